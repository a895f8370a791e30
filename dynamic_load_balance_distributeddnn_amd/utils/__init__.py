from .fault import FaultInjector  # noqa: F401
from .lr_policy import one_cycle_lr  # noqa: F401
from .recorder import StatsRecorder, init_logger  # noqa: F401
