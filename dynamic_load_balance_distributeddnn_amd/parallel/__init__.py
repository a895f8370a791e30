from .grad_sync import GradientSynchronizer  # noqa: F401
from .timing import StepTimer  # noqa: F401
