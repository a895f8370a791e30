from .fault import FaultInjector  # noqa: F401
from .lr_policy import one_cycle_lr  # noqa: F401
from .recorder import StatsRecorder, init_logger  # noqa: F401


def print_layer(model, layer_name):
    """Return a named parameter (reference utils.py:1-4 debug helper)."""
    for name, param in model.named_parameters():
        if name == layer_name:
            return param
    return None
