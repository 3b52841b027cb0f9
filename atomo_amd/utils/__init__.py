from .flat import flatten_params, grads_of
from .timers import PhaseTimers
from .metrics import accuracy

__all__ = ["flatten_params", "grads_of", "PhaseTimers", "accuracy"]
