from .space import hp
from .fmin import fmin, tpe, Trials, LocalTrials, STATUS_OK, STATUS_FAIL

__all__ = ["hp", "fmin", "tpe", "Trials", "LocalTrials", "STATUS_OK", "STATUS_FAIL"]
