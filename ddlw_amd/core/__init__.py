from .config import DataCfg, TrainCfg, setup
from . import tracking

__all__ = ["DataCfg", "TrainCfg", "setup", "tracking"]
