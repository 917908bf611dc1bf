from .model import Model, History, autolog
from .callbacks import (
    Callback,
    ModelCheckpoint,
    EarlyStopping,
    ReduceLROnPlateau,
    LearningRateWarmupCallback,
    BroadcastGlobalVariablesCallback,
    MetricAverageCallback,
)

__all__ = [
    "Model",
    "History",
    "autolog",
    "Callback",
    "ModelCheckpoint",
    "EarlyStopping",
    "ReduceLROnPlateau",
    "LearningRateWarmupCallback",
    "BroadcastGlobalVariablesCallback",
    "MetricAverageCallback",
]
