from .small_cnn import SmallCNN, build_small_cnn
from .resnet import ResNet50, build_resnet50
from .mobilenet_v2 import MobileNetV2, build_model
from ..core.model_io import register_builder

register_builder("small_cnn", build_small_cnn)
register_builder("resnet50", build_resnet50)
register_builder("mobilenet_v2_head", build_model)

__all__ = [
    "SmallCNN",
    "build_small_cnn",
    "ResNet50",
    "build_resnet50",
    "MobileNetV2",
    "build_model",
]
