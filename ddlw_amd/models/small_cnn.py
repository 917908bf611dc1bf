"""3-layer CNN for the CPU-only baseline config (BASELINE.json config 1:
"3-layer CNN on 64x64 synthetic JPEGs, single-process CPU")."""
from __future__ import annotations

import torch
import torch.nn as nn

from ..core.model_io import tag_model


class SmallCNN(nn.Module):
    def __init__(self, img_height: int = 64, img_width: int = 64, channels: int = 3, num_classes: int = 5):
        super().__init__()
        self.features = nn.Sequential(
            nn.Conv2d(channels, 32, 3, stride=2, padding=1, bias=False),
            nn.BatchNorm2d(32),
            nn.ReLU(inplace=True),
            nn.Conv2d(32, 64, 3, stride=2, padding=1, bias=False),
            nn.BatchNorm2d(64),
            nn.ReLU(inplace=True),
            nn.Conv2d(64, 128, 3, stride=2, padding=1, bias=False),
            nn.BatchNorm2d(128),
            nn.ReLU(inplace=True),
        )
        self.pool = nn.AdaptiveAvgPool2d(1)
        self.classifier = nn.Linear(128, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.features(x)
        x = self.pool(x).flatten(1)
        return self.classifier(x)


def build_small_cnn(img_height: int = 64, img_width: int = 64, channels: int = 3, num_classes: int = 5) -> SmallCNN:
    m = SmallCNN(img_height, img_width, channels, num_classes)
    return tag_model(
        m,
        "small_cnn",
        dict(img_height=img_height, img_width=img_width, channels=channels, num_classes=num_classes),
    )
