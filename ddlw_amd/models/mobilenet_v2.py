"""MobileNetV2 + transfer-learning head — the reference's ``build_model()``.

Reference contract (``Part 1 .../02_model_training_single_node.py:159-178``):
``build_model(img_height, img_width, img_channels, num_classes, dropout)`` ->
frozen MobileNetV2 base (include_top=False) -> GlobalAveragePooling ->
Dropout -> Dense(num_classes) producing *logits*.

Faithfulness notes: the base is frozen layer by
layer, so its BatchNorms must run with *moving statistics* even in training
mode — here the base is put in eval() and its params have
``requires_grad=False``; ``FrozenBase.train()`` keeps it in eval mode.

No pretrained weights exist in this offline environment, so the base is
randomly initialised (documented deviation; the benchmark configs use
random-init weights anyway, BASELINE.json).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..core.model_io import tag_model
from ..ops.conv import Conv2d
from ..ops.layers import BatchNormAct2d, GlobalAvgPool2d


class InvertedResidual(nn.Module):
    def __init__(self, in_ch: int, out_ch: int, stride: int, expand: int):
        super().__init__()
        hidden = in_ch * expand
        self.use_res = stride == 1 and in_ch == out_ch
        layers = []
        if expand != 1:
            layers += [
                Conv2d(in_ch, hidden, 1, bias=False),
                BatchNormAct2d(hidden),
                nn.ReLU6(inplace=True),
            ]
        layers += [
            Conv2d(hidden, hidden, 3, stride=stride, padding=1, groups=hidden, bias=False),
            BatchNormAct2d(hidden),
            nn.ReLU6(inplace=True),
            Conv2d(hidden, out_ch, 1, bias=False),
            BatchNormAct2d(out_ch),
        ]
        self.conv = nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x + self.conv(x) if self.use_res else self.conv(x)


# (expand t, out channels c, repeats n, first stride s) — standard MobileNetV2
_V2_CFG = [
    (1, 16, 1, 1),
    (6, 24, 2, 2),
    (6, 32, 3, 2),
    (6, 64, 4, 2),
    (6, 96, 3, 1),
    (6, 160, 3, 2),
    (6, 320, 1, 1),
]


class MobileNetV2(nn.Module):
    """Feature extractor (include_top=False): output B x 1280 x H/32 x W/32."""

    def __init__(self, channels: int = 3):
        super().__init__()
        blocks = [
            Conv2d(channels, 32, 3, stride=2, padding=1, bias=False),
            BatchNormAct2d(32),
            nn.ReLU6(inplace=True),
        ]
        in_ch = 32
        for t, c, n, s in _V2_CFG:
            for i in range(n):
                blocks.append(InvertedResidual(in_ch, c, s if i == 0 else 1, t))
                in_ch = c
        blocks += [
            Conv2d(in_ch, 1280, 1, bias=False),
            BatchNormAct2d(1280),
            nn.ReLU6(inplace=True),
        ]
        self.features = nn.Sequential(*blocks)
        for m in self.modules():
            if isinstance(m, (nn.Conv2d, Conv2d)):
                nn.init.kaiming_normal_(m.weight, mode="fan_out")
            elif isinstance(m, (nn.BatchNorm2d, BatchNormAct2d)):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.features(x)


class FrozenBase(nn.Module):
    """Wraps a base so it stays in eval mode (frozen BN -> moving stats,
    reference ``.../02_model_training_single_node.py:167-169``)."""

    def __init__(self, base: nn.Module):
        super().__init__()
        self.base = base
        for p in self.base.parameters():
            p.requires_grad = False
        self.base.eval()

    def train(self, mode: bool = True) -> "FrozenBase":
        # stay frozen: never switch the base to train-mode BN
        super().train(mode)
        self.base.eval()
        return self

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        with torch.no_grad():
            return self.base(x)


class TransferModel(nn.Module):
    """Frozen base -> GAP -> Dropout -> Dense(num_classes) logits
    (reference ``.../02_model_training_single_node.py:171-176``)."""

    def __init__(self, channels: int = 3, num_classes: int = 5, dropout: float = 0.5):
        super().__init__()
        self.base = FrozenBase(MobileNetV2(channels))
        self.global_average_pooling = GlobalAvgPool2d()
        from ..ops.layers import Dropout as DdlwDropout

        self.dropout = DdlwDropout(dropout)  # K7 kernel on GPU, stock on CPU
        self.classifier = nn.Linear(1280, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.base(x)
        x = self.global_average_pooling(x)  # fused GAP -> (N, C)
        x = self.dropout(x)
        return self.classifier(x.to(self.classifier.weight.dtype))


def build_model(
    img_height: int = 224,
    img_width: int = 224,
    img_channels: int = 3,
    num_classes: int = 5,
    dropout: float = 0.5,
) -> TransferModel:
    m = TransferModel(channels=img_channels, num_classes=num_classes, dropout=dropout)
    return tag_model(
        m,
        "mobilenet_v2_head",
        dict(
            img_height=img_height,
            img_width=img_width,
            img_channels=img_channels,
            num_classes=num_classes,
            dropout=dropout,
        ),
    )
