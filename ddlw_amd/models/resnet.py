"""ResNet-50 — the BASELINE benchmark model (BASELINE.json configs 2-5).

Written from scratch (no torchvision in this environment). Standard v1.5
bottleneck architecture (stride-2 in the 3x3 conv of downsampling blocks),
because that is what "ResNet-50 images/sec" conventionally measures.

Built from ``ddlw_amd.ops.layers`` blocks: BatchNorm+ReLU (and the block-end
BatchNorm+residual-add+ReLU) are ONE fused hand-written CDNA4 HIP kernel each
way when running bf16/channels_last on gfx950, as are the stem maxpool and
the global average pool; on CPU the same layers fall back to stock PyTorch
fp32 ops, which doubles as the numerics oracle for the kernel parity tests
(SURVEY.md §4 item 2). Convolutions use MIOpen or the ddlw MFMA
implicit-GEMM kernels depending on the ops dispatch setting.
"""
from __future__ import annotations

from typing import List

import os

import torch
import torch.nn as nn

from ..core.model_io import tag_model
from ..ops.layers import BatchNormAct2d, GlobalAvgPool2d, MaxPool3x3s2
from ..ops.conv import Conv2d


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch: int, mid_ch: int, stride: int = 1, downsample: nn.Module = None):
        super().__init__()
        self.conv1 = Conv2d(in_ch, mid_ch, 1, bias=False)
        self.bn1 = BatchNormAct2d(mid_ch, relu=True)
        self.conv2 = Conv2d(mid_ch, mid_ch, 3, stride=stride, padding=1, bias=False)
        self.bn2 = BatchNormAct2d(mid_ch, relu=True)
        self.conv3 = Conv2d(mid_ch, mid_ch * self.expansion, 1, bias=False)
        # bn3 + residual add + relu fused into one kernel
        self.bn3 = BatchNormAct2d(mid_ch * self.expansion, relu=True)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from ..ops import block as _block

        if _block.bottleneck_fusable(self, x):
            # whole-block fused Function: the residual-join gradient add is
            # absorbed into conv1's dgrad epilogue (see ops/block.py)
            return _block.bottleneck_forward(self, x)
        if _block.bottleneck_eval_fusable(self, x):
            # eval: BN folded into the conv epilogues (4 kernels per block)
            return _block.bottleneck_eval_forward(self, x)
        identity = self.downsample(x) if self.downsample is not None else x
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), residual=identity)


class Downsample(nn.Module):
    def __init__(self, in_ch: int, out_ch: int, stride: int):
        super().__init__()
        self.conv = Conv2d(in_ch, out_ch, 1, stride=stride, bias=False)
        self.bn = BatchNormAct2d(out_ch, relu=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.bn(self.conv(x))


class ResNet50(nn.Module):
    def __init__(self, num_classes: int = 1000, channels: int = 3):
        super().__init__()
        layers = [3, 4, 6, 3]
        self.in_ch = 64
        self.conv1 = Conv2d(channels, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = BatchNormAct2d(64, relu=True)
        self.maxpool = MaxPool3x3s2()
        self.layer1 = self._make_layer(64, layers[0], stride=1)
        self.layer2 = self._make_layer(128, layers[1], stride=2)
        self.layer3 = self._make_layer(256, layers[2], stride=2)
        self.layer4 = self._make_layer(512, layers[3], stride=2)
        self.avgpool = GlobalAvgPool2d()
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)
        self._init_weights()

    def _make_layer(self, mid_ch: int, blocks: int, stride: int) -> nn.Sequential:
        downsample = None
        out_ch = mid_ch * Bottleneck.expansion
        if stride != 1 or self.in_ch != out_ch:
            downsample = Downsample(self.in_ch, out_ch, stride)
        layer: List[nn.Module] = [Bottleneck(self.in_ch, mid_ch, stride, downsample)]
        self.in_ch = out_ch
        for _ in range(1, blocks):
            layer.append(Bottleneck(out_ch, mid_ch))
        return nn.Sequential(*layer)

    def _init_weights(self) -> None:
        for m in self.modules():
            if isinstance(m, (nn.Conv2d, Conv2d)):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, BatchNormAct2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
        # zero-init the last BN of each block (standard ResNet-50 recipe)
        for m in self.modules():
            if isinstance(m, Bottleneck):
                nn.init.zeros_(m.bn3.weight)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from ..ops import block as _block

        if (not self.training and not torch.is_grad_enabled() and x.is_cuda
                and x.dtype == torch.bfloat16
                and os.environ.get("DDLW_EVAL_FOLD", "1") == "1"
                and os.environ.get("DDLW_STEM", "1") == "1"
                and os.environ.get("DDLW_DISABLE_HIP_OPS", "0") != "1"
                and self.conv1.in_channels == 3):
            from ..ops import conv_gemm as _cg

            # eval: stem conv + folded bn1 + relu in one kernel
            x = _cg.stem_fwd_kernel(
                x.contiguous(memory_format=torch.channels_last),
                self.conv1.weight,
                ep=(*self.bn1.folded_scale_bias(), True))
        else:
            x = self.bn1(self.conv1(x))
        x = self.maxpool(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        x = self.avgpool(x)
        return self.fc(x)


def build_resnet50(num_classes: int = 1000, channels: int = 3) -> ResNet50:
    m = ResNet50(num_classes=num_classes, channels=channels)
    return tag_model(m, "resnet50", dict(num_classes=num_classes, channels=channels))
