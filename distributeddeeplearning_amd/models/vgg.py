"""VGG-16 (configuration D), torchvision-compatible state dict.

BASELINE config 5: VGG-16 bf16 DP=8 — the bandwidth-bound, 138 M-parameter
gradient-fusion stress test (277 MB bf16 grads per step over xGMI;
SURVEY.md §5.8).  Conv+bias+ReLU runs as one fused kernel (the ReLU slots in
``features`` are Identity pass-throughs so the Sequential indices — and
therefore parameter names like ``features.0.weight`` — match torchvision).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops.modules import Conv2d, Dropout, Linear, MaxPool2d
from ..ops import functional as DF

_VGG16_D = [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
            512, 512, 512, "M", 512, 512, 512, "M"]


class _FusedConvReLU(Conv2d):
    """Conv2d whose kernel epilogue applies bias + ReLU (K1 epilogue)."""

    def forward(self, x):
        return DF.conv2d(x, self.weight, self.bias, self.stride, self.padding,
                         relu=True)


class VGG(nn.Module):
    def __init__(self, cfg=_VGG16_D, num_classes=1000, dropout=0.5):
        super().__init__()
        layers: list[nn.Module] = []
        in_ch = 3
        for v in cfg:
            if v == "M":
                layers.append(MaxPool2d(2, stride=2))
            else:
                layers.append(_FusedConvReLU(in_ch, v, 3, padding=1, bias=True))
                layers.append(nn.Identity())  # ReLU fused into the conv above
                in_ch = v
        self.features = nn.Sequential(*layers)
        # torchvision applies AdaptiveAvgPool2d(7) between features and
        # classifier; at 224x224 input features already end at 7x7 so it is
        # the identity — kept as flatten only.
        self.classifier = nn.Sequential(
            Linear(512 * 7 * 7, 4096), nn.ReLU(inplace=True), Dropout(dropout),
            Linear(4096, 4096), nn.ReLU(inplace=True), Dropout(dropout),
            Linear(4096, num_classes),
        )
        for m in self.modules():
            if isinstance(m, Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, Linear):
                nn.init.normal_(m.weight, 0, 0.01)
                nn.init.zeros_(m.bias)

    def forward(self, x):
        x = self.features(x)
        x = torch.flatten(x.contiguous(memory_format=torch.contiguous_format), 1)
        return self.classifier(x)


def vgg16(num_classes=1000, **kw):
    return VGG(_VGG16_D, num_classes, **kw)
