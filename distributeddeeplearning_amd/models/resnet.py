"""ResNet v1.5 (18/34/50/101/152) built on the framework's fused ops.

Architecture and parameter naming match torchvision's ``resnet*`` (which the
reference PyTorch trainer instantiates —
``HorovodPytorch/src/imagenet_pytorch_horovod.py`` uses
``torchvision.models.resnet50``), so checkpoints interchange directly.
torchvision is NOT a dependency: the model is written in-repo against
``distributeddeeplearning_amd.ops`` (fused conv / BN+add+ReLU kernels on
GPU).  Shape table: SURVEY.md Appendix A.
"""
from __future__ import annotations

from typing import List, Type, Union

import os

import torch.nn as nn

# residual-join BN backward-stats fusion (measured -2% e2e; opt-in)
_FUSE_BN3 = os.environ.get("DDL_FUSE_BN3", "0")

from ..ops import functional as DF
from ..ops.modules import (
    BatchNorm2d,
    Conv2d,
    GlobalAvgPool2d,
    Linear,
    MaxPool2d,
)


def conv3x3(in_planes, out_planes, stride=1):
    return Conv2d(in_planes, out_planes, 3, stride=stride, padding=1,
                  bias=False, emit_bn_stats=True)


def conv1x1(in_planes, out_planes, stride=1):
    return Conv2d(in_planes, out_planes, 1, stride=stride, bias=False,
                  emit_bn_stats=True)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = conv3x3(inplanes, planes, stride)
        # bn1's output has exactly one consumer (conv2): its backward
        # stats come from conv2's dgrad epilogue (BnBwdFusion)
        self.bn1 = BatchNorm2d(planes, fuse_bwd_stats=True)
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = BatchNorm2d(planes, fuse_bwd_stats=_FUSE_BN3 == "1")  # join BN
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        # fork-conv fusion: x gets ONE autograd consumer, so the residual
        # join's gradient add folds into conv1's dgrad epilogue
        ident, out = DF.fork_conv2d(x, self.conv1)
        identity = (self.downsample(ident) if self.downsample is not None
                    else ident)
        out = self.bn1(out, relu=True)
        out = self.bn2(self.conv2(out), relu=True, residual=identity)
        return out


class Bottleneck(nn.Module):
    # ResNet v1.5: stride on the 3x3 (conv2), as in torchvision.
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = conv1x1(inplanes, planes)
        # bn1/bn2 each feed exactly one conv: their backward stats come
        # from that conv's dgrad epilogue (BnBwdFusion)
        self.bn1 = BatchNorm2d(planes, fuse_bwd_stats=True)
        self.conv2 = conv3x3(planes, planes, stride)
        self.bn2 = BatchNorm2d(planes, fuse_bwd_stats=True)
        self.conv3 = conv1x1(planes, planes * self.expansion)
        # bn3's backward-stats CAN fuse into the next block's fork-conv
        # accum epilogue (mask mode; capability tested in
        # test_bnbwd_fusion_gpu), but measured -2% end-to-end: the extra
        # epilogue reads sit on the serial dgrad path while the stats
        # kernels they replace overlapped well.  Default off.
        self.bn3 = BatchNorm2d(planes * self.expansion,
                               fuse_bwd_stats=_FUSE_BN3 == "1")
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        ident, out = DF.fork_conv2d(x, self.conv1)
        identity = (self.downsample(ident) if self.downsample is not None
                    else ident)
        out = self.bn1(out, relu=True)
        out = self.bn2(self.conv2(out), relu=True)
        out = self.bn3(self.conv3(out), relu=True, residual=identity)
        return out


class ResNet(nn.Module):
    def __init__(
        self,
        block: Type[Union[BasicBlock, Bottleneck]],
        layers: List[int],
        num_classes: int = 1000,
        zero_init_residual: bool = False,
    ):
        super().__init__()
        self.inplanes = 64
        self.conv1 = Conv2d(3, 64, 7, stride=2, padding=3, bias=False,
                            emit_bn_stats=True)
        self.bn1 = BatchNorm2d(64)
        self.maxpool = MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = GlobalAvgPool2d()
        self.fc = Linear(512 * block.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, Linear):
                nn.init.normal_(m.weight, 0, 0.01)
                nn.init.zeros_(m.bias)

        # Goyal et al. 2017 large-batch recipe: zero-init the last BN gamma
        # of every residual branch (optional; off by default to match the
        # reference scripts' torchvision defaults).
        if zero_init_residual:
            for m in self.modules():
                if isinstance(m, Bottleneck):
                    nn.init.zeros_(m.bn3.weight)
                elif isinstance(m, BasicBlock):
                    nn.init.zeros_(m.bn2.weight)

    def _make_layer(self, block, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * block.expansion, stride),
                BatchNorm2d(planes * block.expansion),
            )
        layers = [block(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * block.expansion
        layers += [block(self.inplanes, planes) for _ in range(1, blocks)]
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.bn1(self.conv1(x), relu=True)
        x = self.maxpool(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        x = self.avgpool(x)
        return self.fc(x)


def resnet18(num_classes=1000, **kw):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, **kw)


def resnet34(num_classes=1000, **kw):
    return ResNet(BasicBlock, [3, 4, 6, 3], num_classes, **kw)


def resnet50(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, **kw)


def resnet101(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes, **kw)


def resnet152(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 8, 36, 3], num_classes, **kw)
