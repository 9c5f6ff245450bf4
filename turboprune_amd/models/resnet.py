"""ResNet family, built directly with masked layers.

The reference constructs torchvision ResNets and then rewrites every
nn.Conv2d/nn.Linear into masked layers by module surgery (reference:
utils/custom_models.py:64-107,176-220). Since torchvision is not part of
this stack, the architectures are implemented here natively — with the
*same module naming scheme* (``conv1``, ``bn1``, ``layer{1..4}.{i}.conv{j}``,
``downsample.0/1``, ``fc``) so state_dicts/checkpoints are interchangeable
with torchvision-derived ones.

Convolutions are ``ConvMask``; the classifier is ``Conv1dMask``
(linear-as-1x1-conv), matching the reference's replacement map
(custom_models.py:219: nn.Linear -> Conv1dMask, nn.Conv2d -> ConvMask).

CIFAR stem surgery (3x3 stride-1 conv1, no maxpool) mirrors
custom_models.py:197-215.
"""

from __future__ import annotations

from typing import List, Optional, Type, Union

import torch
import torch.nn as nn

from turboprune_amd.ops.bn import FusedBatchNorm2d, bn_act
from turboprune_amd.ops.pool import FusedMaxPool2d
from turboprune_amd.ops.mask_layers import Conv1dMask, ConvMask


def conv3x3(in_planes: int, out_planes: int, stride: int = 1,
            groups: int = 1, dilation: int = 1) -> ConvMask:
    return ConvMask(in_channels=in_planes, out_channels=out_planes,
                    kernel_size=3, stride=stride, padding=dilation,
                    groups=groups, bias=False, dilation=dilation)


def conv1x1(in_planes: int, out_planes: int, stride: int = 1) -> ConvMask:
    return ConvMask(in_channels=in_planes, out_channels=out_planes,
                    kernel_size=1, stride=stride, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes: int, planes: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None):
        super().__init__()
        self.conv1 = conv3x3(inplanes, planes, stride)
        self.bn1 = FusedBatchNorm2d(planes)
        self.relu = nn.ReLU(inplace=True)
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = FusedBatchNorm2d(planes)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x
        # fused BN+ReLU and BN+add+ReLU epilogues (ops/bn.py)
        out = bn_act(self.bn1, self.conv1(x), relu=True)
        out = self.conv2(out)
        if self.downsample is not None:
            identity = self.downsample(x)
        return bn_act(self.bn2, out, residual=identity, relu=True)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes: int, planes: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None,
                 base_width: int = 64):
        super().__init__()
        width = int(planes * (base_width / 64.0))
        self.conv1 = conv1x1(inplanes, width)
        self.bn1 = FusedBatchNorm2d(width)
        self.conv2 = conv3x3(width, width, stride)
        self.bn2 = FusedBatchNorm2d(width)
        self.conv3 = conv1x1(width, planes * self.expansion)
        self.bn3 = FusedBatchNorm2d(planes * self.expansion)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x
        out = bn_act(self.bn1, self.conv1(x), relu=True)
        out = bn_act(self.bn2, self.conv2(out), relu=True)
        out = self.conv3(out)
        if self.downsample is not None:
            identity = self.downsample(x)
        return bn_act(self.bn3, out, residual=identity, relu=True)


class ResNet(nn.Module):
    def __init__(self, block: Type[Union[BasicBlock, Bottleneck]],
                 layers: List[int], num_classes: int = 1000,
                 cifar_stem: bool = False, width_per_group: int = 64):
        super().__init__()
        self.inplanes = 64
        self.base_width = width_per_group
        if cifar_stem:
            # CIFAR surgery: 3x3 stride-1 stem, no maxpool
            # (reference: custom_models.py:197-215)
            self.conv1 = ConvMask(in_channels=3, out_channels=64,
                                  kernel_size=3, stride=1, padding=1,
                                  bias=False)
            self.maxpool = nn.Identity()
        else:
            self.conv1 = ConvMask(in_channels=3, out_channels=64,
                                  kernel_size=7, stride=2, padding=3,
                                  bias=False)
            self.maxpool = FusedMaxPool2d(kernel_size=3, stride=2, padding=1)
        self.bn1 = FusedBatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = Conv1dMask(512 * block.expansion, num_classes, bias=True)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)

    def _make_layer(self, block, planes: int, blocks: int,
                    stride: int = 1) -> nn.Sequential:
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * block.expansion, stride),
                FusedBatchNorm2d(planes * block.expansion),
            )
        kw = {} if block is BasicBlock else {"base_width": self.base_width}
        layers = [block(self.inplanes, planes, stride, downsample, **kw)]
        self.inplanes = planes * block.expansion
        layers += [block(self.inplanes, planes, **kw)
                   for _ in range(1, blocks)]
        return nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.maxpool(bn_act(self.bn1, self.conv1(x), relu=True))
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        x = self.avgpool(x)
        x = torch.flatten(x, 1)
        return self.fc(x)


def resnet18(num_classes: int = 1000, cifar_stem: bool = False) -> ResNet:
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, cifar_stem)


def resnet34(num_classes: int = 1000, cifar_stem: bool = False) -> ResNet:
    return ResNet(BasicBlock, [3, 4, 6, 3], num_classes, cifar_stem)


def resnet50(num_classes: int = 1000, cifar_stem: bool = False) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, cifar_stem)


def resnet101(num_classes: int = 1000, cifar_stem: bool = False) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes, cifar_stem)


def resnet152(num_classes: int = 1000, cifar_stem: bool = False) -> ResNet:
    return ResNet(Bottleneck, [3, 8, 36, 3], num_classes, cifar_stem)


def wide_resnet50_2(num_classes: int = 1000,
                    cifar_stem: bool = False) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, cifar_stem,
                  width_per_group=128)


def wide_resnet101_2(num_classes: int = 1000,
                     cifar_stem: bool = False) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes, cifar_stem,
                  width_per_group=128)
