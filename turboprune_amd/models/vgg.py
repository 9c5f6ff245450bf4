"""VGG family with masked layers (torchvision-compatible naming:
``features.N`` / ``classifier.N``).

The reference reaches VGG via torchvision + layer surgery; CIFAR variant
swaps the classifier for a single Linear (reference:
custom_models.py:205-213). Convs are ConvMask; linears are Conv1dMask.
"""

from __future__ import annotations

from typing import List, Union

import torch
import torch.nn as nn

from turboprune_amd.ops.bn import FusedBatchNorm2d
from turboprune_amd.ops.pool import FusedMaxPool2d
from turboprune_amd.ops.mask_layers import Conv1dMask, ConvMask

_CFGS = {
    "vgg11": [64, "M", 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "vgg16": [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
              512, 512, 512, "M", 512, 512, 512, "M"],
    "vgg19": [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M",
              512, 512, 512, 512, "M", 512, 512, 512, 512, "M"],
}


def _make_features(cfg: List[Union[int, str]], batch_norm: bool) -> nn.Sequential:
    layers: List[nn.Module] = []
    in_ch = 3
    for v in cfg:
        if v == "M":
            layers.append(FusedMaxPool2d(kernel_size=2, stride=2))
        else:
            layers.append(ConvMask(in_channels=in_ch, out_channels=int(v),
                                   kernel_size=3, padding=1, bias=True))
            if batch_norm:
                layers.append(FusedBatchNorm2d(int(v)))
            layers.append(nn.ReLU(inplace=True))
            in_ch = int(v)
    return nn.Sequential(*layers)


class VGG(nn.Module):
    def __init__(self, cfg_name: str, batch_norm: bool = True,
                 num_classes: int = 1000, cifar_stem: bool = False):
        super().__init__()
        self.features = _make_features(_CFGS[cfg_name], batch_norm)
        if cifar_stem:
            # CIFAR: single masked-linear classifier on the 512-dim feature
            self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
            self.classifier = Conv1dMask(512, num_classes, bias=True)
        else:
            self.avgpool = nn.AdaptiveAvgPool2d((7, 7))
            self.classifier = nn.Sequential(
                Conv1dMask(512 * 7 * 7, 4096, bias=True),
                nn.ReLU(inplace=True),
                nn.Dropout(0.5),
                Conv1dMask(4096, 4096, bias=True),
                nn.ReLU(inplace=True),
                nn.Dropout(0.5),
                Conv1dMask(4096, num_classes, bias=True),
            )
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
                if m.bias is not None:
                    nn.init.constant_(m.bias, 0)
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.features(x)
        x = self.avgpool(x)
        x = torch.flatten(x, 1)
        return self.classifier(x)


def vgg11_bn(num_classes=1000, cifar_stem=False):
    return VGG("vgg11", True, num_classes, cifar_stem)


def vgg16_bn(num_classes=1000, cifar_stem=False):
    return VGG("vgg16", True, num_classes, cifar_stem)


def vgg19_bn(num_classes=1000, cifar_stem=False):
    return VGG("vgg19", True, num_classes, cifar_stem)
