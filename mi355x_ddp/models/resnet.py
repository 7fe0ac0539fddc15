"""ResNet-50, defined locally (torchvision is not installed in this
environment — SURVEY §7 environment facts).

The profiling entrypoint (reference multigpu_profile.py:14,25 imports
torchvision's resnet50) needs a real convolutional workload whose DDP
gradient payload is ~102 MB across many buckets. Conv/BN/pool
run through MIOpen via PyTorch-ROCm (SURVEY §2.2 N11 — re-writing ~50 conv
shapes by hand is out of scope; the profile stage's purpose is trace
capture and multi-bucket reduction); the final FC layer uses the
hand-written MFMA linear.

Standard bottleneck architecture (He et al. 2015), v1.5 variant (stride on
the 3x3 conv) to match what torchvision's resnet50 computes.
"""

from __future__ import annotations

import torch
from torch import nn

from .toy import HipLinear


def conv3x3(cin: int, cout: int, stride: int = 1) -> nn.Conv2d:
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


def conv1x1(cin: int, cout: int, stride: int = 1) -> nn.Conv2d:
    return nn.Conv2d(cin, cout, 1, stride=stride, bias=False)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin: int, width: int, stride: int = 1,
                 downsample: nn.Module = None):
        super().__init__()
        self.conv1 = conv1x1(cin, width)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = conv3x3(width, width, stride)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = conv1x1(width, width * self.expansion)
        self.bn3 = nn.BatchNorm2d(width * self.expansion)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.relu(out + identity)


class ResNet(nn.Module):
    def __init__(self, layers, num_classes: int = 1000):
        super().__init__()
        self.inplanes = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, layers[0])
        self.layer2 = self._make_layer(128, layers[1], stride=2)
        self.layer3 = self._make_layer(256, layers[2], stride=2)
        self.layer4 = self._make_layer(512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = HipLinear(512 * Bottleneck.expansion, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)

    def _make_layer(self, width: int, blocks: int, stride: int = 1):
        downsample = None
        cout = width * Bottleneck.expansion
        if stride != 1 or self.inplanes != cout:
            downsample = nn.Sequential(conv1x1(self.inplanes, cout, stride),
                                       nn.BatchNorm2d(cout))
        layers = [Bottleneck(self.inplanes, width, stride, downsample)]
        self.inplanes = cout
        layers += [Bottleneck(cout, width) for _ in range(1, blocks)]
        return nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        x = torch.flatten(self.avgpool(x), 1)
        return self.fc(x)


def resnet50(num_classes: int = 1000) -> ResNet:
    return ResNet([3, 4, 6, 3], num_classes)
