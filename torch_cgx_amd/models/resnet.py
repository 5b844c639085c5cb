"""ResNet-50 (bottleneck v1) for the DDP benchmarks.

Standard architecture (He et al. 2015); implemented here because torchvision
is not available in the target image.  Matches the reference's benchmark
model family (the reference example trains torchvision ResNets,
/root/reference/examples/cifar_train.py).
"""

from __future__ import annotations

import torch
import torch.nn as nn


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes: int, planes: int, stride: int = 1,
                 downsample: nn.Module = None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.conv3 = nn.Conv2d(planes, planes * self.expansion, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * self.expansion)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
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
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)

    def _make_layer(self, planes: int, blocks: int, stride: int = 1):
        downsample = None
        if stride != 1 or self.inplanes != planes * Bottleneck.expansion:
            downsample = nn.Sequential(
                nn.Conv2d(self.inplanes, planes * Bottleneck.expansion, 1,
                          stride=stride, bias=False),
                nn.BatchNorm2d(planes * Bottleneck.expansion),
            )
        layers = [Bottleneck(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * Bottleneck.expansion
        for _ in range(1, blocks):
            layers.append(Bottleneck(self.inplanes, planes))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet50(num_classes: int = 1000) -> ResNet:
    return ResNet([3, 4, 6, 3], num_classes)


def resnet18(num_classes: int = 1000) -> nn.Module:
    """Small variant for tests/examples (basic blocks)."""

    class BasicBlock(nn.Module):
        def __init__(self, inp, planes, stride=1, down=None):
            super().__init__()
            self.c1 = nn.Conv2d(inp, planes, 3, stride, 1, bias=False)
            self.b1 = nn.BatchNorm2d(planes)
            self.c2 = nn.Conv2d(planes, planes, 3, 1, 1, bias=False)
            self.b2 = nn.BatchNorm2d(planes)
            self.down = down

        def forward(self, x):
            idt = x if self.down is None else self.down(x)
            out = torch.relu(self.b1(self.c1(x)))
            return torch.relu(self.b2(self.c2(out)) + idt)

    class R18(nn.Module):
        def __init__(self):
            super().__init__()
            self.stem = nn.Sequential(
                nn.Conv2d(3, 64, 7, 2, 3, bias=False), nn.BatchNorm2d(64),
                nn.ReLU(inplace=True), nn.MaxPool2d(3, 2, 1))
            chans = [64, 128, 256, 512]
            blocks = []
            inp = 64
            for i, c in enumerate(chans):
                stride = 1 if i == 0 else 2
                down = None
                if stride != 1 or inp != c:
                    down = nn.Sequential(nn.Conv2d(inp, c, 1, stride,
                                                   bias=False),
                                         nn.BatchNorm2d(c))
                blocks += [BasicBlock(inp, c, stride, down), ]
                inp = c
                blocks += [BasicBlock(c, c)]
            self.body = nn.Sequential(*blocks)
            self.head = nn.Linear(512, num_classes)

        def forward(self, x):
            x = self.body(self.stem(x))
            return self.head(torch.nn.functional.adaptive_avg_pool2d(
                x, 1).flatten(1))

    return R18()
