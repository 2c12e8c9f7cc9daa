"""ResNet-18 (standard BasicBlock architecture, batch-norm).

The reference uses torchvision.models.resnet18(pretrained=True) for the
FMoW drift config (main_fedavg.py:221-222); torchvision and pretrained
checkpoints are unavailable offline, so this is the same architecture with
random initialization (the benchmark contract allows random-init weights).
State-dict key layout matches torchvision's so checkpoints interoperate.
"""

from __future__ import annotations

import torch
from torch import nn


def conv3x3(inp, out, stride=1):
    return nn.Conv2d(inp, out, 3, stride=stride, padding=1, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = conv3x3(inplanes, planes, stride)
        self.bn1 = nn.BatchNorm2d(planes)
        self.relu = nn.ReLU(inplace=True)
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = nn.BatchNorm2d(planes)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        identity = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.relu(out + identity)


class ResNet(nn.Module):
    def __init__(self, layers, num_classes=1000):
        super().__init__()
        self.inplanes = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, layers[0])
        self.layer2 = self._make_layer(128, layers[1], 2)
        self.layer3 = self._make_layer(256, layers[2], 2)
        self.layer4 = self._make_layer(512, layers[3], 2)
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(512, num_classes)

    def _make_layer(self, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes:
            downsample = nn.Sequential(
                nn.Conv2d(self.inplanes, planes, 1, stride=stride,
                          bias=False),
                nn.BatchNorm2d(planes))
        layers = [BasicBlock(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes
        for _ in range(1, blocks):
            layers.append(BasicBlock(self.inplanes, planes))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet18(num_classes=1000):
    return ResNet([2, 2, 2, 2], num_classes=num_classes)


class FlatImageModel(nn.Module):
    """Adapter: the drift data layer stores rows as flat feature vectors
    (the reference batches CSV rows the same way); this reshapes them to
    images for convolutional backbones."""

    def __init__(self, backbone: nn.Module, shape):
        super().__init__()
        self.backbone = backbone
        self.shape = tuple(shape)

    def forward(self, x):
        return self.backbone(x.reshape(-1, *self.shape))
