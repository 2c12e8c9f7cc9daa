"""CIFAR-style CV zoo: resnet56/110, group-norm resnet variant, MobileNet,
DenseNet-121.

Counterparts of the reference fedml_api/model/cv/{resnet.py, resnet_gn.py,
group_normalization.py, mobilenet.py} and torchvision densenet121
(main_fedavg.py:219-220). Standard architectures, random init (no
pretrained weights offline).
"""

from __future__ import annotations

import torch
from torch import nn
import torch.nn.functional as F


def _norm(planes: int, group_norm: bool):
    return nn.GroupNorm(2, planes) if group_norm else nn.BatchNorm2d(planes)


class CifarBasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None,
                 group_norm=False):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, 3, stride, 1, bias=False)
        self.bn1 = _norm(planes, group_norm)
        self.conv2 = nn.Conv2d(planes, planes, 3, 1, 1, bias=False)
        self.bn2 = _norm(planes, group_norm)
        self.downsample = downsample

    def forward(self, x):
        identity = x
        out = F.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return F.relu(out + identity)


class CifarResNet(nn.Module):
    """3-stage CIFAR ResNet (16/32/64 channels); depth = 6n+2."""

    def __init__(self, n_blocks_per_stage: int, num_classes: int = 10,
                 group_norm: bool = False):
        super().__init__()
        self.gn = group_norm
        self.inplanes = 16
        self.conv1 = nn.Conv2d(3, 16, 3, 1, 1, bias=False)
        self.bn1 = _norm(16, group_norm)
        self.layer1 = self._make_layer(16, n_blocks_per_stage, 1)
        self.layer2 = self._make_layer(32, n_blocks_per_stage, 2)
        self.layer3 = self._make_layer(64, n_blocks_per_stage, 2)
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(64, num_classes)

    def _make_layer(self, planes, blocks, stride):
        downsample = None
        if stride != 1 or self.inplanes != planes:
            downsample = nn.Sequential(
                nn.Conv2d(self.inplanes, planes, 1, stride, bias=False),
                _norm(planes, self.gn))
        layers = [CifarBasicBlock(self.inplanes, planes, stride, downsample,
                                  self.gn)]
        self.inplanes = planes
        for _ in range(1, blocks):
            layers.append(CifarBasicBlock(planes, planes,
                                          group_norm=self.gn))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = F.relu(self.bn1(self.conv1(x)))
        x = self.layer3(self.layer2(self.layer1(x)))
        return self.fc(self.avgpool(x).flatten(1))


def resnet56(num_classes: int = 10, group_norm: bool = False):
    return CifarResNet(9, num_classes, group_norm)


def resnet110(num_classes: int = 10, group_norm: bool = False):
    return CifarResNet(18, num_classes, group_norm)


class _DWSep(nn.Module):
    def __init__(self, inp, out, stride):
        super().__init__()
        self.dw = nn.Conv2d(inp, inp, 3, stride, 1, groups=inp, bias=False)
        self.bn1 = nn.BatchNorm2d(inp)
        self.pw = nn.Conv2d(inp, out, 1, 1, 0, bias=False)
        self.bn2 = nn.BatchNorm2d(out)

    def forward(self, x):
        x = F.relu(self.bn1(self.dw(x)))
        return F.relu(self.bn2(self.pw(x)))


class MobileNet(nn.Module):
    """MobileNet v1 (depthwise-separable stacks)."""

    CFG = [(64, 1), (128, 2), (128, 1), (256, 2), (256, 1), (512, 2),
           (512, 1), (512, 1), (512, 1), (512, 1), (512, 1), (1024, 2),
           (1024, 1)]

    def __init__(self, num_classes: int = 1000):
        super().__init__()
        self.stem = nn.Sequential(nn.Conv2d(3, 32, 3, 2, 1, bias=False),
                                  nn.BatchNorm2d(32), nn.ReLU(inplace=True))
        layers = []
        inp = 32
        for out, stride in self.CFG:
            layers.append(_DWSep(inp, out, stride))
            inp = out
        self.features = nn.Sequential(*layers)
        self.pool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(1024, num_classes)

    def forward(self, x):
        x = self.features(self.stem(x))
        return self.fc(self.pool(x).flatten(1))


class _DenseLayer(nn.Module):
    def __init__(self, inp, growth, bn_size=4):
        super().__init__()
        self.norm1 = nn.BatchNorm2d(inp)
        self.conv1 = nn.Conv2d(inp, bn_size * growth, 1, bias=False)
        self.norm2 = nn.BatchNorm2d(bn_size * growth)
        self.conv2 = nn.Conv2d(bn_size * growth, growth, 3, padding=1,
                               bias=False)

    def forward(self, x):
        out = self.conv1(F.relu(self.norm1(x)))
        out = self.conv2(F.relu(self.norm2(out)))
        return torch.cat([x, out], 1)


class _Transition(nn.Module):
    def __init__(self, inp, out):
        super().__init__()
        self.norm = nn.BatchNorm2d(inp)
        self.conv = nn.Conv2d(inp, out, 1, bias=False)

    def forward(self, x):
        return F.avg_pool2d(self.conv(F.relu(self.norm(x))), 2)


class DenseNet(nn.Module):
    def __init__(self, block_config=(6, 12, 24, 16), growth=32,
                 num_init=64, num_classes: int = 1000):
        super().__init__()
        layers = [nn.Conv2d(3, num_init, 7, 2, 3, bias=False),
                  nn.BatchNorm2d(num_init), nn.ReLU(inplace=True),
                  nn.MaxPool2d(3, 2, 1)]
        ch = num_init
        for bi, n in enumerate(block_config):
            for _ in range(n):
                layers.append(_DenseLayer(ch, growth))
                ch += growth
            if bi != len(block_config) - 1:
                layers.append(_Transition(ch, ch // 2))
                ch //= 2
        layers += [nn.BatchNorm2d(ch), nn.ReLU(inplace=True)]
        self.features = nn.Sequential(*layers)
        self.classifier = nn.Linear(ch, num_classes)

    def forward(self, x):
        x = self.features(x)
        x = F.adaptive_avg_pool2d(x, (1, 1)).flatten(1)
        return self.classifier(x)


def densenet121(num_classes: int = 1000):
    return DenseNet((6, 12, 24, 16), 32, 64, num_classes)


class CifarBottleneck(nn.Module):
    """Bottleneck block (1x1 -> 3x3 -> 1x1, expansion 4) — the reference
    GKT server model is built from these
    (resnet56_gkt/resnet_server.py:70-110)."""
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.conv3 = nn.Conv2d(planes, planes * 4, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * 4)
        self.downsample = downsample

    def forward(self, x):
        identity = x
        out = F.relu(self.bn1(self.conv1(x)))
        out = F.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return F.relu(out + identity)


class GKTClientNet(nn.Module):
    """FedGKT edge model at the reference's split
    (resnet56_gkt/resnet_client.py:112-216, resnet5_56): 3x3 stem to 16
    channels (the EXTRACTED FEATURES that ship to the server), one local
    16-channel stage, and a local classifier head for the client-side
    CE + distillation loss."""

    def __init__(self, num_classes: int = 10, n_blocks: int = 1):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 16, 3, 1, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(16)
        self.layer1 = nn.Sequential(*[
            CifarBasicBlock(16, 16) for _ in range(n_blocks)])
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(16, num_classes)

    def forward(self, x):
        feats = F.relu(self.bn1(self.conv1(x)))   # B x 16 x H x W
        h = self.layer1(feats)
        logits = self.fc(self.avgpool(h).flatten(1))
        return logits, feats


class GKTServerNet(nn.Module):
    """FedGKT server model consuming the 16-channel client features
    (resnet56_gkt/resnet_server.py:185-208, resnet56_server =
    Bottleneck [6,6,6]): three bottleneck stages 16 -> 32 -> 64 with
    stride-2 reductions, then the classifier."""

    def __init__(self, num_classes: int = 10, n_blocks: int = 6):
        super().__init__()
        self.inplanes = 16

        def stage(planes, blocks, stride):
            down = None
            if stride != 1 or self.inplanes != planes * 4:
                down = nn.Sequential(
                    nn.Conv2d(self.inplanes, planes * 4, 1, stride,
                              bias=False),
                    nn.BatchNorm2d(planes * 4))
            layers = [CifarBottleneck(self.inplanes, planes, stride, down)]
            self.inplanes = planes * 4
            layers += [CifarBottleneck(self.inplanes, planes)
                       for _ in range(1, blocks)]
            return nn.Sequential(*layers)

        self.layer1 = stage(16, n_blocks, 1)
        self.layer2 = stage(32, n_blocks, 2)
        self.layer3 = stage(64, n_blocks, 2)
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(64 * 4, num_classes)

    def forward(self, feats):
        x = self.layer3(self.layer2(self.layer1(feats)))
        return self.fc(self.avgpool(x).flatten(1))
