"""ResNets: CIFAR-style resnet20..110 (reference VGG/models/resnet.py) and
ImageNet-style resnet18..152 (reference VGG/models/imagenet_resnet.py),
fresh implementations of the standard architectures."""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


def _conv3x3(cin, cout, stride=1):
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, cout, stride=1, downsample=None):
        super().__init__()
        self.conv1 = _conv3x3(cin, cout, stride)
        self.bn1 = nn.BatchNorm2d(cout)
        self.conv2 = _conv3x3(cout, cout)
        self.bn2 = nn.BatchNorm2d(cout)
        self.downsample = downsample

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = F.relu(self.bn1(self.conv1(x)), inplace=True)
        out = self.bn2(self.conv2(out))
        return F.relu(out + identity, inplace=True)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, cout, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(cin, cout, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(cout)
        self.conv2 = nn.Conv2d(cout, cout, 3, stride=stride, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(cout)
        self.conv3 = nn.Conv2d(cout, cout * 4, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(cout * 4)
        self.downsample = downsample

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = F.relu(self.bn1(self.conv1(x)), inplace=True)
        out = F.relu(self.bn2(self.conv2(out)), inplace=True)
        out = self.bn3(self.conv3(out))
        return F.relu(out + identity, inplace=True)


class CifarResNet(nn.Module):
    """depth = 6n+2 basic-block resnet for 32x32 inputs."""

    def __init__(self, depth: int = 20, num_classes: int = 10):
        super().__init__()
        assert (depth - 2) % 6 == 0, "depth must be 6n+2"
        n = (depth - 2) // 6
        self.inplanes = 16
        self.conv1 = _conv3x3(3, 16)
        self.bn1 = nn.BatchNorm2d(16)
        self.layer1 = self._make_layer(16, n)
        self.layer2 = self._make_layer(32, n, stride=2)
        self.layer3 = self._make_layer(64, n, stride=2)
        self.fc = nn.Linear(64, num_classes)

    def _make_layer(self, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes:
            downsample = nn.Sequential(
                nn.Conv2d(self.inplanes, planes, 1, stride=stride, bias=False),
                nn.BatchNorm2d(planes),
            )
        layers = [BasicBlock(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes
        for _ in range(1, blocks):
            layers.append(BasicBlock(planes, planes))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = F.relu(self.bn1(self.conv1(x)), inplace=True)
        x = self.layer3(self.layer2(self.layer1(x)))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


class ImagenetResNet(nn.Module):
    def __init__(self, block, layers, num_classes=1000):
        super().__init__()
        self.inplanes = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.fc = nn.Linear(512 * block.expansion, num_classes)

    def _make_layer(self, block, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                nn.Conv2d(self.inplanes, planes * block.expansion, 1, stride=stride, bias=False),
                nn.BatchNorm2d(planes * block.expansion),
            )
        layers = [block(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * block.expansion
        for _ in range(1, blocks):
            layers.append(block(self.inplanes, planes))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(F.relu(self.bn1(self.conv1(x)), inplace=True))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


def resnet20(**kw):
    return CifarResNet(20, **kw)


def resnet32(**kw):
    return CifarResNet(32, **kw)


def resnet44(**kw):
    return CifarResNet(44, **kw)


def resnet56(**kw):
    return CifarResNet(56, **kw)


def resnet110(**kw):
    return CifarResNet(110, **kw)


def resnet18(**kw):
    return ImagenetResNet(BasicBlock, [2, 2, 2, 2], **kw)


def resnet34(**kw):
    return ImagenetResNet(BasicBlock, [3, 4, 6, 3], **kw)


def resnet50(**kw):
    return ImagenetResNet(Bottleneck, [3, 4, 6, 3], **kw)


def resnet101(**kw):
    return ImagenetResNet(Bottleneck, [3, 4, 23, 3], **kw)


def resnet152(**kw):
    return ImagenetResNet(Bottleneck, [3, 8, 36, 3], **kw)


class _PadChannelDownsample(nn.Module):
    """Parameter-free shortcut: stride-2 subsample + zero-pad channels
    (option-A shortcut of the ResNet paper; reference res_utils.DownsampleA)."""

    def __init__(self, stride=2):
        super().__init__()
        self.pool = nn.AvgPool2d(1, stride=stride)

    def forward(self, x):
        x = self.pool(x)
        return torch.cat([x, x.mul(0.0)], dim=1)


class PreActBlock(nn.Module):
    """Pre-activation basic block (BN-ReLU-conv ordering, He et al. 2016;
    reference preresnet.py behavior: the FIRST block of a stage pre-activates
    the shared input, later blocks keep a plain residual)."""

    def __init__(self, cin, cout, stride=1, downsample=None, both_preact=False):
        super().__init__()
        self.bn1 = nn.BatchNorm2d(cin)
        self.conv1 = _conv3x3(cin, cout, stride)
        self.bn2 = nn.BatchNorm2d(cout)
        self.conv2 = _conv3x3(cout, cout)
        self.downsample = downsample
        self.both_preact = both_preact

    def forward(self, x):
        pre = F.relu(self.bn1(x), inplace=False)
        identity = pre if self.both_preact else x
        if self.downsample is not None:
            identity = self.downsample(identity)
        out = self.conv1(pre)
        out = self.conv2(F.relu(self.bn2(out), inplace=True))
        return out + identity


class CifarPreResNet(nn.Module):
    """Pre-activation 6n+2 resnet for 32x32 (reference preresnet.py)."""

    def __init__(self, depth: int = 20, num_classes: int = 10):
        super().__init__()
        assert (depth - 2) % 6 == 0, "depth must be 6n+2"
        n = (depth - 2) // 6
        self.conv1 = _conv3x3(3, 16)
        self.inplanes = 16
        self.layer1 = self._make_layer(16, n, 1)
        self.layer2 = self._make_layer(32, n, 2)
        self.layer3 = self._make_layer(64, n, 2)
        self.bn = nn.BatchNorm2d(64)
        self.fc = nn.Linear(64, num_classes)

    def _make_layer(self, planes, blocks, stride):
        down = _PadChannelDownsample(stride) if stride != 1 else None
        layers = [PreActBlock(self.inplanes, planes, stride, down,
                              both_preact=True)]
        self.inplanes = planes
        for _ in range(1, blocks):
            layers.append(PreActBlock(planes, planes))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.conv1(x)
        x = self.layer3(self.layer2(self.layer1(x)))
        x = F.relu(self.bn(x), inplace=True)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


class CifarResNetMod(nn.Module):
    """fb.resnet.torch-style cifar resnet (reference resnet_mod.py): same
    6n+2 structure as CifarResNet but with the parameter-free option-A
    zero-pad shortcut instead of a 1x1-conv projection."""

    def __init__(self, depth: int = 20, num_classes: int = 10):
        super().__init__()
        assert (depth - 2) % 6 == 0, "depth must be 6n+2"
        n = (depth - 2) // 6
        self.conv1 = _conv3x3(3, 16)
        self.bn1 = nn.BatchNorm2d(16)
        self.inplanes = 16
        self.layer1 = self._make_layer(16, n, 1)
        self.layer2 = self._make_layer(32, n, 2)
        self.layer3 = self._make_layer(64, n, 2)
        self.fc = nn.Linear(64, num_classes)

    def _make_layer(self, planes, blocks, stride):
        down = _PadChannelDownsample(stride) if stride != 1 else None
        layers = [BasicBlock(self.inplanes, planes, stride, down)]
        self.inplanes = planes
        for _ in range(1, blocks):
            layers.append(BasicBlock(planes, planes))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = F.relu(self.bn1(self.conv1(x)), inplace=True)
        x = self.layer3(self.layer2(self.layer1(x)))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


def preresnet20(**kw):
    return CifarPreResNet(20, **kw)


def preresnet110(**kw):
    return CifarPreResNet(110, **kw)


def resnet_mod20(**kw):
    return CifarResNetMod(20, **kw)


def resnet_mod110(**kw):
    return CifarResNetMod(110, **kw)


def preresnet32(**kw):
    return CifarPreResNet(32, **kw)


def preresnet44(**kw):
    return CifarPreResNet(44, **kw)


def preresnet56(**kw):
    return CifarPreResNet(56, **kw)


def resnet_mod32(**kw):
    return CifarResNetMod(32, **kw)


def resnet_mod44(**kw):
    return CifarResNetMod(44, **kw)


def resnet_mod56(**kw):
    return CifarResNetMod(56, **kw)
