"""ResNet family (CIFAR and ImageNet variants), self-contained.

Used by the acceptance workloads (reference: examples/pytorch-cifar/models/
resnet.py and the ResNet-50 elastic-rescale config of BASELINE.json).
torchvision is not a dependency; these are standard ResNet definitions.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from adaptdl_amd.torch.layers import (FusedBatchNormAct2d,
                                      FusedConv2d)


def _bn(planes, relu=False):
    """BatchNorm2d with optional fused ReLU (CDNA4 kernel on GPU)."""
    return FusedBatchNormAct2d(planes, relu=relu)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_planes, planes, stride=1):
        super().__init__()
        self.conv1 = FusedConv2d(in_planes, planes, 3, stride=stride,
                                 padding=1, bias=False)
        self.bn1 = _bn(planes, relu=True)
        self.conv2 = FusedConv2d(planes, planes, 3, stride=1, padding=1,
                                 bias=False)
        # bn2 fuses the residual add + final ReLU of the block.
        self.bn2 = _bn(planes, relu=True)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_planes != self.expansion * planes:
            self.shortcut = nn.Sequential(
                FusedConv2d(in_planes, self.expansion * planes, 1,
                            stride=stride, bias=False),
                _bn(self.expansion * planes))

    def forward(self, x):
        out = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(out), residual=self.shortcut(x))


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_planes, planes, stride=1):
        super().__init__()
        self.conv1 = nn.Conv2d(in_planes, planes, 1, bias=False)
        self.bn1 = _bn(planes, relu=True)
        self.conv2 = FusedConv2d(planes, planes, 3, stride=stride,
                                 padding=1, bias=False)
        self.bn2 = _bn(planes, relu=True)
        self.conv3 = nn.Conv2d(planes, self.expansion * planes, 1,
                               bias=False)
        # bn3 fuses the residual add + final ReLU of the block.
        self.bn3 = _bn(self.expansion * planes, relu=True)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_planes != self.expansion * planes:
            self.shortcut = nn.Sequential(
                FusedConv2d(in_planes, self.expansion * planes, 1,
                            stride=stride, bias=False),
                _bn(self.expansion * planes))

    def forward(self, x):
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), residual=self.shortcut(x))


class CifarResNet(nn.Module):
    """CIFAR-style ResNet: 3x3 stem, no maxpool, 4x4 final feature map."""

    def __init__(self, block, num_blocks, num_classes=10):
        super().__init__()
        self.in_planes = 64
        self.conv1 = nn.Conv2d(3, 64, 3, stride=1, padding=1, bias=False)
        self.bn1 = _bn(64, relu=True)
        self.layer1 = self._make_layer(block, 64, num_blocks[0], 1)
        self.layer2 = self._make_layer(block, 128, num_blocks[1], 2)
        self.layer3 = self._make_layer(block, 256, num_blocks[2], 2)
        self.layer4 = self._make_layer(block, 512, num_blocks[3], 2)
        self.linear = nn.Linear(512 * block.expansion, num_classes)

    def _make_layer(self, block, planes, n, stride):
        strides = [stride] + [1] * (n - 1)
        layers = []
        for s in strides:
            layers.append(block(self.in_planes, planes, s))
            self.in_planes = planes * block.expansion
        return nn.Sequential(*layers)

    def forward(self, x):
        out = self.bn1(self.conv1(x))
        out = self.layer4(self.layer3(self.layer2(self.layer1(out))))
        out = F.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.linear(out)


class ImageNetResNet(nn.Module):
    """ImageNet-style ResNet: 7x7 stem + maxpool."""

    def __init__(self, block, num_blocks, num_classes=1000):
        super().__init__()
        self.in_planes = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = _bn(64, relu=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, num_blocks[0], 1)
        self.layer2 = self._make_layer(block, 128, num_blocks[1], 2)
        self.layer3 = self._make_layer(block, 256, num_blocks[2], 2)
        self.layer4 = self._make_layer(block, 512, num_blocks[3], 2)
        self.fc = nn.Linear(512 * block.expansion, num_classes)

    _make_layer = CifarResNet._make_layer

    def forward(self, x):
        out = self.maxpool(self.bn1(self.conv1(x)))
        out = self.layer4(self.layer3(self.layer2(self.layer1(out))))
        out = F.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.fc(out)


def ResNet18(num_classes=10):
    return CifarResNet(BasicBlock, [2, 2, 2, 2], num_classes)


def ResNet34(num_classes=10):
    return CifarResNet(BasicBlock, [3, 4, 6, 3], num_classes)


def ResNet50(num_classes=1000):
    return ImageNetResNet(Bottleneck, [3, 4, 6, 3], num_classes)


def ResNet50Cifar(num_classes=10):
    return CifarResNet(Bottleneck, [3, 4, 6, 3], num_classes)
