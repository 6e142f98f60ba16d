"""ResNet-18/50 implemented in-package (no torchvision in this env).

Convolutions go through torch (MIOpen on ROCm, channels_last/NHWC);
BatchNorm (+fused ReLU) and the classifier run on the gfx950 kernels
(ops/bn.py FusedBN2d, ops/linear.py MFMALinear).  Two stem variants:
CIFAR (3x3, no max-pool; config-2 jobs) and ImageNet (7x7/2 +
max-pool; config-3 ResNet-50 jobs).
"""
from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F

from ..ops.bn import FusedBN2d
from ..ops.linear import MFMALinear


def conv3x3(cin, cout, stride=1):
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.conv1 = conv3x3(cin, cout, stride)
        self.bn1 = FusedBN2d(cout, relu=True)
        self.conv2 = conv3x3(cout, cout)
        self.bn2 = FusedBN2d(cout, relu=True)  # fused +residual +relu
        self.down = None
        if stride != 1 or cin != cout:
            self.down = nn.Sequential(
                nn.Conv2d(cin, cout, 1, stride=stride, bias=False),
                FusedBN2d(cout))

    def forward(self, x):
        idt = x if self.down is None else self.down(x)
        out = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(out), residual=idt)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, cmid, stride=1):
        super().__init__()
        cout = cmid * self.expansion
        self.conv1 = nn.Conv2d(cin, cmid, 1, bias=False)
        self.bn1 = FusedBN2d(cmid, relu=True)
        self.conv2 = conv3x3(cmid, cmid, stride)
        self.bn2 = FusedBN2d(cmid, relu=True)
        self.conv3 = nn.Conv2d(cmid, cout, 1, bias=False)
        self.bn3 = FusedBN2d(cout, relu=True)  # fused +residual +relu
        self.down = None
        if stride != 1 or cin != cout:
            self.down = nn.Sequential(
                nn.Conv2d(cin, cout, 1, stride=stride, bias=False),
                FusedBN2d(cout))

    def forward(self, x):
        idt = x if self.down is None else self.down(x)
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), residual=idt)


class ResNet(nn.Module):
    def __init__(self, block, layers, num_classes=1000, cifar=False):
        super().__init__()
        self.cifar = cifar
        if cifar:
            self.stem = nn.Sequential(conv3x3(3, 64),
                                      FusedBN2d(64, relu=True))
        else:
            self.stem = nn.Sequential(
                nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False),
                FusedBN2d(64, relu=True),
                nn.MaxPool2d(3, stride=2, padding=1))
        self.cin = 64
        stages = []
        for i, (n, c) in enumerate(zip(layers, (64, 128, 256, 512))):
            stages.append(self._stage(block, c, n, 1 if i == 0 else 2))
        self.stages = nn.Sequential(*stages)
        feat = 512 * block.expansion
        self.fc = MFMALinear(feat, num_classes)

    def _stage(self, block, c, n, stride):
        blocks = [block(self.cin, c, stride)]
        self.cin = c * block.expansion
        for _ in range(n - 1):
            blocks.append(block(self.cin, c))
        return nn.Sequential(*blocks)

    def forward(self, x):
        x = self.stages(self.stem(x))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


def resnet18_cifar(num_classes=10):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, cifar=True)


def resnet50(num_classes=1000):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, cifar=False)
