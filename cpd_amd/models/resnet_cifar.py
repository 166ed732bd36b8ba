"""CIFAR ResNet-18 — the flagship benchmark model.

Same architecture class as the reference's res_cifar
(example/ResNet18/models/resnet18_cifar.py:1-87): 3x3 stem (no max-pool),
4 stages x 2 BasicBlocks at 64/128/256/512 channels, 4x4 avg-pool, fc to
num_classes.  forward(x, rank=None) keeps the reference's dummy-rank calling
convention (resnet18_cifar.py:73).

``fused_bn=True`` swaps every BatchNorm(+ReLU)(+residual add) chain for the
fused gfx950 kernels (models/fused_bn.py) — identical math, fewer HBM passes.
"""
import torch.nn as nn
import torch.nn.functional as F

from .fused_bn import FusedBNReLU

__all__ = ["res_cifar", "ResNetCifar"]


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_planes, planes, stride=1, fused_bn=False):
        super().__init__()
        self.fused = fused_bn
        self.conv1 = nn.Conv2d(in_planes, planes, 3, stride=stride, padding=1,
                               bias=False)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=1, padding=1,
                               bias=False)
        if fused_bn:
            self.bn1 = FusedBNReLU(planes, relu=True)
            self.bn2 = FusedBNReLU(planes, relu=True)  # + residual add
        else:
            self.bn1 = nn.BatchNorm2d(planes)
            self.bn2 = nn.BatchNorm2d(planes)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_planes != planes:
            sbn = FusedBNReLU(planes, relu=False) if fused_bn else \
                nn.BatchNorm2d(planes)
            self.shortcut = nn.Sequential(
                nn.Conv2d(in_planes, planes, 1, stride=stride, bias=False),
                sbn,
            )

    def forward(self, x):
        if self.fused:
            out = self.bn1(self.conv1(x))
            return self.bn2(self.conv2(out), residual=self.shortcut(x))
        out = F.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        out = out + self.shortcut(x)
        return F.relu(out)


class ResNetCifar(nn.Module):
    def __init__(self, num_classes=10, fused_bn=False):
        super().__init__()
        self.in_planes = 64
        self.fused = fused_bn
        self.conv1 = nn.Conv2d(3, 64, 3, stride=1, padding=1, bias=False)
        self.bn1 = FusedBNReLU(64, relu=True) if fused_bn else \
            nn.BatchNorm2d(64)
        self.layer1 = self._make_layer(64, 2, 1, fused_bn)
        self.layer2 = self._make_layer(128, 2, 2, fused_bn)
        self.layer3 = self._make_layer(256, 2, 2, fused_bn)
        self.layer4 = self._make_layer(512, 2, 2, fused_bn)
        self.fc = nn.Linear(512, num_classes)

    def _make_layer(self, planes, blocks, stride, fused_bn):
        layers = [BasicBlock(self.in_planes, planes, stride, fused_bn)]
        self.in_planes = planes
        for _ in range(blocks - 1):
            layers.append(BasicBlock(planes, planes, 1, fused_bn))
        return nn.Sequential(*layers)

    def forward(self, x, rank=None):
        out = self.bn1(self.conv1(x))
        if not self.fused:
            out = F.relu(out)
        out = self.layer1(out)
        out = self.layer2(out)
        out = self.layer3(out)
        out = self.layer4(out)
        out = F.avg_pool2d(out, 4)
        out = out.flatten(1)
        return self.fc(out)


def res_cifar(num_classes=10, fused_bn=False):
    return ResNetCifar(num_classes=num_classes, fused_bn=fused_bn)
