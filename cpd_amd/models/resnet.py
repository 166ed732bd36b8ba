"""ImageNet ResNet-50 (bottleneck), implemented in-repo: the environment has
no torchvision (the reference imports torchvision.models.resnet50,
example/ResNet50/main.py:67).  Standard v1 architecture: 7x7/2 stem + 3x3/2
max-pool, stages [3,4,6,3] of Bottleneck(4x expansion), global avg-pool, fc.
"""
import torch.nn as nn
import torch.nn.functional as F

from .fused_bn import FusedBNReLU

__all__ = ["resnet50", "ResNet"]


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_planes, planes, stride=1, downsample=None,
                 fused_bn=False):
        super().__init__()
        self.fused = fused_bn
        self.conv1 = nn.Conv2d(in_planes, planes, 1, bias=False)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=stride, padding=1,
                               bias=False)
        self.conv3 = nn.Conv2d(planes, planes * 4, 1, bias=False)
        if fused_bn:
            self.bn1 = FusedBNReLU(planes, relu=True)
            self.bn2 = FusedBNReLU(planes, relu=True)
            self.bn3 = FusedBNReLU(planes * 4, relu=True)  # + residual
        else:
            self.bn1 = nn.BatchNorm2d(planes)
            self.bn2 = nn.BatchNorm2d(planes)
            self.bn3 = nn.BatchNorm2d(planes * 4)
        self.downsample = downsample

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        if self.fused:
            out = self.bn1(self.conv1(x))
            out = self.bn2(self.conv2(out))
            return self.bn3(self.conv3(out), residual=identity)
        out = F.relu(self.bn1(self.conv1(x)))
        out = F.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        return F.relu(out + identity)


class ResNet(nn.Module):
    def __init__(self, layers=(3, 4, 6, 3), num_classes=1000, fused_bn=False):
        super().__init__()
        self.in_planes = 64
        self.fused = fused_bn
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = FusedBNReLU(64, relu=True) if fused_bn else \
            nn.BatchNorm2d(64)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, layers[0], 1, fused_bn)
        self.layer2 = self._make_layer(128, layers[1], 2, fused_bn)
        self.layer3 = self._make_layer(256, layers[2], 2, fused_bn)
        self.layer4 = self._make_layer(512, layers[3], 2, fused_bn)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * 4, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, (nn.BatchNorm2d, FusedBNReLU)):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def _make_layer(self, planes, blocks, stride, fused_bn):
        downsample = None
        if stride != 1 or self.in_planes != planes * 4:
            dbn = FusedBNReLU(planes * 4, relu=False) if fused_bn else \
                nn.BatchNorm2d(planes * 4)
            downsample = nn.Sequential(
                nn.Conv2d(self.in_planes, planes * 4, 1, stride=stride,
                          bias=False),
                dbn,
            )
        layers = [Bottleneck(self.in_planes, planes, stride, downsample,
                             fused_bn)]
        self.in_planes = planes * 4
        for _ in range(blocks - 1):
            layers.append(Bottleneck(self.in_planes, planes,
                                     fused_bn=fused_bn))
        return nn.Sequential(*layers)

    def forward(self, x, rank=None):
        x = self.bn1(self.conv1(x))
        if not self.fused:
            x = F.relu(x)
        x = self.maxpool(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet50(num_classes=1000, fused_bn=False):
    return ResNet((3, 4, 6, 3), num_classes=num_classes, fused_bn=fused_bn)
