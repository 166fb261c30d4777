"""CIFAR ResNets (resnet20/32/44/56/110), written from scratch.

The depth-6n+2 plain-block family the reference trains on CIFAR-10/100
(reference: examples/cifar_resnet.py, selected at
examples/pytorch_cifar10_resnet.py:200-217).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = ["CifarResNet", "resnet20", "resnet32", "resnet44", "resnet56",
           "resnet110", "get_cifar_model"]


class PlainBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.conv1 = nn.Conv2d(cin, cout, 3, stride=stride, padding=1,
                               bias=False)
        self.bn1 = nn.BatchNorm2d(cout)
        self.conv2 = nn.Conv2d(cout, cout, 3, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(cout)
        self.shortcut = nn.Sequential()
        if stride != 1 or cin != cout:
            self.shortcut = nn.Sequential(
                nn.Conv2d(cin, cout, 1, stride=stride, bias=False),
                nn.BatchNorm2d(cout),
            )

    def forward(self, x):
        out = F.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        return F.relu(out + self.shortcut(x))


class CifarResNet(nn.Module):
    def __init__(self, depth: int, num_classes: int = 10):
        super().__init__()
        if (depth - 2) % 6 != 0:
            raise ValueError("CIFAR ResNet depth must be 6n+2")
        n = (depth - 2) // 6
        self.conv1 = nn.Conv2d(3, 16, 3, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(16)
        self.layer1 = self._make_layer(16, 16, n, 1)
        self.layer2 = self._make_layer(16, 32, n, 2)
        self.layer3 = self._make_layer(32, 64, n, 2)
        self.fc = nn.Linear(64, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")

    @staticmethod
    def _make_layer(cin, cout, blocks, stride):
        layers = [PlainBlock(cin, cout, stride)]
        for _ in range(1, blocks):
            layers.append(PlainBlock(cout, cout))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = F.relu(self.bn1(self.conv1(x)))
        x = self.layer3(self.layer2(self.layer1(x)))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


def resnet20(num_classes=10):
    return CifarResNet(20, num_classes)


def resnet32(num_classes=10):
    return CifarResNet(32, num_classes)


def resnet44(num_classes=10):
    return CifarResNet(44, num_classes)


def resnet56(num_classes=10):
    return CifarResNet(56, num_classes)


def resnet110(num_classes=10):
    return CifarResNet(110, num_classes)


_MODELS = {"resnet20": resnet20, "resnet32": resnet32, "resnet44": resnet44,
           "resnet56": resnet56, "resnet110": resnet110}


def get_cifar_model(name: str, num_classes: int = 10) -> nn.Module:
    if name not in _MODELS:
        raise ValueError(f"unknown cifar model {name!r}; have "
                         f"{sorted(_MODELS)}")
    return _MODELS[name](num_classes=num_classes)
