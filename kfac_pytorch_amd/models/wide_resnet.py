"""CIFAR Wide ResNet (WRN-28-10 / WRN-28-20)
(reference capability: examples/cifar_wide_resnet.py)."""

import torch.nn as nn
import torch.nn.functional as F

__all__ = ["WideResNet", "wrn28_10", "wrn28_20"]


class WideBlock(nn.Module):
    def __init__(self, cin, cout, stride, drop_rate=0.0):
        super().__init__()
        self.bn1 = nn.BatchNorm2d(cin)
        self.conv1 = nn.Conv2d(cin, cout, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(cout)
        self.conv2 = nn.Conv2d(cout, cout, 3, padding=1, bias=False)
        self.drop_rate = drop_rate
        self.equal = (cin == cout and stride == 1)
        self.shortcut = None if self.equal else \
            nn.Conv2d(cin, cout, 1, stride=stride, bias=False)

    def forward(self, x):
        out = F.relu(self.bn1(x))
        shortcut = x if self.equal else self.shortcut(out)
        out = self.conv1(out)
        out = F.relu(self.bn2(out))
        if self.drop_rate > 0:
            out = F.dropout(out, p=self.drop_rate, training=self.training)
        return self.conv2(out) + shortcut


class WideResNet(nn.Module):
    def __init__(self, depth: int = 28, widen: int = 10,
                 num_classes: int = 10, drop_rate: float = 0.0):
        super().__init__()
        assert (depth - 4) % 6 == 0, "WRN depth must be 6n+4"
        n = (depth - 4) // 6
        widths = [16, 16 * widen, 32 * widen, 64 * widen]
        self.conv1 = nn.Conv2d(3, widths[0], 3, padding=1, bias=False)
        self.block1 = self._make(widths[0], widths[1], n, 1, drop_rate)
        self.block2 = self._make(widths[1], widths[2], n, 2, drop_rate)
        self.block3 = self._make(widths[2], widths[3], n, 2, drop_rate)
        self.bn = nn.BatchNorm2d(widths[3])
        self.fc = nn.Linear(widths[3], num_classes)

    @staticmethod
    def _make(cin, cout, n, stride, drop_rate):
        layers = [WideBlock(cin, cout, stride, drop_rate)]
        for _ in range(1, n):
            layers.append(WideBlock(cout, cout, 1, drop_rate))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.conv1(x)
        x = self.block3(self.block2(self.block1(x)))
        x = F.relu(self.bn(x))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


def wrn28_10(num_classes=10):
    return WideResNet(28, 10, num_classes)


def wrn28_20(num_classes=10):
    return WideResNet(28, 20, num_classes)
