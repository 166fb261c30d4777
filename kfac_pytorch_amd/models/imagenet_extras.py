"""Additional ImageNet model families the reference trainer exposes
(reference: examples/pytorch_imagenet_resnet.py:235-258 -- densenet,
vgg16, inceptionv3/v4, mobilenetv2; examples/imagenet_inceptionv4.py).

Own implementations (not torchvision imports, which this offline image
does not bundle weights for): conv layers are plain ``nn.Conv2d`` /
``nn.Linear`` so K-FAC's hooks attach to every preconditionable module.
All models take ``num_classes`` and 224x224 inputs (inception-v3/v4
accept 299x299 as in the reference trainer's val pipeline but work on
224 too -- global pooling at the head).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = [
    "DenseNet", "densenet121", "densenet201",
    "InceptionV3", "inception_v3",
    "InceptionV4", "inception_v4",
    "MobileNetV2", "mobilenet_v2",
    "vgg16_imagenet",
]


# ---------------------------------------------------------------------------
# DenseNet (densenet121 / densenet201)
# ---------------------------------------------------------------------------
class _DenseLayer(nn.Module):
    def __init__(self, in_ch: int, growth: int, bn_size: int = 4):
        super().__init__()
        self.norm1 = nn.BatchNorm2d(in_ch)
        self.conv1 = nn.Conv2d(in_ch, bn_size * growth, 1, bias=False)
        self.norm2 = nn.BatchNorm2d(bn_size * growth)
        self.conv2 = nn.Conv2d(bn_size * growth, growth, 3, padding=1,
                               bias=False)

    def forward(self, x):
        out = self.conv1(F.relu(self.norm1(x)))
        out = self.conv2(F.relu(self.norm2(out)))
        return torch.cat([x, out], 1)


class _Transition(nn.Sequential):
    def __init__(self, in_ch: int, out_ch: int):
        super().__init__(nn.BatchNorm2d(in_ch), nn.ReLU(inplace=True),
                         nn.Conv2d(in_ch, out_ch, 1, bias=False),
                         nn.AvgPool2d(2))


class DenseNet(nn.Module):
    def __init__(self, block_config=(6, 12, 24, 16), growth: int = 32,
                 num_init: int = 64, num_classes: int = 1000):
        super().__init__()
        self.stem = nn.Sequential(
            nn.Conv2d(3, num_init, 7, stride=2, padding=3, bias=False),
            nn.BatchNorm2d(num_init), nn.ReLU(inplace=True),
            nn.MaxPool2d(3, stride=2, padding=1))
        ch = num_init
        blocks = []
        for i, n in enumerate(block_config):
            for _ in range(n):
                blocks.append(_DenseLayer(ch, growth))
                ch += growth
            if i != len(block_config) - 1:
                blocks.append(_Transition(ch, ch // 2))
                ch //= 2
        self.features = nn.Sequential(*blocks)
        self.norm_final = nn.BatchNorm2d(ch)
        self.classifier = nn.Linear(ch, num_classes)

    def forward(self, x):
        x = self.features(self.stem(x))
        x = F.relu(self.norm_final(x))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.classifier(x)


def densenet121(num_classes=1000):
    return DenseNet((6, 12, 24, 16), num_classes=num_classes)


def densenet201(num_classes=1000):
    return DenseNet((6, 12, 48, 32), num_classes=num_classes)


# ---------------------------------------------------------------------------
# Inception v3 / v4 (compact faithful blocks)
# ---------------------------------------------------------------------------
class _ConvBN(nn.Sequential):
    def __init__(self, in_ch, out_ch, **kw):
        super().__init__(nn.Conv2d(in_ch, out_ch, bias=False, **kw),
                         nn.BatchNorm2d(out_ch), nn.ReLU(inplace=True))


class _InceptionA3(nn.Module):
    """Inception-v3 35x35 block (5x5 + double-3x3 + pool branches)."""

    def __init__(self, in_ch, pool_ch):
        super().__init__()
        self.b1 = _ConvBN(in_ch, 64, kernel_size=1)
        self.b5 = nn.Sequential(_ConvBN(in_ch, 48, kernel_size=1),
                                _ConvBN(48, 64, kernel_size=5, padding=2))
        self.b3 = nn.Sequential(_ConvBN(in_ch, 64, kernel_size=1),
                                _ConvBN(64, 96, kernel_size=3, padding=1),
                                _ConvBN(96, 96, kernel_size=3, padding=1))
        self.bp = _ConvBN(in_ch, pool_ch, kernel_size=1)

    def forward(self, x):
        pool = F.avg_pool2d(x, 3, stride=1, padding=1)
        return torch.cat([self.b1(x), self.b5(x), self.b3(x),
                          self.bp(pool)], 1)


class _InceptionB3(nn.Module):
    """Inception-v3 17x17 block with 1x7/7x1 factorized convs."""

    def __init__(self, in_ch, mid):
        super().__init__()
        self.b1 = _ConvBN(in_ch, 192, kernel_size=1)
        self.b7 = nn.Sequential(
            _ConvBN(in_ch, mid, kernel_size=1),
            _ConvBN(mid, mid, kernel_size=(1, 7), padding=(0, 3)),
            _ConvBN(mid, 192, kernel_size=(7, 1), padding=(3, 0)))
        self.b77 = nn.Sequential(
            _ConvBN(in_ch, mid, kernel_size=1),
            _ConvBN(mid, mid, kernel_size=(7, 1), padding=(3, 0)),
            _ConvBN(mid, mid, kernel_size=(1, 7), padding=(0, 3)),
            _ConvBN(mid, mid, kernel_size=(7, 1), padding=(3, 0)),
            _ConvBN(mid, 192, kernel_size=(1, 7), padding=(0, 3)))
        self.bp = _ConvBN(in_ch, 192, kernel_size=1)

    def forward(self, x):
        pool = F.avg_pool2d(x, 3, stride=1, padding=1)
        return torch.cat([self.b1(x), self.b7(x), self.b77(x),
                          self.bp(pool)], 1)


class _Reduction(nn.Module):
    """Grid-size reduction: stride-2 conv branches + maxpool."""

    def __init__(self, in_ch, k1, k2):
        super().__init__()
        self.b3 = _ConvBN(in_ch, k1, kernel_size=3, stride=2)
        self.b33 = nn.Sequential(
            _ConvBN(in_ch, k2, kernel_size=1),
            _ConvBN(k2, k2, kernel_size=3, padding=1),
            _ConvBN(k2, k2, kernel_size=3, stride=2))

    def forward(self, x):
        return torch.cat([self.b3(x), self.b33(x),
                          F.max_pool2d(x, 3, stride=2)], 1)


class InceptionV3(nn.Module):
    def __init__(self, num_classes: int = 1000):
        super().__init__()
        self.stem = nn.Sequential(
            _ConvBN(3, 32, kernel_size=3, stride=2),
            _ConvBN(32, 32, kernel_size=3),
            _ConvBN(32, 64, kernel_size=3, padding=1),
            nn.MaxPool2d(3, stride=2),
            _ConvBN(64, 80, kernel_size=1),
            _ConvBN(80, 192, kernel_size=3),
            nn.MaxPool2d(3, stride=2))
        self.mixed_a = nn.Sequential(_InceptionA3(192, 32),
                                     _InceptionA3(256, 64),
                                     _InceptionA3(288, 64))
        self.red_a = _Reduction(288, 384, 96)
        self.mixed_b = nn.Sequential(_InceptionB3(768, 128),
                                     _InceptionB3(768, 160),
                                     _InceptionB3(768, 160),
                                     _InceptionB3(768, 192))
        self.red_b = _Reduction(768, 320, 192)
        self.head_conv = _ConvBN(1280, 2048, kernel_size=1)
        self.fc = nn.Linear(2048, num_classes)

    def forward(self, x):
        x = self.stem(x)
        x = self.red_a(self.mixed_a(x))
        x = self.red_b(self.mixed_b(x))
        x = self.head_conv(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


class _InceptionA4(nn.Module):
    """Inception-v4 35x35 block (reference: examples/imagenet_inceptionv4.py
    Inception_A)."""

    def __init__(self, in_ch=384):
        super().__init__()
        self.b1 = _ConvBN(in_ch, 96, kernel_size=1)
        self.b3 = nn.Sequential(_ConvBN(in_ch, 64, kernel_size=1),
                                _ConvBN(64, 96, kernel_size=3, padding=1))
        self.b33 = nn.Sequential(_ConvBN(in_ch, 64, kernel_size=1),
                                 _ConvBN(64, 96, kernel_size=3, padding=1),
                                 _ConvBN(96, 96, kernel_size=3, padding=1))
        self.bp = _ConvBN(in_ch, 96, kernel_size=1)

    def forward(self, x):
        pool = F.avg_pool2d(x, 3, stride=1, padding=1)
        return torch.cat([self.b1(x), self.b3(x), self.b33(x),
                          self.bp(pool)], 1)


class _InceptionB4(nn.Module):
    """Inception-v4 17x17 block."""

    def __init__(self, in_ch=1024):
        super().__init__()
        self.b1 = _ConvBN(in_ch, 384, kernel_size=1)
        self.b7 = nn.Sequential(
            _ConvBN(in_ch, 192, kernel_size=1),
            _ConvBN(192, 224, kernel_size=(1, 7), padding=(0, 3)),
            _ConvBN(224, 256, kernel_size=(7, 1), padding=(3, 0)))
        self.b77 = nn.Sequential(
            _ConvBN(in_ch, 192, kernel_size=1),
            _ConvBN(192, 192, kernel_size=(7, 1), padding=(3, 0)),
            _ConvBN(192, 224, kernel_size=(1, 7), padding=(0, 3)),
            _ConvBN(224, 224, kernel_size=(7, 1), padding=(3, 0)),
            _ConvBN(224, 256, kernel_size=(1, 7), padding=(0, 3)))
        self.bp = _ConvBN(in_ch, 128, kernel_size=1)

    def forward(self, x):
        pool = F.avg_pool2d(x, 3, stride=1, padding=1)
        return torch.cat([self.b1(x), self.b7(x), self.b77(x),
                          self.bp(pool)], 1)


class _InceptionC4(nn.Module):
    """Inception-v4 8x8 block with split 1x3/3x1 branches."""

    def __init__(self, in_ch=1536):
        super().__init__()
        self.b1 = _ConvBN(in_ch, 256, kernel_size=1)
        self.b3_stem = _ConvBN(in_ch, 384, kernel_size=1)
        self.b3_a = _ConvBN(384, 256, kernel_size=(1, 3), padding=(0, 1))
        self.b3_b = _ConvBN(384, 256, kernel_size=(3, 1), padding=(1, 0))
        self.b33_stem = nn.Sequential(
            _ConvBN(in_ch, 384, kernel_size=1),
            _ConvBN(384, 448, kernel_size=(3, 1), padding=(1, 0)),
            _ConvBN(448, 512, kernel_size=(1, 3), padding=(0, 1)))
        self.b33_a = _ConvBN(512, 256, kernel_size=(1, 3), padding=(0, 1))
        self.b33_b = _ConvBN(512, 256, kernel_size=(3, 1), padding=(1, 0))
        self.bp = _ConvBN(in_ch, 256, kernel_size=1)

    def forward(self, x):
        pool = F.avg_pool2d(x, 3, stride=1, padding=1)
        s3 = self.b3_stem(x)
        s33 = self.b33_stem(x)
        return torch.cat([self.b1(x), self.b3_a(s3), self.b3_b(s3),
                          self.b33_a(s33), self.b33_b(s33),
                          self.bp(pool)], 1)


class InceptionV4(nn.Module):
    """Inception-v4 (4xA + 7xB + 3xC as in the reference,
    examples/imagenet_inceptionv4.py Inceptionv4 ctor)."""

    def __init__(self, num_classes: int = 1000):
        super().__init__()
        self.stem = nn.Sequential(
            _ConvBN(3, 32, kernel_size=3, stride=2),
            _ConvBN(32, 32, kernel_size=3),
            _ConvBN(32, 64, kernel_size=3, padding=1),
            nn.MaxPool2d(3, stride=2),
            _ConvBN(64, 96, kernel_size=1),
            _ConvBN(96, 192, kernel_size=3),
            _ConvBN(192, 384, kernel_size=3, stride=2, padding=1))
        self.blocks_a = nn.Sequential(*[_InceptionA4() for _ in range(4)])
        self.red_a = _Reduction(384, 384, 256)
        self.blocks_b = nn.Sequential(*[_InceptionB4() for _ in range(7)])
        self.red_b = _Reduction(1024, 256, 256)
        self.blocks_c = nn.Sequential(*[_InceptionC4() for _ in range(3)])
        self.fc = nn.Linear(1536, num_classes)

    def forward(self, x):
        x = self.stem(x)
        x = self.red_a(self.blocks_a(x))
        x = self.red_b(self.blocks_b(x))
        x = self.blocks_c(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


def inception_v3(num_classes=1000):
    return InceptionV3(num_classes)


def inception_v4(num_classes=1000):
    return InceptionV4(num_classes)


# ---------------------------------------------------------------------------
# MobileNetV2
# ---------------------------------------------------------------------------
class _InvertedResidual(nn.Module):
    def __init__(self, in_ch, out_ch, stride, expand):
        super().__init__()
        hid = in_ch * expand
        self.use_res = stride == 1 and in_ch == out_ch
        layers = []
        if expand != 1:
            layers.append(_ConvBN(in_ch, hid, kernel_size=1))
        layers += [
            nn.Conv2d(hid, hid, 3, stride=stride, padding=1, groups=hid,
                      bias=False),
            nn.BatchNorm2d(hid), nn.ReLU6(inplace=True),
            nn.Conv2d(hid, out_ch, 1, bias=False), nn.BatchNorm2d(out_ch)]
        self.conv = nn.Sequential(*layers)

    def forward(self, x):
        out = self.conv(x)
        return x + out if self.use_res else out


class MobileNetV2(nn.Module):
    CFG = [(1, 16, 1, 1), (6, 24, 2, 2), (6, 32, 3, 2), (6, 64, 4, 2),
           (6, 96, 3, 1), (6, 160, 3, 2), (6, 320, 1, 1)]

    def __init__(self, num_classes: int = 1000):
        super().__init__()
        layers = [_ConvBN(3, 32, kernel_size=3, stride=2, padding=1)]
        ch = 32
        for t, c, n, s in self.CFG:
            for i in range(n):
                layers.append(_InvertedResidual(ch, c, s if i == 0 else 1, t))
                ch = c
        layers.append(_ConvBN(ch, 1280, kernel_size=1))
        self.features = nn.Sequential(*layers)
        self.classifier = nn.Linear(1280, num_classes)

    def forward(self, x):
        x = self.features(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.classifier(x)


def mobilenet_v2(num_classes=1000):
    return MobileNetV2(num_classes)


# ---------------------------------------------------------------------------
# VGG-16 (ImageNet head)
# ---------------------------------------------------------------------------
def vgg16_imagenet(num_classes=1000):
    cfg = [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
           512, 512, 512, "M", 512, 512, 512, "M"]
    layers = []
    ch = 3
    for v in cfg:
        if v == "M":
            layers.append(nn.MaxPool2d(2))
        else:
            layers += [nn.Conv2d(ch, v, 3, padding=1), nn.BatchNorm2d(v),
                       nn.ReLU(inplace=True)]
            ch = v
    return nn.Sequential(
        *layers, nn.AdaptiveAvgPool2d(7), nn.Flatten(),
        nn.Linear(512 * 49, 4096), nn.ReLU(inplace=True), nn.Dropout(0.5),
        nn.Linear(4096, 4096), nn.ReLU(inplace=True), nn.Dropout(0.5),
        nn.Linear(4096, num_classes))
