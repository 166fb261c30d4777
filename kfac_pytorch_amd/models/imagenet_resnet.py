"""ImageNet-scale ResNet family, written from scratch (no torchvision
dependency in this image).

Covers the architectures the reference trains
(reference: examples/imagenet_resnet.py, selected at
examples/pytorch_imagenet_resnet.py:235-258): resnet18/34/50/101/152 and
the wide/resnext variants via constructor args.  K-FAC preconditions the
``Conv2d``/``Linear`` modules these define; BatchNorm stays first-order.
"""

from __future__ import annotations

from typing import List, Optional, Type, Union

import torch
import torch.nn as nn

__all__ = [
    "ResNet", "resnet18", "resnet34", "resnet50", "resnet101", "resnet152",
    "resnext50_32x4d", "wide_resnet50_2", "get_imagenet_model",
]


def conv3x3(cin, cout, stride=1, groups=1, dilation=1):
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=dilation,
                     groups=groups, bias=False, dilation=dilation)


def conv1x1(cin, cout, stride=1):
    return nn.Conv2d(cin, cout, 1, stride=stride, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None,
                 groups=1, base_width=64, dilation=1):
        super().__init__()
        if groups != 1 or base_width != 64:
            raise ValueError("BasicBlock only supports groups=1, width=64")
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


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None,
                 groups=1, base_width=64, dilation=1):
        super().__init__()
        width = int(planes * (base_width / 64.0)) * groups
        self.conv1 = conv1x1(inplanes, width)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = conv3x3(width, width, stride, groups, dilation)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = conv1x1(width, planes * self.expansion)
        self.bn3 = nn.BatchNorm2d(planes * self.expansion)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        identity = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.relu(out + identity)


class ResNet(nn.Module):
    def __init__(self, block: Type[Union[BasicBlock, Bottleneck]],
                 layers: List[int], num_classes: int = 1000,
                 groups: int = 1, width_per_group: int = 64,
                 zero_init_residual: bool = False):
        super().__init__()
        self.inplanes = 64
        self.dilation = 1
        self.groups = groups
        self.base_width = width_per_group
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(512 * block.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)
        if zero_init_residual:
            for m in self.modules():
                if isinstance(m, Bottleneck):
                    nn.init.constant_(m.bn3.weight, 0)
                elif isinstance(m, BasicBlock):
                    nn.init.constant_(m.bn2.weight, 0)

    def _make_layer(self, block, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * block.expansion, stride),
                nn.BatchNorm2d(planes * block.expansion),
            )
        layers = [block(self.inplanes, planes, stride, downsample,
                        self.groups, self.base_width, self.dilation)]
        self.inplanes = planes * block.expansion
        for _ in range(1, blocks):
            layers.append(block(self.inplanes, planes, groups=self.groups,
                                base_width=self.base_width,
                                dilation=self.dilation))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = torch.flatten(self.avgpool(x), 1)
        return self.fc(x)


def resnet18(num_classes=1000, **kw):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, **kw)


def resnet34(num_classes=1000, **kw):
    return ResNet(BasicBlock, [3, 4, 6, 3], num_classes, **kw)


def resnet50(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, **kw)


def resnet101(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes, **kw)


def resnet152(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 8, 36, 3], num_classes, **kw)


def resnext50_32x4d(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes,
                  groups=32, width_per_group=4, **kw)


def wide_resnet50_2(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes,
                  width_per_group=128, **kw)


_MODELS = {
    "resnet18": resnet18, "resnet34": resnet34, "resnet50": resnet50,
    "resnet101": resnet101, "resnet152": resnet152,
    "resnext50": resnext50_32x4d, "wide_resnet50": wide_resnet50_2,
}


def get_imagenet_model(name: str, num_classes: int = 1000) -> nn.Module:
    """Name -> ImageNet model.  Covers the reference trainer's model list
    (reference: examples/pytorch_imagenet_resnet.py:235-258): resnets +
    resnext here, densenet/vgg/inception/mobilenet in imagenet_extras."""
    if name not in _MODELS:
        from kfac_pytorch_amd.models import imagenet_extras as ex
        from kfac_pytorch_amd.models import vit as _vit
        extras = {
            "densenet121": ex.densenet121, "densenet201": ex.densenet201,
            "inceptionv3": ex.inception_v3, "inceptionv4": ex.inception_v4,
            "mobilenetv2": ex.mobilenet_v2, "vgg16": ex.vgg16_imagenet,
            "vit_tiny": _vit.vit_tiny, "vit_small": _vit.vit_small,
        }
        if name in extras:
            return extras[name](num_classes=num_classes)
        raise ValueError(f"unknown imagenet model {name!r}; "
                         f"have {sorted(_MODELS) + sorted(extras)}")
    return _MODELS[name](num_classes=num_classes)
