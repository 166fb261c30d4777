"""CIFAR VGG-16/19 (reference capability: examples/cifar_vgg.py,
selected at examples/pytorch_cifar10_resnet.py:200-217)."""

import torch.nn as nn

__all__ = ["VGG", "vgg16", "vgg19"]

_CFG = {
    "vgg16": [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
              512, 512, 512, "M", 512, 512, 512, "M"],
    "vgg19": [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M",
              512, 512, 512, 512, "M", 512, 512, 512, 512, "M"],
}


class VGG(nn.Module):
    def __init__(self, name: str = "vgg16", num_classes: int = 10):
        super().__init__()
        layers = []
        cin = 3
        for v in _CFG[name]:
            if v == "M":
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers += [nn.Conv2d(cin, v, 3, padding=1),
                           nn.BatchNorm2d(v), nn.ReLU(inplace=True)]
                cin = v
        self.features = nn.Sequential(*layers)
        self.classifier = nn.Linear(512, num_classes)

    def forward(self, x):
        x = self.features(x).flatten(1)
        return self.classifier(x)


def vgg16(num_classes=10):
    return VGG("vgg16", num_classes)


def vgg19(num_classes=10):
    return VGG("vgg19", num_classes)
