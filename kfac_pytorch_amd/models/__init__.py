from kfac_pytorch_amd.models.cifar_resnet import get_cifar_model
from kfac_pytorch_amd.models.imagenet_resnet import get_imagenet_model

__all__ = ["get_cifar_model", "get_imagenet_model"]
