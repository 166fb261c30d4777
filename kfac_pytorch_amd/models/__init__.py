from kfac_pytorch_amd.models.bert import BertShapeForQA, make_bert_base_squad
from kfac_pytorch_amd.models.cifar_resnet import get_cifar_model
from kfac_pytorch_amd.models.cifar_vgg import vgg16, vgg19
from kfac_pytorch_amd.models.imagenet_resnet import get_imagenet_model
from kfac_pytorch_amd.models.rnn_lm import LSTMLanguageModel
from kfac_pytorch_amd.models.transformer import (Seq2SeqTransformer,
                                                 make_transformer)
from kfac_pytorch_amd.models.wide_resnet import wrn28_10, wrn28_20

__all__ = [
    "get_cifar_model", "get_imagenet_model", "vgg16", "vgg19", "wrn28_10",
    "wrn28_20", "Seq2SeqTransformer", "make_transformer", "BertShapeForQA",
    "make_bert_base_squad", "LSTMLanguageModel",
]
