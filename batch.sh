#!/bin/bash
# Throughput-mode sweep over the BASELINE efficiency configs
# (reference analog: batch.sh:17-32).
set -e
cd "$(dirname "$0")"
nw="${nworkers:-8}"
# ResNet-110 bs128, damping 0.03
nworkers=$nw model=resnet110 batch_size=128 damping=0.03 extra="--speed --iters-per-epoch 60 --epochs 1" bash train_cifar.sh
# VGG-16 bs128
nworkers=$nw model=vgg16 batch_size=128 damping=0.03 extra="--speed --iters-per-epoch 60 --epochs 1" bash train_cifar.sh
# ResNet-50 bs32 (headline)
nworkers=$nw model=resnet50 batch_size=32 damping=0.002 extra="--speed --iters-per-epoch 60 --epochs 1" bash train_imagenet.sh
# DenseNet-201 bs16
nworkers=$nw model=densenet201 batch_size=16 damping=0.002 extra="--speed --iters-per-epoch 60 --epochs 1" bash train_imagenet.sh
# Inception-v4 bs16
nworkers=$nw model=inceptionv4 batch_size=16 damping=0.002 extra="--speed --iters-per-epoch 60 --epochs 1" bash train_imagenet.sh
# Transformer bs128
nproc=$nw script=examples/train_transformer.py bash launch_torch.sh --batch-size 128 --speed --iters-per-epoch 60 --epochs 1
# BERT-base bs4
nproc=$nw script=examples/train_bert_squad.py bash launch_torch.sh --batch-size 4 --speed --iters-per-epoch 60 --epochs 1
