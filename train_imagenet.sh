#!/bin/bash
# ImageNet K-FAC experiment driver (reference analog: train_imagenet.sh).
# Headline config: ResNet-50 bs32/GPU eigen_dp damping 0.002 freq 1
# (reference: train_imagenet.sh:4-23, batch.sh:27-29).
source "$(dirname "$0")/configs/envs.conf"
nworkers="${nworkers:-8}"
model="${model:-resnet50}"
batch_size="${batch_size:-32}"
kfac="${kfac:-eigen_dp}"
damping="${damping:-0.002}"
epochs="${epochs:-55}"
kfac_update_freq="${kfac_update_freq:-1}"
fac_update_freq="${fac_update_freq:-1}"
lr="${lr:-0.0125}"
extra="${extra:-}"
nproc="$nworkers" script=examples/train_imagenet.py bash "$(dirname "$0")/launch_torch.sh" \
    --model "$model" --batch-size "$batch_size" --kfac-name "$kfac" \
    --damping "$damping" --epochs "$epochs" \
    --kfac-update-freq "$kfac_update_freq" \
    --fac-update-freq "$fac_update_freq" --base-lr "$lr" $extra
