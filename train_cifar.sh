#!/bin/bash
# CIFAR-10/100 K-FAC experiment driver (reference analog: train_cifar10.sh).
# Env-var-overridable hyperparameters; headline config resnet110 bs128x4GPU
# damping 0.03 (reference: batch.sh:25, train_cifar10.sh:4-27).
source "$(dirname "$0")/configs/envs.conf"
nworkers="${nworkers:-4}"
model="${model:-resnet110}"
batch_size="${batch_size:-128}"
kfac="${kfac:-eigen_dp}"
damping="${damping:-0.03}"
epochs="${epochs:-100}"
kfac_update_freq="${kfac_update_freq:-10}"
fac_update_freq="${fac_update_freq:-1}"
lr="${lr:-0.1}"
extra="${extra:-}"
nproc="$nworkers" script=examples/train_cifar.py bash "$(dirname "$0")/launch_torch.sh" \
    --model "$model" --batch-size "$batch_size" --kfac-name "$kfac" \
    --damping "$damping" --epochs "$epochs" \
    --kfac-update-freq "$kfac_update_freq" \
    --fac-update-freq "$fac_update_freq" --base-lr "$lr" $extra
