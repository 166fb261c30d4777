#!/bin/bash
# Multi-30k Transformer K-FAC experiment driver (reference analog:
# train_multi30k.sh / batch.sh:27-29): Transformer-base bs128x8GPU,
# vocabulary-size exclusion on the tied pre-softmax projection.
source "$(dirname "$0")/configs/envs.conf"
nworkers="${nworkers:-8}"
batch_size="${batch_size:-128}"
kfac="${kfac:-eigen_dp}"
damping="${damping:-0.003}"
epochs="${epochs:-100}"
kfac_update_freq="${kfac_update_freq:-10}"
fac_update_freq="${fac_update_freq:-1}"
extra="${extra:-}"
nproc="$nworkers" script=examples/train_transformer.py \
    bash "$(dirname "$0")/launch_torch.sh" \
    --batch-size "$batch_size" --kfac-name "$kfac" \
    --damping "$damping" --epochs "$epochs" \
    --kfac-update-freq "$kfac_update_freq" \
    --fac-update-freq "$fac_update_freq" $extra
