#!/bin/bash
# SQuAD BERT K-FAC experiment driver (reference analog: train_squad.sh /
# batch.sh:31-32): BERT-base bs4x8GPU, exclude_vocabulary_size=30522.
source "$(dirname "$0")/configs/envs.conf"
nworkers="${nworkers:-8}"
batch_size="${batch_size:-4}"
kfac="${kfac:-eigen_dp}"
damping="${damping:-0.003}"
epochs="${epochs:-2}"
kfac_update_freq="${kfac_update_freq:-10}"
fac_update_freq="${fac_update_freq:-1}"
extra="${extra:-}"
nproc="$nworkers" script=examples/train_bert_squad.py \
    bash "$(dirname "$0")/launch_torch.sh" \
    --batch-size "$batch_size" --kfac-name "$kfac" \
    --damping "$damping" --epochs "$epochs" \
    --kfac-update-freq "$kfac_update_freq" \
    --fac-update-freq "$fac_update_freq" $extra
