#!/bin/bash
# Damping / update-frequency sweep grid (reference analog: batch-hyper.sh:6-25).
set -e
cd "$(dirname "$0")"
for damping in 0.001 0.002 0.003 0.01 0.03; do
  for freq in 1 10 50 100; do
    nworkers="${nworkers:-4}" model="${model:-resnet110}" damping=$damping \
      kfac_update_freq=$freq extra="--speed --iters-per-epoch 60 --epochs 1" \
      bash train_cifar.sh
  done
done
