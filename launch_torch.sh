#!/bin/bash
# Single-node launcher: one process per GPU over RCCL.
# (Reference analog: launch_torch.sh -- its ssh multi-node loop maps to
# running this per node with MASTER_ADDR pointing at node 0.)
#
# Usage: nproc=8 script=examples/train_cifar.py bash launch_torch.sh [args...]

nproc="${nproc:-$(python -c 'import torch; print(torch.cuda.device_count() or 1)')}"
script="${script:-examples/train_cifar.py}"
master_addr="${master_addr:-127.0.0.1}"
master_port="${master_port:-29500}"

exec python -m torch.distributed.run \
    --nnodes="${nnodes:-1}" --nproc-per-node "$nproc" \
    --master-addr "$master_addr" --master-port "$master_port" \
    "$script" "$@"
