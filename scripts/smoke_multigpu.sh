#!/bin/bash
# Multi-GPU smoke for an 8x MI355X node: runs the flagship bench at the
# requested width with the torch-group comm path AND the native
# multi-stream RCCL path (KFAC_NATIVE_COMM=1), plus the fused
# multi_bcast research path on the MPD 'eigen' algorithm.  Exits
# nonzero on the first failure.  Usage: bash scripts/smoke_multigpu.sh [N]
set -e
N=${1:-8}
cd "$(dirname "$0")/.."
# dmabuf IPC is the only mode this driver stack supports; RCCL fails
# with hipIpcGetMemHandle errors without it
export HSA_ENABLE_IPC_MODE_LEGACY=0
RUN="python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
     --master-addr 127.0.0.1 --master-port 29755 bench.py --gpus $N \
     --steps 5 --warmup 2"
echo "== torch-group comm path (default) =="
$RUN
echo "== native RCCL communicator path (KFAC_NATIVE_COMM=1) =="
KFAC_NATIVE_COMM=1 $RUN
echo "== MPD eigen + fused native multi_bcast =="
KFAC_NATIVE_COMM=1 KFAC_NATIVE_MULTIBCAST=1 $RUN --kfac-name eigen
echo "multi-GPU smoke OK at N=$N"
