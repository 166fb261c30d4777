#!/bin/bash
# torch.distributed launcher: one process per GPU over RCCL (xGMI
# intra-node, IB/RoCE across nodes).
#
# Single node (default):
#   nproc=8 script=examples/train_cifar.py bash launch_torch.sh [args...]
#
# Multi node (reference analog: launch_torch.sh's ssh loop over
# gpu1..gpu16, reference :31-43): give a hostfile (one hostname per
# line, first = master / rank-0 node) and this script ssh-launches one
# torchrun per node, mirroring the reference's cluster workflow:
#   hostfile=configs/cluster2 nproc=8 bash launch_torch.sh [args...]
#
# configs/envs.conf provides per-cluster env (python path, NCCL_SOCKET_IFNAME
# for the RCCL bootstrap NIC, NCCL_IB_HCA for RDMA).

set -e
cd "$(dirname "$0")"
[ -f configs/envs.conf ] && source configs/envs.conf

nproc="${nproc:-$(python -c 'import torch; print(torch.cuda.device_count() or 1)')}"
script="${script:-examples/train_cifar.py}"
master_port="${master_port:-29500}"

if [ -z "$hostfile" ]; then
    master_addr="${master_addr:-127.0.0.1}"
    exec python -m torch.distributed.run \
        --nnodes="${nnodes:-1}" --nproc-per-node "$nproc" \
        --master-addr "$master_addr" --master-port "$master_port" \
        "$script" "$@"
fi

mapfile -t hosts < <(grep -v '^\s*#' "$hostfile" | awk 'NF{print $1}')
nnodes="${#hosts[@]}"
master_addr="${master_addr:-${hosts[0]}}"
repo="$(pwd)"
echo "launching $nnodes nodes x $nproc procs (master $master_addr)"

pids=()
for i in "${!hosts[@]}"; do
    h="${hosts[$i]}"
    cmd="cd $repo && \
        HSA_ENABLE_IPC_MODE_LEGACY=0 \
        python -m torch.distributed.run \
        --nnodes=$nnodes --node-rank=$i --nproc-per-node $nproc \
        --master-addr $master_addr --master-port $master_port \
        $script $*"
    if [ "$i" -eq 0 ] && { [ "$h" = "$(hostname)" ] || [ "$h" = "127.0.0.1" ] || [ "$h" = "localhost" ]; }; then
        bash -c "$cmd" &
    else
        ssh -o StrictHostKeyChecking=no "$h" "$cmd" &
    fi
    pids+=($!)
done

rc=0
for p in "${pids[@]}"; do
    wait "$p" || rc=$?
done
exit $rc
