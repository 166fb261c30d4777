#!/bin/bash
# One-lease, one-HEAD performance table for profiles/PERFORMANCE.md.
cd "$(dirname "$0")/.."
run() {
    name=$1; shift
    KFAC_PHASE_TIMING=1 timeout 330 python bench.py --steps 10 --warmup 3 "$@" \
        2>/dev/null | grep -E "^KFAC_PHASES|^\{" | tail -2 \
        | sed "s/^/[$name] /"
}
run eigen_dp
run eigen      --kfac-name eigen
run inverse    --kfac-name inverse
run inverse_dp --kfac-name inverse_dp
run eigen_dp_bs64  --batch-size 64
run eigen_dp_bs128 --batch-size 128
run inverse_dp_bs128 --kfac-name inverse_dp --batch-size 128
run eigen_dp_freq10 --kfac-update-freq 10
run sgd_only   --kfac-name none
