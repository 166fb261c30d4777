#!/usr/bin/env python3
"""Probe subgroup / rotating-group collectives under torch.distributed.

Reference analog: scripts/test_allgather.py (Horovod dynamic
process_sets: disjoint-group allreduce on 4 ranks, plus a DDP
allreduce probe). Here the same behaviors over torch.distributed
(RCCL on GPUs, gloo on CPU):

  * disjoint half-world subgroups, allreduce inside each;
  * rotating duplicate world groups with concurrent owner-rooted
    broadcasts (the K-FAC bucket-broadcast pattern).

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
        --master-addr 127.0.0.1 scripts/test_groups.py
"""

import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    if "RANK" not in os.environ:
        sys.exit("launch with torch.distributed.run (needs >= 2 ranks)")
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
    dist.init_process_group("nccl" if use_cuda else "gloo",
                            init_method="env://")
    import kfac_pytorch_amd.parallel.comm as comm_mod
    comm_mod.init("Torch")
    comm = comm_mod.get_comm()
    dev = "cuda" if use_cuda else "cpu"

    # disjoint subgroups: first half / second half
    half = world // 2
    g1 = comm.new_group(list(range(half)))
    g2 = comm.new_group(list(range(half, world)))
    mine = g1 if rank < half else g2
    t = torch.full((4,), float(rank), device=dev)
    dist.all_reduce(t, group=mine)
    expect = sum(range(half)) if rank < half else sum(range(half, world))
    assert torch.allclose(t, torch.full((4,), float(expect), device=dev)), \
        (rank, t)

    # rotating duplicate world groups, concurrent owner broadcasts
    n = comm.ensure_rotating_groups(min(4, world))
    handles = []
    tensors = []
    for r in range(world):
        buf = torch.full((8,), float(rank * 100 + r), device=dev)
        tensors.append(buf)
        handles.append(comm.broadcast_async_(
            buf, src=r, group=comm.rotating_group(r)))
    comm.synchronize(handles)
    for r, buf in enumerate(tensors):
        assert torch.allclose(
            buf, torch.full((8,), float(r * 100 + r), device=dev)), (r, buf)

    if rank == 0:
        print(f"group probes OK: world={world}, rotating_groups={n}, "
              f"device={dev}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
