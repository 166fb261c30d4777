#!/usr/bin/env python3
"""Multi-30k-shape Transformer training with distributed K-FAC.

Reference analog: examples/pytorch_multi30k_transformer.py -- K-FAC with
``exclude_vocabulary_size`` to skip the tied pre-softmax projection
(:288-297), Adam-vs-SGD+KFAC switch (:277-286), label smoothing.
Synthetic translation-shaped data (no network)."""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from examples import common  # noqa: E402


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--vocab-size", type=int, default=9521)
    p.add_argument("--seq-len", type=int, default=32)
    p.add_argument("--d-model", type=int, default=512)
    p.add_argument("--layers", type=int, default=6)
    p.add_argument("--optimizer", default="sgd", choices=["sgd", "adam"])
    p.add_argument("--label-smoothing", type=float, default=0.1)
    common.add_common_args(p)
    p.set_defaults(batch_size=128, base_lr=0.1, damping=0.003)
    args = p.parse_args()

    rank, world, local_rank, use_cuda = common.initialize_distributed()
    torch.manual_seed(args.seed)
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")

    from kfac_pytorch_amd.models import make_transformer
    from kfac_pytorch_amd.utils import LabelSmoothLoss
    model = make_transformer(vocab=args.vocab_size, d_model=args.d_model,
                             num_layers=args.layers,
                             max_len=args.seq_len + 2).to(device)
    ddp_model = model
    if world > 1:
        ddp_model = torch.nn.parallel.DistributedDataParallel(
            model, device_ids=[local_rank] if use_cuda else None,
            find_unused_parameters=False)

    if args.optimizer == "adam":
        optimizer = torch.optim.Adam(model.parameters(),
                                     lr=args.base_lr * world * 1e-3,
                                     betas=(0.9, 0.98))
        precond = None
    else:
        optimizer = torch.optim.SGD(model.parameters(),
                                    lr=args.base_lr * world,
                                    momentum=args.momentum,
                                    weight_decay=args.weight_decay)
        import kfac_pytorch_amd as kfac
        precond = None
        if args.kfac_name != "none":
            KFAC = kfac.get_kfac_module(args.kfac_name)
            # exclude the vocab-sized (tied) pre-softmax projection
            # (reference :288-297)
            precond = KFAC(model, lr=args.base_lr * world,
                           damping=args.damping,
                           fac_update_freq=args.fac_update_freq,
                           kfac_update_freq=args.kfac_update_freq,
                           kl_clip=args.kl_clip,
                           factor_decay=args.factor_decay,
                           exclude_vocabulary_size=args.vocab_size,
                           exclude_parts=args.exclude_parts)

    crit = LabelSmoothLoss(args.label_smoothing)
    g = torch.Generator().manual_seed(args.seed + rank)

    def batches():
        out = []
        for _ in range(args.iters_per_epoch):
            src = torch.randint(4, args.vocab_size,
                                (args.batch_size, args.seq_len),
                                generator=g)
            trg = torch.randint(4, args.vocab_size,
                                (args.batch_size, args.seq_len),
                                generator=g)
            out.append((src, trg))
        return out

    def criterion(logits, trg_out):
        return crit(logits.reshape(-1, args.vocab_size),
                    trg_out.reshape(-1))

    import time
    for epoch in range(args.epochs):
        model.train()
        iter_times = []
        last = time.perf_counter()
        for it, (src, trg) in enumerate(batches()):
            src, trg = src.to(device), trg.to(device)
            optimizer.zero_grad(set_to_none=False)
            if args.dtype == "bf16" and use_cuda:
                with torch.autocast("cuda", dtype=torch.bfloat16):
                    logits = ddp_model(src, trg[:, :-1])
                    loss = criterion(logits, trg[:, 1:])
            else:
                logits = ddp_model(src, trg[:, :-1])
                loss = criterion(logits, trg[:, 1:])
            loss.backward()
            if precond is not None:
                precond.step()
            optimizer.step()
            now = time.perf_counter()
            iter_times.append(now - last)
            last = now
            if rank == 0 and (it + 1) % args.display == 0:
                common.logger.info("epoch %d iter %d loss %.4f %.3fs",
                                   epoch, it + 1, loss.item(),
                                   iter_times[-1])
        if args.speed:
            common.report_speed(iter_times, args.batch_size, world, rank)


if __name__ == "__main__":
    main()
