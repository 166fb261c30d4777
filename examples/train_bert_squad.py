#!/usr/bin/env python3
"""BERT-base SQuAD-shape span-prediction training with distributed K-FAC.

Reference analog: examples/pytorch_squad_bert.py -- bs 4/GPU, AdamW
baseline vs SGD+K-FAC, ``exclude_vocabulary_size=30522`` (:394,450).
Synthetic SQuAD-shaped data (no network; random token ids and spans).
The 768/3072-dim per-layer factors make this the large-eigensolve
stress config (BASELINE.json config #5)."""

import argparse
import os
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from examples import common  # noqa: E402


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--vocab-size", type=int, default=30522)
    p.add_argument("--seq-len", type=int, default=384)
    p.add_argument("--optimizer", default="sgd", choices=["sgd", "adamw"])
    common.add_common_args(p)
    p.set_defaults(batch_size=4, base_lr=0.001, damping=0.003,
                   kfac_update_freq=10, fac_update_freq=10)
    args = p.parse_args()

    rank, world, local_rank, use_cuda = common.initialize_distributed()
    torch.manual_seed(args.seed)
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")

    from kfac_pytorch_amd.models import make_bert_base_squad
    model = make_bert_base_squad(vocab_size=args.vocab_size).to(device)
    ddp_model = model
    if world > 1:
        ddp_model = torch.nn.parallel.DistributedDataParallel(
            model, device_ids=[local_rank] if use_cuda else None)

    precond = None
    if args.optimizer == "adamw":
        optimizer = torch.optim.AdamW(model.parameters(), lr=3e-5)
    else:
        optimizer = torch.optim.SGD(model.parameters(),
                                    lr=args.base_lr * world,
                                    momentum=args.momentum)
        if args.kfac_name != "none":
            import kfac_pytorch_amd as kfac
            KFAC = kfac.get_kfac_module(args.kfac_name)
            precond = KFAC(model, lr=args.base_lr * world,
                           damping=args.damping,
                           fac_update_freq=args.fac_update_freq,
                           kfac_update_freq=args.kfac_update_freq,
                           kl_clip=args.kl_clip,
                           factor_decay=args.factor_decay,
                           exclude_vocabulary_size=args.vocab_size,
                           exclude_parts=args.exclude_parts)

    g = torch.Generator().manual_seed(args.seed + rank)
    for epoch in range(args.epochs):
        model.train()
        iter_times = []
        last = time.perf_counter()
        for it in range(args.iters_per_epoch):
            ids = torch.randint(0, args.vocab_size,
                                (args.batch_size, args.seq_len),
                                generator=g).to(device)
            starts = torch.randint(0, args.seq_len, (args.batch_size,),
                                   generator=g).to(device)
            ends = torch.randint(0, args.seq_len, (args.batch_size,),
                                 generator=g).to(device)
            optimizer.zero_grad(set_to_none=False)
            if args.dtype == "bf16" and use_cuda:
                with torch.autocast("cuda", dtype=torch.bfloat16):
                    s_logits, e_logits = ddp_model(ids)
                    loss = (F.cross_entropy(s_logits, starts) +
                            F.cross_entropy(e_logits, ends)) / 2
            else:
                s_logits, e_logits = ddp_model(ids)
                loss = (F.cross_entropy(s_logits, starts) +
                        F.cross_entropy(e_logits, ends)) / 2
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
