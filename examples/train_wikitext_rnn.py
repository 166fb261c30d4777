#!/usr/bin/env python3
"""WikiText-2 LSTM language-model trainer with K-FAC.

Capability analog of the reference's RNN example
(reference: examples/pytorch_wikitext_rnn.py + wikitext_models.py).
The reference trainer calls a stale K-FAC API
(``kfac.KFAC(TCov=..., TInv=...)``, pytorch_wikitext_rnn.py:196-202)
and does not run against its own package; this one uses the current
API.  K-FAC preconditions the decoder Linear (LSTM weights stay
first-order, like the reference's intent); the vocab-sized decoder can
be excluded with ``--exclude-vocab``.

Synthetic WikiText-2-shaped token streams (vocab 33278) -- no network
for the real dataset in this environment.

Run:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 examples/train_wikitext_rnn.py --speed
"""

import argparse
import os
import sys

import torch
import torch.distributed as dist
import torch.nn as nn

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from examples.common import (add_common_args,  # noqa: E402
                             initialize_distributed, report_speed,
                             train_loop)


def parse_args():
    p = argparse.ArgumentParser(description=__doc__)
    add_common_args(p)
    p.add_argument("--vocab-size", type=int, default=33278)
    p.add_argument("--emb", type=int, default=256)
    p.add_argument("--hidden", type=int, default=256)
    p.add_argument("--layers", type=int, default=2)
    p.add_argument("--bptt", type=int, default=35,
                   help="sequence length (reference default)")
    p.add_argument("--exclude-vocab", dest="exclude_vocab",
                   action="store_true", default=True,
                   help="exclude the vocab-sized decoder from K-FAC "
                        "(default: a 33k-dim factor is not worth "
                        "eigendecomposing; reference LM trainers exclude "
                        "vocab projections too)")
    p.add_argument("--include-vocab", dest="exclude_vocab",
                   action="store_false")
    p.set_defaults(batch_size=20, base_lr=20.0 / 20, damping=0.003)
    return p.parse_args()


def main():
    args = parse_args()
    rank, world, local_rank, use_cuda = initialize_distributed()
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")
    torch.manual_seed(args.seed + rank)

    from kfac_pytorch_amd.models import LSTMLanguageModel
    model = LSTMLanguageModel(vocab_size=args.vocab_size, emb=args.emb,
                              hidden=args.hidden,
                              layers=args.layers).to(device)
    ddp = (torch.nn.parallel.DistributedDataParallel(
        model, device_ids=[local_rank] if use_cuda else None)
        if world > 1 else model)

    optimizer = torch.optim.SGD(model.parameters(),
                                lr=args.base_lr * world,
                                momentum=args.momentum)
    if args.exclude_vocab:
        args_vocab = args.vocab_size
    else:
        args_vocab = None
    import kfac_pytorch_amd as kfac
    pre = None
    sched = None
    if args.kfac_name != "none":
        KFAC = kfac.get_kfac_module(args.kfac_name)
        pre = KFAC(model, lr=args.base_lr * world, damping=args.damping,
                   fac_update_freq=args.fac_update_freq,
                   kfac_update_freq=args.kfac_update_freq,
                   kl_clip=args.kl_clip, factor_decay=args.factor_decay,
                   exclude_vocabulary_size=args_vocab,
                   exclude_parts=args.exclude_parts)

    vocab = args.vocab_size

    class TokenBatches:
        """Synthetic (data, target) token batches of WikiText shape."""

        def __init__(self, n):
            self.n = n

        def __len__(self):
            return self.n

        def __iter__(self):
            g = torch.Generator().manual_seed(args.seed + rank)
            for _ in range(self.n):
                x = torch.randint(0, vocab, (args.batch_size, args.bptt),
                                  generator=g)
                y = torch.randint(0, vocab, (args.batch_size, args.bptt),
                                  generator=g)
                yield x, y

    class LMLoss(nn.Module):
        def __init__(self):
            super().__init__()
            self.ce = nn.CrossEntropyLoss()

        def forward(self, out, target):
            logits = out[0] if isinstance(out, tuple) else out
            return self.ce(logits.reshape(-1, vocab), target.reshape(-1))

    for epoch in range(args.epochs):
        loss, iter_times = train_loop(
            model, ddp, optimizer, pre, LMLoss(),
            TokenBatches(args.iters_per_epoch), args, rank, use_cuda)
        if rank == 0 and not args.speed:
            print(f"epoch {epoch}: loss={loss:.4f}")
    report_speed(iter_times, args.batch_size, world, rank)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
