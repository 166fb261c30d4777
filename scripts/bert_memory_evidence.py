"""BERT-scale evidence for BASELINE config #5: peak HBM, per-step
time and the big-factor eigensolve cost of eigen_dp K-FAC on the
BERT-base SQuAD shape (bs 4, seq 384, vocab excluded -- reference:
examples/pytorch_squad_bert.py:394,450, batch.sh:31-32).

    python scripts/bert_memory_evidence.py [--steps 8]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import torch
import torch.distributed as dist
import torch.nn.functional as F


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--batch-size", type=int, default=4)
    ap.add_argument("--seq-len", type=int, default=384)
    args = ap.parse_args()

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29763")
        dist.init_process_group("gloo", world_size=1, rank=0,
                                init_method="env://")
    import kfac_pytorch_amd.backend as backend
    backend.init("Torch")
    import kfac_pytorch_amd as kfac
    from kfac_pytorch_amd.models.bert import make_bert_base_squad

    assert torch.cuda.is_available(), "GPU evidence script"
    dev = torch.device("cuda")
    torch.manual_seed(0)
    model = make_bert_base_squad().to(dev)
    pre = kfac.KFAC_EIGEN_DP(model, damping=0.002,
                             exclude_vocabulary_size=30522)
    opt = torch.optim.AdamW(model.parameters(), lr=3e-5)

    bs, sl = args.batch_size, args.seq_len
    ids = torch.randint(0, 30522, (bs, sl), device=dev)
    starts = torch.randint(0, sl, (bs,), device=dev)
    ends = torch.randint(0, sl, (bs,), device=dev)

    torch.cuda.reset_peak_memory_stats()
    os.environ["KFAC_PHASE_TIMING"] = "1"
    times = []
    for step in range(args.steps):
        t0 = time.perf_counter()
        opt.zero_grad(set_to_none=False)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            s_logits, e_logits = model(ids)
            loss = (F.cross_entropy(s_logits, starts)
                    + F.cross_entropy(e_logits, ends))
        loss.backward()
        pre.step()
        opt.step()
        torch.cuda.synchronize()
        times.append(time.perf_counter() - t0)
        if step == args.steps - 1 and hasattr(pre, "phase_times"):
            tot = {k: v / args.steps * 1e3
                   for k, v in pre.phase_times.items()}
            print("phase ms/step:",
                  {k: round(v, 1) for k, v in tot.items()}, flush=True)

    # per-factor eigensolve cost at the BERT dims (3072/768 + vocab-free)
    from kfac_pytorch_amd.ops.factors import factor_dims
    dims = sorted({factor_dims(m) for m in pre.modules}, reverse=True)
    peak = torch.cuda.max_memory_allocated() / 2**30
    total = torch.cuda.get_device_properties(0).total_memory / 2**30
    n_steps = len(times[2:]) or 1
    print(f"BERT-base SQuAD shape bs={bs} seq={sl}: "
          f"{sum(times[2:]) / n_steps * 1e3:.1f} ms/step "
          f"({len(pre.modules)} hooked layers, factor dims {dims[:6]}...)",
          flush=True)
    print(f"peak HBM {peak:.2f} GiB of {total:.0f} GiB "
          f"({peak / total * 100:.1f}% -- headroom "
          f"{total - peak:.0f} GiB)", flush=True)

    import json
    print(json.dumps({"bert_ms_per_step":
                      round(sum(times[2:]) / n_steps * 1e3, 1),
                      "peak_hbm_gib": round(peak, 2),
                      "total_hbm_gib": round(total, 1)}), flush=True)


if __name__ == "__main__":
    main()
