"""Training-harness utilities (reference: examples/utils.py).

Metric averaging across ranks, rank-0 checkpointing, warmup/step and
polynomial LR schedules, label smoothing, and the phase timers the
reference prints per-iteration time breakdowns with
(reference: examples/pytorch_cifar10_resnet.py:289-339)."""

from __future__ import annotations

import os
import time
from typing import Dict, List, Optional

import torch
import torch.nn.functional as F

import kfac_pytorch_amd.parallel.comm as comm_mod


class Metric:
    """Running average of a scalar, allreduce-averaged across ranks
    (reference: examples/utils.py:39-52)."""

    def __init__(self, name: str):
        self.name = name
        self.total = torch.zeros(1)
        self.n = 0

    def update(self, val, n: int = 1):
        if not torch.is_tensor(val):
            val = torch.tensor([float(val)])
        val = val.detach().cpu().reshape(1).float()
        if comm_mod.is_initialized() and comm_mod.get_comm().size() > 1:
            comm = comm_mod.get_comm()
            comm.allreduce(val, op=comm.Average)
        self.total += val
        self.n += n

    @property
    def avg(self) -> float:
        return (self.total / max(self.n, 1)).item()


class LabelSmoothLoss(torch.nn.Module):
    """Cross-entropy with label smoothing
    (reference: examples/utils.py:20-32)."""

    def __init__(self, smoothing: float = 0.0):
        super().__init__()
        self.smoothing = smoothing

    def forward(self, input, target):
        log_prob = F.log_softmax(input, dim=-1)
        weight = input.new_ones(input.size()) * \
            (self.smoothing / (input.size(-1) - 1.0))
        weight.scatter_(-1, target.unsqueeze(-1), 1.0 - self.smoothing)
        return (-weight * log_prob).sum(dim=-1).mean()


def create_lr_schedule(workers: int, warmup_epochs: float,
                       decay_schedule: List[int], alpha: float = 0.1):
    """Linear warmup to scaled LR then multiplicative decay at the listed
    epochs (reference: examples/utils.py:54-66). Returns an epoch->factor
    callable for LambdaLR."""
    decay_schedule = sorted(decay_schedule, reverse=True)

    def lr_schedule(epoch):
        lr_adj = 1.0
        if epoch < warmup_epochs:
            lr_adj = 1.0 / workers * (epoch * (workers - 1) /
                                      warmup_epochs + 1)
        else:
            for e in decay_schedule:
                if epoch >= e:
                    lr_adj *= alpha
        return lr_adj

    return lr_schedule


def polynomial_decay_lr(warmup_epochs: float, total_epochs: int,
                        power: float = 2.0):
    """Linear warmup then polynomial decay
    (reference: examples/utils.py:68-80)."""

    def lr_schedule(epoch):
        if epoch < warmup_epochs:
            return epoch / max(warmup_epochs, 1e-8)
        return (1.0 - (epoch - warmup_epochs) /
                max(total_epochs - warmup_epochs, 1e-8)) ** power

    return lr_schedule


def save_checkpoint(model, optimizer, filepath: str, epoch: int,
                    preconditioner=None, scheduler=None):
    """Rank-0 checkpoint save (reference: examples/utils.py:11-18).
    Unlike the reference, optionally persists K-FAC factor state so
    preconditioning resumes warm instead of from identity
    (reference gap: SURVEY.md S5 'Checkpoint / resume')."""
    if comm_mod.is_initialized() and comm_mod.get_comm().rank() != 0:
        return
    state = {
        "epoch": epoch,
        "model": model.state_dict(),
        "optimizer": optimizer.state_dict(),
    }
    if scheduler is not None:
        state["scheduler"] = scheduler.state_dict()
    if preconditioner is not None:
        state["kfac_steps"] = preconditioner.steps
    torch.save(state, filepath.format(epoch=epoch))


def load_checkpoint(model, optimizer, filepath: str, scheduler=None,
                    map_location="cpu") -> int:
    """Load a checkpoint; returns the stored epoch."""
    state = torch.load(filepath, map_location=map_location,
                       weights_only=False)
    model.load_state_dict(state["model"])
    optimizer.load_state_dict(state["optimizer"])
    if scheduler is not None and "scheduler" in state:
        scheduler.load_state_dict(state["scheduler"])
    return int(state.get("epoch", 0))


class PhaseTimers:
    """Wall-clock phase timers for the IO / FW+BW / COMM / KFAC / UPDATE
    breakdown (reference: examples/pytorch_cifar10_resnet.py:289-339).
    CUDA-synchronizing when a GPU is present so phases attribute
    correctly."""

    PHASES = ("io", "fwbw", "comm", "kfac", "update")

    def __init__(self, cuda: Optional[bool] = None):
        self.cuda = torch.cuda.is_available() if cuda is None else cuda
        self.times: Dict[str, List[float]] = {p: [] for p in self.PHASES}
        self._t0: Optional[float] = None
        self._phase: Optional[str] = None

    def start(self, phase: str):
        if self.cuda:
            torch.cuda.synchronize()
        self._phase = phase
        self._t0 = time.perf_counter()

    def stop(self):
        if self._phase is None:
            return
        if self.cuda:
            torch.cuda.synchronize()
        self.times[self._phase].append(time.perf_counter() - self._t0)
        self._phase = None

    def summary(self) -> Dict[str, float]:
        return {p: (sum(v) / len(v) if v else 0.0)
                for p, v in self.times.items()}

    def reset(self):
        for v in self.times.values():
            v.clear()

    def format(self) -> str:
        s = self.summary()
        total = sum(s.values())
        parts = " ".join(f"{p}={t * 1000:.1f}ms" for p, t in s.items())
        return f"iter={total * 1000:.1f}ms [{parts}]"


class MultiEpochsDataLoader(torch.utils.data.DataLoader):
    """DataLoader whose worker pool and iterator persist across epochs
    (capability analog of the reference's MultiEpochsDataLoader,
    examples/utils.py:93-121 -- avoids worker restart cost per epoch)."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._DataLoader__initialized = False
        self.batch_sampler = _RepeatSampler(self.batch_sampler)
        self._DataLoader__initialized = True
        self.iterator = super().__iter__()

    def __len__(self):
        return len(self.batch_sampler.sampler)

    def __iter__(self):
        for _ in range(len(self)):
            yield next(self.iterator)


class _RepeatSampler:
    """Endlessly repeating batch sampler (examples/utils.py:106-121)."""

    def __init__(self, sampler):
        self.sampler = sampler

    def __iter__(self):
        while True:
            yield from iter(self.sampler)


def generate_pseudo_labels(output: torch.Tensor) -> torch.Tensor:
    """Sample labels from the model's predictive distribution -- the
    Monte-Carlo Fisher ('F1mc') option (reference:
    examples/utils.py:83-90; --kfac-type arg at
    examples/pytorch_cifar10_resnet.py:74-75)."""
    probs = torch.softmax(output.detach(), dim=-1)
    return torch.multinomial(probs, 1).squeeze(-1)
