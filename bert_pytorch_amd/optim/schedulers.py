"""Warmup LR schedulers (reference semantics: src/schedulers.py:21-158).

All schedulers scale each group's base LR by a factor of
``progress = current_step / total_steps``:

* warmup region (progress < warmup): linear ramp ``progress / warmup``.
* after warmup: constant / linear-to-zero / cosine / polynomial decay.

``LinearWarmUpScheduler`` and ``PolyWarmUpScheduler`` read the true
optimizer step from ``optimizer.param_groups[0]['step']`` when present
(kept by FusedLAMB/FusedAdam/BertAdam here) so resume and the two-phase
hand-off track the optimizer, not the scheduler's own call count.
"""

from __future__ import annotations

import math
from typing import List

import torch


class LRScheduler:
    def __init__(self, optimizer: torch.optim.Optimizer, last_epoch: int = -1):
        if not isinstance(optimizer, torch.optim.Optimizer):
            raise TypeError(f"{type(optimizer).__name__} is not an Optimizer")
        self.optimizer = optimizer
        for group in optimizer.param_groups:
            group.setdefault("initial_lr", group["lr"])
        self.base_lrs = [g["initial_lr"] for g in optimizer.param_groups]
        self.last_epoch = last_epoch
        self.step()

    def _current_step(self) -> int:
        return self.last_epoch + 1

    def get_lr(self) -> List[float]:  # pragma: no cover - abstract
        raise NotImplementedError

    def step(self, epoch: int | None = None) -> None:
        self.last_epoch = epoch if epoch is not None else self.last_epoch + 1
        for group, lr in zip(self.optimizer.param_groups, self.get_lr()):
            group["lr"] = lr

    def state_dict(self) -> dict:
        return {
            k: v for k, v in self.__dict__.items() if k != "optimizer"
        }

    def load_state_dict(self, state: dict) -> None:
        self.__dict__.update(state)


class _OptStepScheduler(LRScheduler):
    """Track optimizer.param_groups[0]['step'] (reference behavior for
    the Linear/Poly schedulers used in pretraining)."""

    def step(self, epoch: int | None = None) -> None:
        group0 = self.optimizer.param_groups[0]
        if "step" in group0:
            self.last_epoch = group0["step"] + 1
        else:
            self.last_epoch = 1
        for group, lr in zip(self.optimizer.param_groups, self.get_lr()):
            group["lr"] = lr


class ConstantWarmUpScheduler(LRScheduler):
    def __init__(self, optimizer, warmup: float, total_steps: int, last_epoch=-1):
        self.warmup = warmup
        self.total_steps = total_steps
        super().__init__(optimizer, last_epoch)

    def get_lr(self):
        progress = self.last_epoch / self.total_steps
        if progress < self.warmup:
            return [lr * progress / self.warmup for lr in self.base_lrs]
        return list(self.base_lrs)


class CosineWarmUpScheduler(LRScheduler):
    def __init__(self, optimizer, warmup: float, total_steps: int, last_epoch=-1):
        self.warmup = warmup
        self.total_steps = total_steps
        super().__init__(optimizer, last_epoch)

    def get_lr(self):
        progress = self.last_epoch / self.total_steps
        if progress < self.warmup:
            return [lr * progress / self.warmup for lr in self.base_lrs]
        return [
            lr * (0.5 * (1.0 + math.cos(math.pi + progress)))
            for lr in self.base_lrs
        ]


class LinearWarmUpScheduler(_OptStepScheduler):
    def __init__(self, optimizer, warmup: float, total_steps: int, last_epoch=-1):
        self.warmup = warmup
        self.total_steps = total_steps
        super().__init__(optimizer, last_epoch)

    def get_lr(self):
        progress = self.last_epoch / self.total_steps
        if progress < self.warmup:
            return [lr * progress / self.warmup for lr in self.base_lrs]
        return [
            lr * max((progress - 1.0) / (self.warmup - 1.0), 0.0)
            for lr in self.base_lrs
        ]


class PolyWarmUpScheduler(_OptStepScheduler):
    def __init__(
        self, optimizer, warmup: float, total_steps: int, degree: float = 0.5,
        last_epoch=-1,
    ):
        self.warmup = warmup
        self.total_steps = total_steps
        self.degree = degree
        super().__init__(optimizer, last_epoch)

    def get_lr(self):
        progress = self.last_epoch / self.total_steps
        if progress < self.warmup:
            return [lr * progress / self.warmup for lr in self.base_lrs]
        return [lr * ((1.0 - progress) ** self.degree) for lr in self.base_lrs]


def warmup_exp_decay_exp(
    global_step: int,
    decay_rate: float,
    decay_steps: int,
    total_steps: int,
    warmup: float = 0.002,
    degree: float = 2.0,
) -> float:
    """LambdaLR factor used by the NER runner (reference: :144-158)."""
    x = global_step / total_steps
    warmup_end = warmup * total_steps
    if warmup == 0.0:
        return 1.0
    if x < warmup:
        return (x / warmup) ** degree
    return decay_rate ** ((global_step - warmup_end) / decay_steps)
