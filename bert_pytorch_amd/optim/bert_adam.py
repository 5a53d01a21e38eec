"""BertAdam — fp32 Adam with decoupled weight decay, per-parameter grad
clipping and an internal warmup LR schedule.

Reference: src/optimization.py:36-174 (used by the fp32 SQuAD path).
No bias correction (BERT convention). Schedules mirror the reference's
``warmup_cosine`` / ``warmup_constant`` / ``warmup_linear``.
"""

from __future__ import annotations

import math

import torch
from torch.optim import Optimizer


def warmup_cosine(x: float, warmup: float = 0.002) -> float:
    if x < warmup:
        return x / warmup
    return 0.5 * (1.0 + math.cos(math.pi * x))


def warmup_constant(x: float, warmup: float = 0.002) -> float:
    if x < warmup:
        return x / warmup
    return 1.0


def warmup_linear(x: float, warmup: float = 0.002) -> float:
    if x < warmup:
        return x / warmup
    return max((x - 1.0) / (warmup - 1.0), 0.0)


SCHEDULES = {
    "warmup_cosine": warmup_cosine,
    "warmup_constant": warmup_constant,
    "warmup_linear": warmup_linear,
}


class BertAdam(Optimizer):
    def __init__(
        self,
        params,
        lr: float,
        warmup: float = -1,
        t_total: int = -1,
        schedule: str = "warmup_linear",
        b1: float = 0.9,
        b2: float = 0.999,
        e: float = 1e-6,
        weight_decay: float = 0.01,
        max_grad_norm: float = 1.0,
    ):
        if lr < 0.0:
            raise ValueError(f"invalid learning rate {lr}")
        if schedule not in SCHEDULES:
            raise ValueError(f"invalid schedule {schedule}")
        if not 0.0 <= warmup < 1.0 and warmup != -1:
            raise ValueError(f"invalid warmup {warmup}")
        defaults = dict(
            lr=lr, schedule=schedule, warmup=warmup, t_total=t_total,
            b1=b1, b2=b2, e=e, weight_decay=weight_decay,
            max_grad_norm=max_grad_norm,
        )
        super().__init__(params, defaults)

    def get_lr(self):
        lrs = []
        for group in self.param_groups:
            for p in group["params"]:
                state = self.state[p]
                if len(state) == 0:
                    return [0]
                if group["t_total"] != -1:
                    schedule_fct = SCHEDULES[group["schedule"]]
                    lrs.append(
                        group["lr"]
                        * schedule_fct(
                            state["step"] / group["t_total"], group["warmup"]
                        )
                    )
                else:
                    lrs.append(group["lr"])
        return lrs

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            group["step"] = group.get("step", 0) + 1
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad
                if grad.is_sparse:
                    raise RuntimeError("BertAdam does not support sparse gradients")
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["next_m"] = torch.zeros_like(p)
                    state["next_v"] = torch.zeros_like(p)
                m, v = state["next_m"], state["next_v"]
                beta1, beta2 = group["b1"], group["b2"]

                if group["max_grad_norm"] > 0:
                    torch.nn.utils.clip_grad_norm_([p], group["max_grad_norm"])

                m.mul_(beta1).add_(grad, alpha=1 - beta1)
                v.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
                update = m / (v.sqrt() + group["e"])
                if group["weight_decay"] > 0.0:
                    update = update + group["weight_decay"] * p

                if group["t_total"] != -1:
                    schedule_fct = SCHEDULES[group["schedule"]]
                    lr_scheduled = group["lr"] * schedule_fct(
                        state["step"] / group["t_total"], group["warmup"]
                    )
                else:
                    lr_scheduled = group["lr"]

                p.add_(update, alpha=-lr_scheduled)
                state["step"] += 1
        return loss
