"""Fused Adam/AdamW optimizer (HIP multi-tensor, gfx950).

Replaces ``apex.optimizers.FusedAdam`` (reference call sites:
src/optimization.py:25, run_squad.py:982-988 [bias_correction=False],
run_ner.py:243-244). ``adam_w_mode=True`` gives decoupled weight decay
inside the fused update. Kernel source: csrc/optim/multi_tensor.hip.
"""

from __future__ import annotations

import os
from typing import List

import torch
from torch.optim import Optimizer

from .. import ops
from .lamb import _use_native


class FusedAdam(Optimizer):
    def __init__(
        self,
        params,
        lr: float = 1e-3,
        bias_correction: bool = True,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        adam_w_mode: bool = True,
        weight_decay: float = 0.0,
        amsgrad: bool = False,
        set_grad_none: bool = True,
    ):
        if amsgrad:
            raise RuntimeError("FusedAdam does not support amsgrad")
        defaults = dict(
            lr=lr, bias_correction=bias_correction, betas=betas, eps=eps,
            weight_decay=weight_decay,
        )
        super().__init__(params, defaults)
        self.adam_w_mode = adam_w_mode
        self.set_grad_none = set_grad_none

    def zero_grad(self, set_to_none: bool | None = None):  # noqa: D102
        if set_to_none is None:
            set_to_none = self.set_grad_none
        super().zero_grad(set_to_none=set_to_none)

    def load_state_dict(self, state_dict):  # noqa: D102
        from .lamb import _restore_fp32_state  # noqa: PLC0415

        super().load_state_dict(state_dict)
        _restore_fp32_state(self, state_dict, ("exp_avg", "exp_avg_sq"))

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            params = [p for p in group["params"] if p.grad is not None]
            if not params:
                continue
            group["step"] = group.get("step", 0) + 1
            step = group["step"]
            grads, ms, vs = [], [], []
            for p in params:
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                state["step"] = step
                grads.append(p.grad)
                ms.append(state["exp_avg"])
                vs.append(state["exp_avg_sq"])
            beta1, beta2 = group["betas"]
            if _use_native(params):
                ops.extension().fused_adam(
                    params, grads, ms, vs,
                    float(group["lr"]), beta1, beta2, group["eps"],
                    group["weight_decay"], step,
                    bool(group["bias_correction"]), bool(self.adam_w_mode),
                )
            else:
                self._eager_group(params, group, step)
        return loss

    def _eager_group(self, params: List[torch.Tensor], group: dict, step: int):
        beta1, beta2 = group["betas"]
        bc1 = 1.0 - beta1**step if group["bias_correction"] else 1.0
        bc2 = 1.0 - beta2**step if group["bias_correction"] else 1.0
        wd = group["weight_decay"]
        for p in params:
            state = self.state[p]
            g = p.grad.float()
            m, v = state["exp_avg"], state["exp_avg_sq"]
            if not self.adam_w_mode and wd != 0.0:
                g = g + wd * p.float()  # L2 mode: decay into gradient
            m.mul_(beta1).add_(g, alpha=1 - beta1)
            v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
            update = (m / bc1) / ((v / bc2).sqrt() + group["eps"])
            if self.adam_w_mode and wd != 0.0:
                update = update + wd * p.float()
            p.add_(update.to(p.dtype), alpha=-group["lr"])
