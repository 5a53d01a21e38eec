"""Fused LAMB optimizer (HIP multi-tensor, gfx950).

Replaces ``apex.optimizers.FusedLAMB`` (reference call site:
run_pretraining.py:39,295; kernels named at src/optimization.py:31-32).
Semantics: NVLAMB-style — global gradient-norm clipping to
``max_grad_norm``, Adam moments with bias correction, per-tensor trust
ratio ``||w|| / ||update||`` applied only to weight-decayed groups
(``use_nvlamb=False``), decoupled weight decay inside the update.

GPU path: three HIP kernels on one stream, no host synchronization —
(1) multi-tensor grad-norm² reduction, (2) stage1 moments+update+
per-tensor norms (update overwrites the grad buffer), (3) stage2
trust-ratio apply. Kernel source: csrc/optim/multi_tensor.hip.

The eager (CPU / reference) path below is the numerics oracle the HIP
kernels are tested against.
"""

from __future__ import annotations

import math
import os
from typing import List

import torch
from torch.optim import Optimizer

from .. import ops


def _use_native(params: List[torch.Tensor]) -> bool:
    if not params or not params[0].is_cuda:
        return False
    if os.environ.get("BPA_FORCE_EAGER") == "1":
        return False
    ops.extension()  # raises if missing: GPU runs must use HIP kernels
    return True


class FusedLAMB(Optimizer):
    def __init__(
        self,
        params,
        lr: float = 1e-3,
        bias_correction: bool = True,
        betas=(0.9, 0.999),
        eps: float = 1e-6,
        weight_decay: float = 0.01,
        amsgrad: bool = False,
        adam_w_mode: bool = True,
        grad_averaging: bool = True,
        set_grad_none: bool = True,
        max_grad_norm: float = 1.0,
        use_nvlamb: bool = False,
    ):
        if amsgrad:
            raise RuntimeError("FusedLAMB does not support amsgrad")
        defaults = dict(
            lr=lr,
            bias_correction=bias_correction,
            betas=betas,
            eps=eps,
            weight_decay=weight_decay,
            grad_averaging=grad_averaging,
            max_grad_norm=max_grad_norm,
        )
        super().__init__(params, defaults)
        self.adam_w_mode = adam_w_mode
        self.set_grad_none = set_grad_none
        self.use_nvlamb = use_nvlamb

    def zero_grad(self, set_to_none: bool | None = None):  # noqa: D102
        if set_to_none is None:
            set_to_none = self.set_grad_none
        super().zero_grad(set_to_none=set_to_none)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        all_grads = [
            p.grad for group in self.param_groups for p in group["params"]
            if p.grad is not None
        ]
        if not all_grads:
            return loss

        if _use_native(all_grads):
            self._step_native(all_grads)
        else:
            self._step_eager(all_grads)
        return loss

    # -- native (HIP) ----------------------------------------------------
    def _step_native(self, all_grads):
        ext = ops.extension()
        max_grad_norm = self.param_groups[0]["max_grad_norm"]
        gnorm_sq = ext.multi_tensor_l2norm_sq(all_grads)  # [1] f32 device
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
            wd = group["weight_decay"]
            use_ratio = self.use_nvlamb or wd != 0.0
            ext.fused_lamb(
                params, grads, ms, vs, gnorm_sq,
                float(group["lr"]), beta1, beta2, group["eps"], wd,
                step, bool(group["bias_correction"]),
                bool(group["grad_averaging"]), float(max_grad_norm),
                bool(use_ratio),
            )

    # -- eager reference -------------------------------------------------
    def _step_eager(self, all_grads):
        max_grad_norm = self.param_groups[0]["max_grad_norm"]
        gnorm = math.sqrt(
            sum(float(g.float().pow(2).sum()) for g in all_grads)
        )
        clip_scale = 1.0
        if max_grad_norm > 0 and gnorm > max_grad_norm:
            clip_scale = max_grad_norm / gnorm

        for group in self.param_groups:
            params = [p for p in group["params"] if p.grad is not None]
            if not params:
                continue
            group["step"] = group.get("step", 0) + 1
            step = group["step"]
            beta1, beta2 = group["betas"]
            bc1 = 1.0 - beta1**step if group["bias_correction"] else 1.0
            bc2 = 1.0 - beta2**step if group["bias_correction"] else 1.0
            wd = group["weight_decay"]
            use_ratio = self.use_nvlamb or wd != 0.0
            for p in params:
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                state["step"] = step
                g = p.grad.float() * clip_scale
                m, v = state["exp_avg"], state["exp_avg_sq"]
                m.mul_(beta1).add_(g, alpha=(1 - beta1) if group["grad_averaging"] else 1.0)
                v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                update = (m / bc1) / ((v / bc2).sqrt() + group["eps"])
                if wd != 0.0:
                    update = update + wd * p.float()
                ratio = 1.0
                if use_ratio:
                    wn = float(p.float().norm())
                    un = float(update.norm())
                    if wn > 0 and un > 0:
                        ratio = wn / un
                p.add_(update.to(p.dtype), alpha=-group["lr"] * ratio)
