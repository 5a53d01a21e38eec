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


def _restore_fp32_state(optimizer, state_dict, keys) -> None:
    """Re-load the named per-param state entries as fp32 (undoing the
    param-dtype cast torch's Optimizer.load_state_dict applies)."""
    id_map = {}
    for saved_g, g in zip(state_dict["param_groups"], optimizer.param_groups):
        for old_id, p in zip(saved_g["params"], g["params"]):
            id_map[old_id] = p
    for old_id, saved in state_dict["state"].items():
        p = id_map.get(old_id)
        if p is None or p.dtype == torch.float32:
            continue
        for k in keys:
            t = saved.get(k)
            if torch.is_tensor(t):
                optimizer.state[p][k] = t.to(
                    device=p.device, dtype=torch.float32
                )


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
        master_weights: bool = False,
    ):
        """``master_weights=True`` keeps an fp32 master copy (plus a
        persistent fp32 grad buffer) for every non-fp32 parameter: the
        update runs entirely in fp32 and the bf16/fp16 parameter is the
        cast-down of the master. This is the pure-bf16 training mode —
        the model holds bf16 weights (no per-microbatch autocast weight
        casts, bf16 gradient all-reduce) while LAMB math stays fp32."""
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
        self.master_weights = master_weights

    def zero_grad(self, set_to_none: bool | None = None):  # noqa: D102
        if set_to_none is None:
            set_to_none = self.set_grad_none
        super().zero_grad(set_to_none=set_to_none)

    def state_dict(self):  # noqa: D102
        sd = super().state_dict()
        # drop per-step scratch WITHOUT mutating the live state (torch
        # packs references to the optimizer's state dicts, not copies)
        sd["state"] = {
            k: {kk: vv for kk, vv in v.items() if kk != "grad32"}
            for k, v in sd["state"].items()
        }
        return sd

    def load_state_dict(self, state_dict):  # noqa: D102
        super().load_state_dict(state_dict)
        # torch's Optimizer.load_state_dict casts floating state to the
        # param dtype — which would silently truncate fp32 moments and
        # masters to bf16 for low-precision params. Restore them from
        # the ORIGINAL state_dict tensors.
        _restore_fp32_state(self, state_dict,
                            ("exp_avg", "exp_avg_sq", "master"))

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

    def _master_lists(self, params):
        """(work_params, work_grads, cast_back) for one group: fp32
        params pass through; low-precision params get fp32 master +
        persistent fp32 grad buffers (batched _foreach_ casts)."""
        work_p, work_g = [], []
        lo_p, lo_masters, lo_g32, lo_grads = [], [], [], []
        for p in params:
            state = self.state[p]
            if p.dtype == torch.float32:
                work_p.append(p)
                work_g.append(p.grad)
                continue
            if "master" not in state:
                state["master"] = p.detach().float().contiguous()
            if "grad32" not in state:  # scratch; absent after resume
                state["grad32"] = torch.empty_like(state["master"])
            lo_p.append(p)
            lo_masters.append(state["master"])
            lo_g32.append(state["grad32"])
            lo_grads.append(p.grad)
            work_p.append(state["master"])
            work_g.append(state["grad32"])
        if lo_g32:
            torch._foreach_copy_(lo_g32, lo_grads)  # bf16 -> fp32 cast

        def cast_back():
            if lo_p:
                torch._foreach_copy_(lo_p, lo_masters)  # fp32 -> bf16

        return work_p, work_g, cast_back

    # -- native (HIP) ----------------------------------------------------
    def _step_native(self, all_grads):
        ext = ops.extension()
        max_grad_norm = self.param_groups[0]["max_grad_norm"]

        per_group = []
        norm_grads = []
        for group in self.param_groups:
            params = [p for p in group["params"] if p.grad is not None]
            if not params:
                continue
            if self.master_weights:
                work_p, work_g, cast_back = self._master_lists(params)
            else:
                work_p = params
                work_g = [p.grad for p in params]
                cast_back = None
            per_group.append((group, work_p, work_g, cast_back))
            norm_grads.extend(work_g)

        gnorm_sq = ext.multi_tensor_l2norm_sq(norm_grads)  # [1] f32 device
        for group, work_p, work_g, cast_back in per_group:
            group["step"] = group.get("step", 0) + 1
            step = group["step"]
            ms, vs = [], []
            for p, wp in zip(
                [p for p in group["params"] if p.grad is not None], work_p
            ):
                state = self.state[p]
                if "exp_avg" not in state:
                    state["exp_avg"] = torch.zeros_like(
                        wp, dtype=torch.float32
                    )
                    state["exp_avg_sq"] = torch.zeros_like(
                        wp, dtype=torch.float32
                    )
                state["step"] = step
                ms.append(state["exp_avg"])
                vs.append(state["exp_avg_sq"])
            beta1, beta2 = group["betas"]
            wd = group["weight_decay"]
            use_ratio = self.use_nvlamb or wd != 0.0
            ext.fused_lamb(
                work_p, work_g, ms, vs, gnorm_sq,
                float(group["lr"]), beta1, beta2, group["eps"], wd,
                step, bool(group["bias_correction"]),
                bool(group["grad_averaging"]), float(max_grad_norm),
                bool(use_ratio),
            )
            if cast_back is not None:
                cast_back()

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
                use_master = (
                    self.master_weights and p.dtype != torch.float32
                )
                if "exp_avg" not in state:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                    if use_master:
                        state["master"] = p.detach().float()
                state["step"] = step
                w = state["master"] if use_master else p
                g = p.grad.float() * clip_scale
                m, v = state["exp_avg"], state["exp_avg_sq"]
                m.mul_(beta1).add_(g, alpha=(1 - beta1) if group["grad_averaging"] else 1.0)
                v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                update = (m / bc1) / ((v / bc2).sqrt() + group["eps"])
                if wd != 0.0:
                    update = update + wd * w.float()
                ratio = 1.0
                if use_ratio:
                    wn = float(w.float().norm())
                    un = float(update.norm())
                    if wn > 0 and un > 0:
                        ratio = wn / un
                if use_master:
                    w.add_(update, alpha=-group["lr"] * ratio)
                    p.copy_(w.to(p.dtype))
                else:
                    p.add_(update.to(p.dtype), alpha=-group["lr"] * ratio)
