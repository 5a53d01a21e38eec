"""Fused Adam/AdamW optimizer (HIP multi-tensor, gfx950).

Replaces ``apex.optimizers.FusedAdam`` (reference call sites:
src/optimization.py:25, run_squad.py:982-988 [bias_correction=False],
run_ner.py:243-244). ``adam_w_mode=True`` gives decoupled weight decay
inside the fused update. Kernel source: csrc/optim/multi_tensor.hip.
"""

from __future__ import annotations

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
        master_weights: bool = False,
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
        from .lamb import _restore_fp32_state  # noqa: PLC0415

        super().load_state_dict(state_dict)
        _restore_fp32_state(self, state_dict,
                            ("exp_avg", "exp_avg_sq", "master"))

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
            work_p, grads, ms, vs = [], [], [], []
            lo_p, lo_masters, lo_g32, lo_grads = [], [], [], []
            for p in params:
                state = self.state[p]
                use_master = (
                    self.master_weights and p.dtype != torch.float32
                )
                if "exp_avg" not in state:
                    if use_master:
                        state["master"] = p.detach().float().contiguous()
                        state["grad32"] = torch.empty_like(state["master"])
                    w = state.get("master", p)
                    state["exp_avg"] = torch.zeros_like(w, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(w, dtype=torch.float32)
                state["step"] = step
                if use_master:
                    if "master" not in state:
                        # resume from a checkpoint saved without masters
                        # (e.g. fp32 run switched to --pure_bf16): lazily
                        # seed the master from the param, like FusedLAMB
                        state["master"] = p.detach().float().contiguous()
                    if "grad32" not in state:
                        state["grad32"] = torch.empty_like(state["master"])
                    lo_p.append(p)
                    lo_masters.append(state["master"])
                    lo_g32.append(state["grad32"])
                    lo_grads.append(p.grad)
                    work_p.append(state["master"])
                    grads.append(state["grad32"])
                else:
                    work_p.append(p)
                    grads.append(p.grad)
                ms.append(state["exp_avg"])
                vs.append(state["exp_avg_sq"])
            if lo_g32:
                torch._foreach_copy_(lo_g32, lo_grads)  # bf16 -> fp32
            beta1, beta2 = group["betas"]
            if _use_native(work_p):
                ops.extension().fused_adam(
                    work_p, grads, ms, vs,
                    float(group["lr"]), beta1, beta2, group["eps"],
                    group["weight_decay"], step,
                    bool(group["bias_correction"]), bool(self.adam_w_mode),
                )
            else:
                self._eager_group_work(work_p, grads, group, step)
            if lo_p:
                torch._foreach_copy_(lo_p, lo_masters)  # fp32 -> bf16
        return loss

    def _eager_group_work(self, work_p: List[torch.Tensor],
                          grads: List[torch.Tensor], group: dict, step: int):
        """Eager update over the working (fp32-master-or-param) lists;
        moments are matched positionally via the caller's ms/vs order."""
        beta1, beta2 = group["betas"]
        bc1 = 1.0 - beta1**step if group["bias_correction"] else 1.0
        bc2 = 1.0 - beta2**step if group["bias_correction"] else 1.0
        wd = group["weight_decay"]
        states = [s for s in self._iter_states(work_p)]
        for p, g_t, state in zip(work_p, grads, states):
            g = g_t.float()
            m, v = state["exp_avg"], state["exp_avg_sq"]
            if not self.adam_w_mode and wd != 0.0:
                g = g + wd * p.float()  # L2 mode: decay into gradient
            m.mul_(beta1).add_(g, alpha=1 - beta1)
            v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
            update = (m / bc1) / ((v / bc2).sqrt() + group["eps"])
            if self.adam_w_mode and wd != 0.0:
                update = update + wd * p.float()
            p.add_(update.to(p.dtype), alpha=-group["lr"])

    def _iter_states(self, work_p):
        # work_p entries are either the param itself or its fp32 master;
        # map back to the owning param's state dict
        masters = {id(s.get("master")): s for s in self.state.values()
                   if isinstance(s, dict) and "master" in s}
        for w in work_p:
            if w in self.state:
                yield self.state[w]
            else:
                yield masters[id(w)]
