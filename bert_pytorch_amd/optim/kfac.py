"""K-FAC second-order preconditioner (MI355X-native, in-repo).

Replaces the reference's external ``kfac_pytorch`` dependency
(reference call sites: run_pretraining.py:30-34,321-355,405-409; ctor
params kfac_inv_interval=10, skip_layers=['BertLMPredictionHead',
'embedding'] at run_pretraining.py:146-149,136-137).

Per eligible linear layer l with input activations a [N, in] and
pre-activation output gradients g [N, out], maintains running Kronecker
factors

    A_l = E[a aT]   (bias handled by appending a ones column)
    G_l = E[g gT]   (per-sample scale: g is multiplied by N because the
                     loss is batch-mean reduced)

accumulated in forward/backward hooks each training micro-batch, and
every ``inv_update_interval`` optimizer steps recomputes damped inverses

    A_l^-1 = (A_l + sqrt(damping) I)^-1,  G_l^-1 = (G_l + sqrt(damping) I)^-1

via symmetric eigendecomposition. ``step()`` preconditions the gradients
in-place:  grad_l  <-  nu * G_l^-1 grad_l A_l^-1   with the K-FAC KL
clip  nu = min(1, sqrt(kl_clip / sum_l lr^2 * <precond_l, grad_l>)).

Distributed strategy (RCCL over xGMI; differs deliberately from the
reference's HYBRID_OPT worker fractions): factors are all-reduce
averaged across ranks right before each inverse recompute, the
per-layer eigendecompositions are round-robin assigned to ranks, and
the resulting inverses are broadcast — one collective burst every
``inv_update_interval`` steps instead of per-step gradient-worker
traffic, which suits xGMI's 7 fat point-to-point links.
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional, Sequence

import torch
import torch.distributed as dist
from torch import nn

__all__ = ["KFAC"]

_DEFAULT_SKIP = ("BertLMPredictionHead", "embedding")


def _is_dist() -> bool:
    return dist.is_available() and dist.is_initialized()


class _LayerState:
    __slots__ = ("module", "name", "has_bias", "A", "G", "A_inv", "G_inv")

    def __init__(self, module: nn.Module, name: str, has_bias: bool):
        self.module = module
        self.name = name
        self.has_bias = has_bias
        self.A: Optional[torch.Tensor] = None
        self.G: Optional[torch.Tensor] = None
        self.A_inv: Optional[torch.Tensor] = None
        self.G_inv: Optional[torch.Tensor] = None

    def weight(self) -> nn.Parameter:
        mod = self.module
        return mod.qkv_weight if hasattr(mod, "qkv_weight") else mod.weight

    def bias(self) -> Optional[nn.Parameter]:
        if not self.has_bias:
            return None
        mod = self.module
        return mod.qkv_bias if hasattr(mod, "qkv_bias") else mod.bias


class KFAC:
    """Hook-based distributed K-FAC gradient preconditioner.

    Eligible layers: plain ``nn.Linear`` and the packed QKV projection
    (the reference's kfac likewise only matched ``linear`` modules, so
    fused LinearActivation/embeddings/LM head stay first-order).
    """

    def __init__(
        self,
        model: nn.Module,
        optimizer: Optional[torch.optim.Optimizer] = None,
        factor_update_interval: int = 1,
        inv_update_interval: int = 10,
        damping: float = 0.003,
        factor_decay: float = 0.95,
        kl_clip: float = 0.001,
        lr: float = 1.0,
        skip_layers: Sequence[str] = _DEFAULT_SKIP,
        inv_dtype: torch.dtype = torch.float32,
    ):
        if hasattr(model, "module"):  # unwrap DDP
            model = model.module
        self.optimizer = optimizer
        self.factor_update_interval = max(1, int(factor_update_interval))
        self.inv_update_interval = max(1, int(inv_update_interval))
        self.damping = float(damping)
        self.factor_decay = float(factor_decay)
        self.kl_clip = float(kl_clip)
        self.lr = float(lr)
        self.inv_dtype = inv_dtype
        self._steps = 0
        self._hooks: List[torch.utils.hooks.RemovableHandle] = []
        self.layers: List[_LayerState] = []
        self._register(model, [s.lower() for s in skip_layers])

    # -- layer discovery ---------------------------------------------------

    def _register(self, model: nn.Module, skip: List[str]) -> None:
        skipped_roots: List[str] = []
        for name, mod in model.named_modules():
            cls = mod.__class__.__name__.lower()
            low = name.lower()
            if any(s in cls or s in low.split(".")[-1] for s in skip):
                skipped_roots.append(name)
        for name, mod in model.named_modules():
            if any(name == r or name.startswith(r + ".") for r in skipped_roots):
                continue
            if isinstance(mod, nn.Linear):
                has_bias = mod.bias is not None
            elif hasattr(mod, "qkv_weight"):
                has_bias = getattr(mod, "qkv_bias", None) is not None
            else:
                continue
            state = _LayerState(mod, name, has_bias)
            self.layers.append(state)
            self._hooks.append(mod.register_forward_hook(self._make_fwd_hook(state)))

    def _make_fwd_hook(self, state: _LayerState):
        def hook(mod: nn.Module, inputs, output):
            if not mod.training or not torch.is_grad_enabled():
                return
            if self._steps % self.factor_update_interval != 0:
                return
            a = inputs[0].detach()
            self._update_A(state, a)
            if isinstance(output, torch.Tensor) and output.requires_grad:
                output.register_hook(lambda g: self._update_G(state, g))

        return hook

    @torch.no_grad()
    def _update_A(self, state: _LayerState, a: torch.Tensor) -> None:
        a = a.reshape(-1, a.shape[-1]).float()
        if state.has_bias:
            ones = torch.ones(a.shape[0], 1, dtype=a.dtype, device=a.device)
            a = torch.cat([a, ones], dim=1)
        cov = a.t() @ a / a.shape[0]
        if state.A is None:
            state.A = cov
        else:
            state.A.mul_(self.factor_decay).add_(cov, alpha=1 - self.factor_decay)

    @torch.no_grad()
    def _update_G(self, state: _LayerState, g: torch.Tensor) -> None:
        g = g.reshape(-1, g.shape[-1]).float()
        n = g.shape[0]
        # batch-mean loss => per-sample grads are g*n; cov = (g n)T (g n)/n
        cov = g.t() @ g * n
        if state.G is None:
            state.G = cov
        else:
            state.G.mul_(self.factor_decay).add_(cov, alpha=1 - self.factor_decay)

    # -- inverses ----------------------------------------------------------

    @torch.no_grad()
    def _allreduce_factors(self) -> None:
        if not _is_dist() or dist.get_world_size() == 1:
            return
        world = dist.get_world_size()
        tensors = []
        for st in self.layers:
            if st.A is not None:
                tensors.append(st.A)
            if st.G is not None:
                tensors.append(st.G)
        if not tensors:
            return
        flat = torch.cat([t.reshape(-1) for t in tensors])
        dist.all_reduce(flat)
        flat.div_(world)
        off = 0
        for t in tensors:
            t.copy_(flat[off : off + t.numel()].view_as(t))
            off += t.numel()

    @torch.no_grad()
    def _damped_inverse(self, factor: torch.Tensor) -> torch.Tensor:
        pi = math.sqrt(self.damping)
        d, q = torch.linalg.eigh(
            factor + pi * torch.eye(factor.shape[0], device=factor.device)
        )
        d = d.clamp_min(1e-10)
        inv = (q / d) @ q.t()
        return inv.to(self.inv_dtype)

    @torch.no_grad()
    def _update_inverses(self) -> None:
        self._allreduce_factors()
        world = dist.get_world_size() if _is_dist() else 1
        rank = dist.get_rank() if _is_dist() else 0
        for i, st in enumerate(self.layers):
            if st.A is None or st.G is None:
                continue
            owner = i % world
            if rank == owner:
                st.A_inv = self._damped_inverse(st.A)
                st.G_inv = self._damped_inverse(st.G)
            elif st.A_inv is None:
                st.A_inv = torch.empty_like(st.A, dtype=self.inv_dtype)
                st.G_inv = torch.empty_like(st.G, dtype=self.inv_dtype)
            if world > 1:
                dist.broadcast(st.A_inv, src=owner)
                dist.broadcast(st.G_inv, src=owner)

    # -- precondition ------------------------------------------------------

    @torch.no_grad()
    def step(self) -> None:
        """Precondition ``p.grad`` in place. Call after unscale_, before
        ``optimizer.step()`` (reference: run_pretraining.py:405-409)."""
        if self._steps % self.inv_update_interval == 0:
            self._update_inverses()
        self._steps += 1

        lr = self.lr
        if self.optimizer is not None:
            lr = float(self.optimizer.param_groups[0]["lr"])

        updates = []
        vg_sum = torch.zeros((), dtype=torch.float32)
        for st in self.layers:
            w = st.weight()
            if w.grad is None or st.A_inv is None:
                continue
            gw = w.grad.float()
            b = st.bias()
            if b is not None and b.grad is not None:
                m = torch.cat([gw, b.grad.float().unsqueeze(1)], dim=1)
            else:
                m = gw
            v = st.G_inv.float() @ m @ st.A_inv.float()
            vg_sum = vg_sum.to(v.device) + (v * m).sum() * (lr * lr)
            updates.append((st, m, v))

        if not updates:
            return
        nu = min(1.0, math.sqrt(self.kl_clip / max(float(vg_sum), 1e-20)))
        for st, _m, v in updates:
            w = st.weight()
            b = st.bias()
            if b is not None and b.grad is not None:
                w.grad.copy_((v[:, :-1] * nu).to(w.grad.dtype))
                b.grad.copy_((v[:, -1] * nu).to(b.grad.dtype))
            else:
                w.grad.copy_((v * nu).to(w.grad.dtype))

    # -- state -------------------------------------------------------------

    def state_dict(self) -> Dict:
        layers = {}
        for st in self.layers:
            layers[st.name] = {
                "A": st.A,
                "G": st.G,
                "A_inv": st.A_inv,
                "G_inv": st.G_inv,
            }
        return {"steps": self._steps, "layers": layers}

    def load_state_dict(self, state: Dict) -> None:
        self._steps = int(state.get("steps", 0))
        saved = state.get("layers", {})
        for st in self.layers:
            if st.name not in saved:
                continue
            entry = saved[st.name]
            dev = st.weight().device
            for attr in ("A", "G", "A_inv", "G_inv"):
                t = entry.get(attr)
                setattr(st, attr, t.to(dev) if t is not None else None)

    def remove_hooks(self) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks.clear()
