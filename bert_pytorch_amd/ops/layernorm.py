"""Fused LayerNorm (HIP, gfx950).

Replaces Apex ``FusedLayerNormAffineFunction`` (reference:
src/modeling.py:299-335). Forward: one wave-per-row Welford-free two-pass
reduction in registers/DPP; backward: fused dx + two-stage partial
reduction for dgamma/dbeta. Kernel source: csrc/ops/layernorm.hip.
"""

from __future__ import annotations

import torch

from . import _reference, extension, use_native


class _FusedLayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        ext = extension()
        x2d = x.contiguous().view(-1, x.shape[-1])
        y, mean, rstd = ext.ln_fwd(x2d, weight, bias, eps)
        ctx.save_for_backward(x2d, weight, mean, rstd)
        ctx.shape = x.shape
        return y.view(x.shape)

    @staticmethod
    def backward(ctx, dy):
        ext = extension()
        x2d, weight, mean, rstd = ctx.saved_tensors
        dy2d = dy.contiguous().view(-1, dy.shape[-1])
        dx, dw, db = ext.ln_bwd(dy2d, x2d, weight, mean, rstd)
        return dx.view(ctx.shape), dw, db, None


def fused_layer_norm(
    x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor, eps: float = 1e-12
) -> torch.Tensor:
    if use_native(x):
        return _FusedLayerNorm.apply(x, weight, bias, eps)
    return _reference.layer_norm(x, weight, bias, eps)


class FusedLayerNorm(torch.nn.Module):
    """Drop-in ``BertLayerNorm`` (same parameter names: weight, bias)."""

    def __init__(self, hidden_size: int, eps: float = 1e-12) -> None:
        super().__init__()
        self.weight = torch.nn.Parameter(torch.ones(hidden_size))
        self.bias = torch.nn.Parameter(torch.zeros(hidden_size))
        self.eps = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return fused_layer_norm(x, self.weight, self.bias, self.eps)
