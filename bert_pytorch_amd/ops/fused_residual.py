"""Fused bias + dropout + residual-add + LayerNorm (HIP, gfx950).

One kernel for the residual junction the reference spells as four ops
(``BertSelfOutput``/``BertOutput``: dense-bias add, dropout, residual
add, LayerNorm — src/modeling.py:432-443, 468-479). The dropout mask is
stored (uint8) so backward is exact. Kernel source:
csrc/ops/fused_residual.hip.
"""

from __future__ import annotations

from typing import Optional

import torch

from . import _reference, extension, use_native
from .rng import next_philox


class _FusedBiasDropoutResidualLN(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias, residual, weight, ln_bias, p, training, eps):
        ext = extension()
        x2d = x.contiguous().view(-1, x.shape[-1])
        r2d = residual.contiguous().view(-1, x.shape[-1])
        seed, offset = next_philox(x2d.numel()) if (training and p > 0) else (0, 0)
        # z = dropout(x + bias) + residual (saved for LN backward)
        y, z, mask, mean, rstd = ext.bias_dropout_residual_ln_fwd(
            x2d, bias, r2d, weight, ln_bias, p if training else 0.0, eps, seed, offset
        )
        ctx.save_for_backward(z, mask, weight, mean, rstd)
        ctx.p = p if training else 0.0
        ctx.has_bias = bias is not None
        ctx.shape = x.shape
        return y.view(x.shape)

    @staticmethod
    def backward(ctx, dy):
        ext = extension()
        z, mask, weight, mean, rstd = ctx.saved_tensors
        dy2d = dy.contiguous().view(-1, dy.shape[-1])
        dx, dbias, dz, dw, db = ext.bias_dropout_residual_ln_bwd(
            dy2d, z, mask, weight, mean, rstd, ctx.p, ctx.has_bias
        )
        return (
            dx.view(ctx.shape),
            dbias if ctx.has_bias else None,
            dz.view(ctx.shape),
            dw,
            db,
            None,
            None,
            None,
        )


def fused_bias_dropout_residual_ln(
    x: torch.Tensor,
    bias: Optional[torch.Tensor],
    residual: torch.Tensor,
    weight: torch.Tensor,
    ln_bias: torch.Tensor,
    p: float,
    training: bool,
    eps: float = 1e-12,
) -> torch.Tensor:
    if use_native(x):
        return _FusedBiasDropoutResidualLN.apply(
            x, bias, residual, weight, ln_bias, p, training, eps
        )
    return _reference.bias_dropout_residual_ln(
        x, bias, residual, weight, ln_bias, p, training, eps
    )
