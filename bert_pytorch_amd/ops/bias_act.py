"""Fused bias + exact-erf GELU (HIP, gfx950).

Replaces the reference's jit-scripted ``bias_gelu_training``
(src/modeling.py:126-139) and the bias+act fusion point of
``LinearActivation`` (src/modeling.py:141-185). Kernel source:
csrc/ops/bias_act.hip.
"""

from __future__ import annotations

import torch

from . import _reference, extension, use_native


class _FusedBiasGelu(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        ext = extension()
        x2d = x.contiguous().view(-1, x.shape[-1])
        y = ext.bias_gelu_fwd(x2d, bias)
        ctx.save_for_backward(x2d, bias)
        ctx.shape = x.shape
        return y.view(x.shape)

    @staticmethod
    def backward(ctx, dy):
        ext = extension()
        x2d, bias = ctx.saved_tensors
        dy2d = dy.contiguous().view(-1, dy.shape[-1])
        dx, dbias = ext.bias_gelu_bwd(dy2d, x2d, bias)
        return dx.view(ctx.shape), dbias


def fused_bias_gelu(x: torch.Tensor, bias: torch.Tensor) -> torch.Tensor:
    if use_native(x):
        return _FusedBiasGelu.apply(x, bias)
    return _reference.bias_gelu(x, bias)
