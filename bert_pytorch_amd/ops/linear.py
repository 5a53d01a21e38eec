"""Linear projection with HIP column-sum bias gradient (gfx950).

``fused_linear`` behaves like ``F.linear`` (the forward IS one
hipBLASLt GEMM with fused bias epilogue) but its backward computes the
bias gradient with the streaming two-stage ``col_sum`` HIP kernel
instead of ATen's generic reduce — the packed-QKV projection's bias
grad over [B*S, 3H] bf16 was one of the last eager reduce hotspots.

Autocast handling is explicit: the wrapper casts x/w/b to the autocast
dtype (tracked, so grads flow back through the casts to the fp32
masters exactly as torch autocast would) and runs the Function with
autocast disabled so backward dtypes match the saved tensors.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from . import extension, use_native


class _FusedLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        x2 = x.reshape(-1, x.shape[-1])
        ctx.save_for_backward(x2, weight)
        ctx.x_shape = x.shape
        y = F.linear(x2, weight, bias)
        return y.view(*x.shape[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, dy):
        from .matmul import _wgrad  # noqa: PLC0415

        x2, weight = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx = dy2 @ weight
        dw = _wgrad(dy2, x2.contiguous())
        db = extension().col_sum(dy2)
        if db.dtype != dy2.dtype:
            db = db.to(dy2.dtype)
        return dx.view(ctx.x_shape), dw, db


def fused_linear(x: torch.Tensor, weight: torch.Tensor,
                 bias: torch.Tensor) -> torch.Tensor:
    if not use_native(x):
        return F.linear(x, weight, bias)
    if torch.is_autocast_enabled() and x.is_cuda:
        dt = torch.get_autocast_dtype("cuda")
        x, weight, bias = x.to(dt), weight.to(dt), bias.to(dt)
    with torch.autocast("cuda", enabled=False):
        return _FusedLinear.apply(x, weight, bias)
