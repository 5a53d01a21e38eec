"""Bias-free linear with the hand-written split-K MFMA wgrad kernel.

``linear_nobias`` behaves like ``F.linear(x, w, None)`` (forward and
dgrad stay on hipBLASLt, which is already at ~1 PF/s for those shapes)
but computes the WEIGHT gradient with the in-repo split-K MFMA kernel
(csrc/ops/wgrad.hip): the library's wgrad GEMMs run at only ~580-790
TF/s on the encoder's skinny [out,in] x K=tokens shapes (64-256
workgroups on 256 CUs - measured in benchmarks/gemm_shapes.py, and an
exhaustive TunableOp re-search does not improve them), and they are
~24% of the phase-1 training step. Accumulation is fp32 end-to-end.

Reference ops being replaced: the implicit wgrad GEMMs behind
src/modeling.py's nn.Linear calls (attention out-proj :432-443, FFN
:458-479). The QKV projection's wgrad routes through the same kernel
via ops/linear.py.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from . import extension, use_native


def _wgrad(dy2: torch.Tensor, x2: torch.Tensor) -> torch.Tensor:
    """dW = dy2^T @ x2 via the split-K kernel when supported."""
    if (
        dy2.is_cuda
        and dy2.dtype == torch.bfloat16
        and x2.dtype == torch.bfloat16
        and extension().wgrad_tn_profitable(
            dy2.shape[0], dy2.shape[1], x2.shape[1]
        )
    ):
        return extension().wgrad_tn(dy2, x2)
    return dy2.t() @ x2


class _LinearNoBias(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight):
        x2 = x.reshape(-1, x.shape[-1])
        ctx.save_for_backward(x2, weight)
        ctx.x_shape = x.shape
        y = F.linear(x2, weight, None)
        return y.view(*x.shape[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, dy):
        x2, weight = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx = dy2 @ weight
        dw = _wgrad(dy2, x2.contiguous())
        return dx.view(ctx.x_shape), dw


def linear_nobias(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """F.linear(x, w, None) with the HIP split-K wgrad backward."""
    if not use_native(x):
        return F.linear(x, weight, None)
    if torch.is_autocast_enabled() and x.is_cuda:
        dt = torch.get_autocast_dtype("cuda")
        x, weight = x.to(dt), weight.to(dt)
    with torch.autocast("cuda", enabled=False):
        return _LinearNoBias.apply(x, weight)
