"""Fused FFN block: GELU folded into the hipBLASLt GEMM epilogues.

Replaces the reference's LinearActivation + BertOutput.dense GEMM pair
(src/modeling.py:141-185, 458-479 — the bias+GELU fusion point) with

    forward:  act = GELU(x @ w1^T + b1)   [GELU_AUX_BIAS epilogue,
              pre-activation saved to AUX]
              out = act @ w2^T            [plain tuned GEMM]
    backward: dpre = (dout @ w2) * gelu'(aux), db1 = colsum(dpre)
              [DGELU_BGRAD epilogue on the FFN2 dgrad GEMM]
              dw2 = dout^T @ act ; dw1 = dpre^T @ x ; dx = dpre @ w1

so the standalone bias-GELU kernels and their two extra [M,4096]
round-trips disappear. Epilogue algo selection is benchmarked once per
shape per process (csrc/ops/gemm_epilogue.cpp). FFN2's bias stays fused
in the downstream bias+dropout+residual+LN kernel.

Note: hipBLASLt's GELU is the tanh approximation; the eager/CPU
reference and the standalone kernels use exact-erf GELU (max relative
difference ~1e-3, inside bf16 rounding).
"""

from __future__ import annotations

import torch

from . import extension, use_native


class _FusedFFN(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2d, w1, b1, w2):
        ext = extension()
        act, aux = ext.gemm_bias_gelu_fwd(x2d, w1, b1)
        out = act @ w2.t()
        ctx.save_for_backward(x2d, w1, w2, aux, act)
        ctx.b1_dtype = b1.dtype
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = extension()
        x2d, w1, w2, aux, act = ctx.saved_tensors
        dout = dout.contiguous()
        dpre, db1 = ext.gemm_dgelu_bgrad(dout, w2, aux)
        dw2 = dout.t() @ act
        dw1 = dpre.t() @ x2d
        dx = dpre @ w1
        if db1.dtype != ctx.b1_dtype:
            db1 = db1.to(ctx.b1_dtype)
        return dx, dw1, db1, dw2


_aux_supported = None


def ffn_supported(x: torch.Tensor) -> bool:
    """True only when hipBLASLt offers the GELU_AUX_BIAS/DGELU_BGRAD
    epilogues (ROCm 7.2's gfx950 library does NOT - probed once; the
    standalone bias-GELU kernels then run instead)."""
    global _aux_supported
    if not use_native(x):
        return False
    # The epilogue GEMMs require bf16 inputs; outside autocast an fp32
    # forward must take the standalone bias-GELU path instead (mirrors
    # fused_attention's dtype gate).
    if x.dtype != torch.bfloat16 and not (
        x.is_cuda and torch.is_autocast_enabled()
    ):
        return False
    if _aux_supported is None:
        _aux_supported = bool(extension().gemm_gelu_aux_supported())
    return _aux_supported


def fused_ffn(
    x: torch.Tensor,
    w1: torch.Tensor,
    b1: torch.Tensor,
    w2: torch.Tensor,
) -> torch.Tensor:
    """x [*, H] -> GELU(x@w1^T+b1) @ w2^T, bf16 GEMM epilogues.

    Caller guarantees the native path (CUDA + extension); autocast
    casting is handled here like ops.fused_linear.
    """
    shape = x.shape
    x2d = x.reshape(-1, shape[-1])
    if torch.is_autocast_enabled() and x.is_cuda:
        dt = torch.get_autocast_dtype("cuda")
        x2d, w1, b1, w2 = x2d.to(dt), w1.to(dt), b1.to(dt), w2.to(dt)
    with torch.autocast("cuda", enabled=False):
        out = _FusedFFN.apply(x2d.contiguous(), w1, b1, w2)
    return out.view(*shape[:-1], w2.shape[0])
