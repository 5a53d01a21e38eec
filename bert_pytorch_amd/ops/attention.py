"""Fused scaled-dot-product attention with padding mask (HIP, gfx950).

Flash-style forward (online softmax over key tiles, MFMA
``mfma_f32_16x16x32_bf16`` tiles, K/V staged through LDS) replacing the
reference's five-op attention path (QK^T, +mask, softmax, dropout, PV —
src/modeling.py:403-429). Input is the packed QKV GEMM output
``[B, S, 3H]`` (no transpose kernels); the padding mask enters as per-
sequence valid lengths. Backward recomputes probabilities flash-style.
Kernel source: csrc/ops/attention.hip.
"""

from __future__ import annotations

import torch

from . import _reference, extension, use_native
from .rng import next_philox


class _FusedAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv, seqlens, num_heads, p, training):
        ext = extension()
        qkv = qkv.contiguous()
        bsz, seq, three_h = qkv.shape
        p_eff = p if training else 0.0
        seed, offset = (
            next_philox(bsz * num_heads * seq * seq) if p_eff > 0 else (0, 0)
        )
        out, lse, dmask = ext.attention_fwd(
            qkv, seqlens, num_heads, p_eff, seed, offset
        )
        ctx.save_for_backward(qkv, seqlens, out, lse, dmask)
        ctx.meta = (num_heads, p_eff, seed, offset)
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = extension()
        qkv, seqlens, out, lse, dmask = ctx.saved_tensors
        num_heads, p_eff, seed, offset = ctx.meta
        dqkv = ext.attention_bwd(
            dout.contiguous(), qkv, seqlens, out, lse, dmask, num_heads,
            p_eff, seed, offset
        )
        return dqkv, None, None, None, None


def _kernel_supported(qkv: torch.Tensor, num_heads: int) -> bool:
    """The gfx950 MFMA kernel covers the BERT/RoBERTa operating points:
    bf16, head_dim 64, S % 16 == 0. Other shapes (tiny test configs,
    fp32-without-autocast) run the eager composite on device."""
    head_dim = qkv.shape[-1] // 3 // num_heads
    return (
        qkv.dtype == torch.bfloat16
        and head_dim == 64
        and qkv.shape[1] % 16 == 0
    )


def fused_attention(
    qkv: torch.Tensor,
    seqlens: torch.Tensor,
    num_heads: int,
    p: float,
    training: bool,
) -> torch.Tensor:
    """qkv [B,S,3H] packed, seqlens [B] int32 -> context [B,S,H]."""
    if use_native(qkv) and _kernel_supported(qkv, num_heads):
        return _FusedAttention.apply(qkv, seqlens, num_heads, p, training)
    return _reference.attention(qkv, seqlens, num_heads, p, training)
