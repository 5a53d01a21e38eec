"""Eager fp32 reference composites for every fused HIP op.

These are the single source of truth for the *math* of each fused kernel:
the CPU path runs them directly (differentiable via autograd), and the GPU
numerics tests compare each HIP kernel against them in fp32
(reference semantics: src/modeling.py:118-139 gelu/bias_gelu,
:282-335 LayerNorm, :376-429 attention, run_pretraining.py:58-72 loss).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn.functional as F


def gelu(x: torch.Tensor) -> torch.Tensor:
    """Exact (erf) GELU — the reference's gelu (src/modeling.py:118-120)."""
    return x * 0.5 * (1.0 + torch.erf(x / math.sqrt(2.0)))


def bias_gelu(x: torch.Tensor, bias: torch.Tensor) -> torch.Tensor:
    return gelu(x + bias)


def layer_norm(
    x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor, eps: float = 1e-12
) -> torch.Tensor:
    return F.layer_norm(x, (x.shape[-1],), weight, bias, eps)


def bias_dropout_residual_ln(
    x: torch.Tensor,
    bias: Optional[torch.Tensor],
    residual: torch.Tensor,
    weight: torch.Tensor,
    ln_bias: torch.Tensor,
    p: float,
    training: bool,
    eps: float = 1e-12,
) -> torch.Tensor:
    """y = LN(dropout(x + bias) + residual)."""
    if bias is not None:
        x = x + bias
    x = F.dropout(x, p=p, training=training)
    return layer_norm(x + residual, weight, ln_bias, eps)


def embedding_ln_dropout(
    input_ids: torch.Tensor,
    token_type_ids: Optional[torch.Tensor],
    word_emb: torch.Tensor,
    pos_emb: torch.Tensor,
    tok_emb: Optional[torch.Tensor],
    weight: torch.Tensor,
    ln_bias: torch.Tensor,
    p: float,
    training: bool,
    eps: float = 1e-12,
) -> torch.Tensor:
    """y = dropout(LN(word[ids] + pos[0..S-1] + tok[type_ids]))."""
    seq_len = input_ids.shape[1]
    e = F.embedding(input_ids, word_emb) + pos_emb[:seq_len].unsqueeze(0)
    if tok_emb is not None and token_type_ids is not None:
        e = e + F.embedding(token_type_ids, tok_emb)
    e = layer_norm(e, weight, ln_bias, eps)
    return F.dropout(e, p=p, training=training)


def attention(
    qkv: torch.Tensor,
    seqlens: torch.Tensor,
    num_heads: int,
    p: float,
    training: bool,
) -> torch.Tensor:
    """Scaled-dot-product attention with a padding mask.

    qkv: [B, S, 3*H] packed (query | key | value, each H wide).
    seqlens: [B] int — number of valid (non-pad) tokens per sequence.
    Returns [B, S, H]. Padding keys are masked out; padded query rows
    produce (dropout of) uniform-attention values, matching the additive
    -10000 mask semantics of the reference (src/modeling.py:413-425,
    862-870) up to the softmax over all-masked rows.
    """
    bsz, seq, three_h = qkv.shape
    hidden = three_h // 3
    head_dim = hidden // num_heads
    q, k, v = qkv.split(hidden, dim=-1)

    def shape(t):  # [B, S, H] -> [B, nh, S, dh]
        return t.view(bsz, seq, num_heads, head_dim).transpose(1, 2)

    q, k, v = shape(q), shape(k), shape(v)
    scores = torch.matmul(q, k.transpose(-1, -2)) / math.sqrt(head_dim)
    key_pad = torch.arange(seq, device=qkv.device).unsqueeze(0) >= seqlens.unsqueeze(1)
    scores = scores.masked_fill(key_pad[:, None, None, :], -10000.0)
    probs = F.softmax(scores, dim=-1)
    probs = F.dropout(probs, p=p, training=training)
    ctx = torch.matmul(probs, v)  # [B, nh, S, dh]
    return ctx.transpose(1, 2).reshape(bsz, seq, hidden)


def cross_entropy(
    logits: torch.Tensor, labels: torch.Tensor, ignore_index: int = -1
) -> torch.Tensor:
    return F.cross_entropy(
        logits.float(), labels, ignore_index=ignore_index, reduction="mean"
    )
