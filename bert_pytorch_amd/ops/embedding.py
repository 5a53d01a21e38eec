"""Fused embedding gather + add + LayerNorm + dropout (HIP, gfx950).

One kernel for ``BertEmbeddings``'s word+position(+token-type) gather,
sum, LayerNorm and dropout (reference: src/modeling.py:338-373).
Backward scatter-adds into a dense fp32 word-embedding gradient with
device-scope atomics. Kernel source: csrc/ops/embedding.hip.
"""

from __future__ import annotations

from typing import Optional

import torch

from . import _reference, extension, use_native
from .rng import next_philox


class _FusedEmbeddingLNDropout(torch.autograd.Function):
    @staticmethod
    def forward(
        ctx, input_ids, token_type_ids, word_emb, pos_emb, tok_emb, weight,
        ln_bias, p, training, eps,
    ):
        ext = extension()
        ids = input_ids.contiguous()
        tt = token_type_ids.contiguous() if token_type_ids is not None else None
        seed, offset = (
            next_philox(ids.numel() * word_emb.shape[1])
            if (training and p > 0)
            else (0, 0)
        )
        if word_emb.is_cuda and torch.is_autocast_enabled():
            out_dtype = torch.get_autocast_gpu_dtype()
        else:
            out_dtype = word_emb.dtype
        y, z, mask, mean, rstd = ext.embedding_ln_dropout_fwd(
            ids, tt, word_emb, pos_emb, tok_emb, weight, ln_bias,
            p if training else 0.0, eps, seed, offset, out_dtype,
        )
        ctx.save_for_backward(ids, tt, z, mask, weight, mean, rstd)
        ctx.p = p if training else 0.0
        ctx.vocab = word_emb.shape[0]
        ctx.max_pos = pos_emb.shape[0]
        ctx.n_types = tok_emb.shape[0] if tok_emb is not None else 0
        ctx.has_tok = tok_emb is not None
        ctx.table_dtype = word_emb.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = extension()
        ids, tt, z, mask, weight, mean, rstd = ctx.saved_tensors
        d_word, d_pos, d_tok, dw, db = ext.embedding_ln_dropout_bwd(
            dy.contiguous(), ids, tt, z, mask, weight, mean, rstd, ctx.p,
            ctx.vocab, ctx.max_pos, ctx.n_types,
        )
        if ctx.table_dtype != d_word.dtype:
            # pure-bf16 mode: grads must match the low-precision tables
            d_word = d_word.to(ctx.table_dtype)
            d_pos = d_pos.to(ctx.table_dtype)
            if ctx.has_tok:
                d_tok = d_tok.to(ctx.table_dtype)
        return (
            None,
            None,
            d_word,
            d_pos,
            d_tok if ctx.has_tok else None,
            dw,
            db,
            None,
            None,
            None,
        )


def fused_embedding_ln_dropout(
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
    if use_native(word_emb):
        return _FusedEmbeddingLNDropout.apply(
            input_ids, token_type_ids, word_emb, pos_emb, tok_emb, weight,
            ln_bias, p, training, eps,
        )
    return _reference.embedding_ln_dropout(
        input_ids, token_type_ids, word_emb, pos_emb, tok_emb, weight,
        ln_bias, p, training, eps,
    )
