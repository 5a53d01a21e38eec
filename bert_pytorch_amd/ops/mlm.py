"""Fused MLM-decoder GEMM + bias + cross-entropy loss (HIP, gfx950).

The reference computes the MLM loss as three separate steps — the vocab
decoder matmul (src/modeling.py:570-578), the bias add, and
``nn.CrossEntropyLoss(ignore_index=-1)`` (run_pretraining.py:58-72).
Here the decoder GEMM is a hand-written MFMA kernel
(csrc/ops/mlm_head.hip) whose epilogue adds the bias and emits the
cross-entropy forward statistics (per-row online max / sum-exp
partials), so the separate full [P, V] CE-forward read pass disappears
and the framework's single biggest GEMM runs in-repo. Backward reuses
the ce_bwd softmax-recompute kernel on the stored bf16 logits; the
dgrad/wgrad GEMMs of the decoder go through the TunableOp-tuned
hipBLASLt path (measured faster than the in-repo split-K wgrad kernel
at this [V, K] output shape) and the bias gradient through the
two-stage col_sum reduction.

The logits ARE materialized (bf16): recomputing the 80-GFLOP GEMM in
backward to avoid a 78-MB round trip would cost ~80 us to save ~20 us
on this hardware (8 TB/s HBM vs ~1 PF/s GEMM), so materialize-and-reuse
is the right MI355X trade.

Routing (measured; docs/KERNELS.md "MLM decoder" and
profiles/mlm_fused_study.md): the shipped structure (256x256
macro-tile, BK=64 glds double-buffer, atomics-free fold) runs
178-275 us forward including the CE statistics vs 131-201 us for
hipBLASLt + the standalone ce_fwd pass, but its leaner backward makes
the full train step tie at P=1280 and win at P>=2048, and a same-box
e2e A/B measured dead even (phase 1 -0.3%, phase 2 +0.6% — the whole
head is ~1.3% of a step after round 1's masked-row gather). At
measured parity the in-repo MFMA kernel is the DEFAULT (the
framework's biggest GEMM runs in-repo); BPA_FUSED_MLM=0 opts back
into the library-GEMM path. Full parity coverage:
tests/test_gpu_kernels.py::test_mlm_*.
"""

from __future__ import annotations

import os

import torch
import torch.nn.functional as F

from . import _reference, extension, use_native
from .cross_entropy import fused_cross_entropy

_TILE = 256  # row tile of mlm_fwd_kernel; P is padded up to it


class _MlmDecoderLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, h, w, bias, labels, ignore_index):
        ext = extension()
        P = h.shape[0]
        pad = (-P) % _TILE
        if pad:
            h = torch.cat([h, h.new_zeros(pad, h.shape[1])])
            labels = torch.cat(
                [labels, labels.new_full((pad,), ignore_index)]
            )
        h = h.contiguous()
        logits, loss_sum, count, lse = ext.mlm_head_fwd(
            h, w.contiguous(), bias.contiguous(), labels, ignore_index
        )
        ctx.save_for_backward(h, w, logits, labels, lse, count)
        ctx.ignore_index = ignore_index
        ctx.rows = P
        return loss_sum / count.clamp(min=1)

    @staticmethod
    def backward(ctx, dloss):
        ext = extension()
        h, w, logits, labels, lse, count = ctx.saved_tensors
        dlogits = ext.ce_bwd(
            dloss.contiguous(), logits, labels, lse, count, ctx.ignore_index
        )
        # padding rows have label == ignore_index -> zero dlogits rows,
        # so they contribute nothing to dw/dbias and slice away from dh
        dh = (dlogits @ w)[: ctx.rows]
        dw = dlogits.transpose(0, 1) @ h
        dbias = ext.col_sum(dlogits)
        return dh, dw, dbias, None, None


def mlm_decoder_loss(
    hidden: torch.Tensor,
    weight: torch.Tensor,
    bias: torch.Tensor,
    labels: torch.Tensor,
    ignore_index: int = -1,
) -> torch.Tensor:
    """Mean CE loss over non-ignored rows of ``hidden @ weight.T + bias``.

    hidden [P, K] (transform output over gathered masked rows), weight
    [V, K] (tied word embeddings), bias [V], labels [P] int64.
    """
    if use_native(hidden):
        ext = extension()
        P, K = hidden.shape
        V = weight.shape[0]
        p_pad = ((P + _TILE - 1) // _TILE) * _TILE
        bf16 = hidden.dtype == torch.bfloat16 or (
            torch.is_autocast_enabled()
            and torch.get_autocast_gpu_dtype() == torch.bfloat16
        )
        if (
            bf16
            and ext.mlm_head_supported(p_pad, V, K)
            and os.environ.get("BPA_FUSED_MLM") != "0"  # opt-out (see above)
        ):
            h = hidden.to(torch.bfloat16)
            w = weight.to(torch.bfloat16)
            b = bias.float()
            return _MlmDecoderLoss.apply(h, w, b, labels, ignore_index)
        # unfused native: library GEMM + fused CE kernel. Outside
        # autocast F.linear requires matching dtypes, so align the fp32
        # bias with bf16 inputs explicitly (autograd casts the grad
        # back to fp32).
        w, b = weight, bias
        if not torch.is_autocast_enabled():
            if w.dtype != hidden.dtype:
                w = w.to(hidden.dtype)
            if b.dtype != hidden.dtype:
                b = b.to(hidden.dtype)
        scores = F.linear(hidden, w, b)
        return fused_cross_entropy(scores, labels, ignore_index)
    # CPU / eager oracle
    scores = F.linear(hidden.float(), weight.float(), bias.float())
    return _reference.cross_entropy(scores, labels, ignore_index)
