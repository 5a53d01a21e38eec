"""Fused log-softmax + NLL cross-entropy with ignore_index (HIP, gfx950).

Replaces ``nn.CrossEntropyLoss(ignore_index=-1)`` over the MLM vocab
(reference: run_pretraining.py:58-72) with one kernel that never
materializes the [N, V] log-softmax: forward stores only the per-row
logsumexp; backward recomputes softmax on the fly.
Kernel source: csrc/ops/cross_entropy.hip.
"""

from __future__ import annotations

import torch

from . import _reference, extension, use_native


class _FusedCrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, ignore_index):
        ext = extension()
        logits = logits.contiguous()
        loss_sum, count, lse = ext.ce_fwd(logits, labels, ignore_index)
        ctx.save_for_backward(logits, labels, lse, count)
        ctx.ignore_index = ignore_index
        # mean over non-ignored rows; 0.0 if none (count clamped >= 1)
        return loss_sum / count.clamp(min=1)

    @staticmethod
    def backward(ctx, dloss):
        ext = extension()
        logits, labels, lse, count = ctx.saved_tensors
        dlogits = ext.ce_bwd(
            dloss.contiguous(), logits, labels, lse, count, ctx.ignore_index
        )
        return dlogits, None, None


def fused_cross_entropy(
    logits: torch.Tensor, labels: torch.Tensor, ignore_index: int = -1
) -> torch.Tensor:
    if use_native(logits):
        return _FusedCrossEntropy.apply(logits, labels, ignore_index)
    return _reference.cross_entropy(logits, labels, ignore_index)
