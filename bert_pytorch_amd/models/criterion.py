"""Pretraining loss (reference: run_pretraining.py:58-72).

MLM cross-entropy (ignore_index=-1, mean over masked tokens) + NSP
cross-entropy. Accepts either the reference's full [B, S, V] prediction
scores or the gathered [P, V] masked-rows fast path that
``BertForPreTraining.forward`` emits when given labels.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from .. import ops


class BertPretrainingCriterion(torch.nn.Module):
    def __init__(self, vocab_size: int):
        super().__init__()
        self.vocab_size = vocab_size

    def forward(
        self,
        prediction_scores: torch.Tensor,
        seq_relationship_score: Optional[torch.Tensor],
        masked_lm_labels: torch.Tensor,
        next_sentence_labels: Optional[torch.Tensor],
    ) -> torch.Tensor:
        if prediction_scores.dim() == 0:
            # model already fused the MLM head into the loss
            # (BertForPreTraining.forward(compute_mlm_loss=True))
            loss = prediction_scores
        elif prediction_scores.dim() == 3:  # full-scores API (reference)
            loss = ops.fused_cross_entropy(
                prediction_scores.view(-1, self.vocab_size),
                masked_lm_labels.view(-1),
                ignore_index=-1,
            )
        else:  # gathered masked rows
            loss = ops.fused_cross_entropy(
                prediction_scores, masked_lm_labels, ignore_index=-1
            )
        if seq_relationship_score is not None and next_sentence_labels is not None:
            loss = loss + F.cross_entropy(
                seq_relationship_score.view(-1, 2).float(),
                next_sentence_labels.view(-1),
                ignore_index=-1,
            )
        return loss
