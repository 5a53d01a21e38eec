"""Multi-tensor gradient clipping (HIP, gfx950).

Replaces the reference's amp_C-based ``GradientClipper``
(run_squad.py:703-725; amp_C bindings src/optimization.py:30-33):
one multi-tensor L2-norm kernel + one multi-tensor scale kernel.
"""

from __future__ import annotations

import math
from typing import Iterable

import torch

from .. import ops
from .lamb import _use_native


def clip_grad_norm_(parameters: Iterable[torch.Tensor], max_norm: float) -> float:
    """Global-norm clip over all parameter grads. Returns the pre-clip norm."""
    grads = [p.grad for p in parameters if p.grad is not None]
    if not grads:
        return 0.0
    if _use_native(grads):
        ext = ops.extension()
        gnorm_sq = ext.multi_tensor_l2norm_sq(grads)
        # scale = max_norm / max(norm, max_norm)  (<=1, no-op when under)
        ext.multi_tensor_clip_scale(grads, gnorm_sq, float(max_norm))
        return float(gnorm_sq.sqrt())
    total = math.sqrt(sum(float(g.float().pow(2).sum()) for g in grads))
    if total > max_norm > 0:
        for g in grads:
            g.mul_(max_norm / total)
    return total


class GradientClipper:
    """API-compatible with the reference's GradientClipper."""

    def __init__(self, max_grad_norm: float):
        self.max_norm = max_grad_norm

    def step(self, parameters: Iterable[torch.Tensor]) -> None:
        clip_grad_norm_(list(parameters), self.max_norm)
