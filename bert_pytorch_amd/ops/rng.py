"""Philox counter-state for fused-dropout kernels.

Each fused kernel that needs randomness takes a (seed, offset) pair and
runs Philox4x32-10 counters derived from them; backward kernels that
regenerate masks receive the same pair. The seed derives from the torch
seed (so ``torch.manual_seed`` reproduces runs); the offset is a
monotonically increasing per-process counter.
"""

from __future__ import annotations

import torch

_seed: int | None = None
_offset: int = 0


def reset(seed: int | None = None) -> None:
    global _seed, _offset
    _seed = seed
    _offset = 0


def next_philox(n_elements: int) -> tuple[int, int]:
    """Reserve ``n_elements`` Philox outputs; return (seed, base_offset)."""
    global _seed, _offset
    if _seed is None:
        _seed = torch.initial_seed() & 0x7FFFFFFFFFFFFFFF
    base = _offset
    # Each Philox call yields 4 32-bit words; round up generously.
    _offset += (n_elements + 3) // 4 + 1
    return _seed, base
