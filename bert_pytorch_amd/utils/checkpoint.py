"""Checkpoint save/load with the reference's layout and dict schema.

Layout (reference: run_pretraining.py:243-265, 505-528):
``{output_dir}/pretrain_ckpts/ckpt_{global_step}.pt`` holding
``{'model', 'optimizer', 'sampler', 'epoch', ['preconditioner'],
['scaler']}``; rank 0 writes; a rolling window of the most recent
``keep`` files is retained; resume picks the max step from filenames.
"""

from __future__ import annotations

import glob
import os
import re
from typing import Any, Dict, Optional, Tuple

import torch

CKPT_DIR = "pretrain_ckpts"
_CKPT_RE = re.compile(r"ckpt_(\d+)\.pt$")


def checkpoint_dir(output_dir: str) -> str:
    return os.path.join(output_dir, CKPT_DIR)


def find_latest(output_dir: str) -> Optional[Tuple[str, int]]:
    """Return (path, global_step) of the newest checkpoint, or None."""
    best = None
    for path in glob.glob(os.path.join(checkpoint_dir(output_dir), "ckpt_*.pt")):
        m = _CKPT_RE.search(path)
        if m:
            step = int(m.group(1))
            if best is None or step > best[1]:
                best = (path, step)
    return best


def save(
    output_dir: str,
    global_step: int,
    state: Dict[str, Any],
    keep: int = 3,
) -> str:
    os.makedirs(checkpoint_dir(output_dir), exist_ok=True)
    path = os.path.join(checkpoint_dir(output_dir), f"ckpt_{global_step}.pt")
    torch.save(state, path)
    # rolling window
    ckpts = []
    for p in glob.glob(os.path.join(checkpoint_dir(output_dir), "ckpt_*.pt")):
        m = _CKPT_RE.search(p)
        if m:
            ckpts.append((int(m.group(1)), p))
    for _, stale in sorted(ckpts)[:-keep] if keep > 0 else []:
        os.remove(stale)
    return path


def load(path: str) -> Dict[str, Any]:
    return torch.load(path, map_location="cpu", weights_only=False)
