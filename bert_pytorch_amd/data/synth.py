"""Synthetic masked-LM shard generator.

Writes HDF5 shards with the reference schema
(utils/encode_data.py:204-210): ``input_ids`` int32 [N, S],
``special_token_positions`` int32 [N, 3] (or [N, 2] for the RoBERTa
no-NSP layout), ``next_sentence_labels`` int8 [N]. Used as the test
fixture and as bench.py's data source (no network for real corpora).
"""

from __future__ import annotations

import os
from typing import List

import numpy as np

from . import h5lite

CLS_ID = 101
SEP_ID = 102
MASK_ID = 103
FIRST_REGULAR_ID = 1000


def make_shard(
    path: str,
    num_samples: int,
    seq_len: int,
    vocab_size: int,
    nsp: bool = True,
    seed: int = 0,
    min_len_frac: float = 0.7,
) -> None:
    rng = np.random.default_rng(seed)
    input_ids = np.zeros((num_samples, seq_len), dtype=np.int32)
    n_special = 3 if nsp else 2
    special = np.zeros((num_samples, n_special), dtype=np.int32)
    nsl = rng.integers(0, 2, size=num_samples).astype(np.int8)
    for i in range(num_samples):
        total = int(rng.integers(int(seq_len * min_len_frac), seq_len + 1))
        total = max(total, 5)
        lo = FIRST_REGULAR_ID if vocab_size > FIRST_REGULAR_ID + 1 else MASK_ID + 1
        body = rng.integers(lo, vocab_size, size=total - n_special).astype(np.int32)
        row = [CLS_ID]
        if nsp:
            split = int(rng.integers(1, len(body)))
            row += list(body[:split]) + [SEP_ID] + list(body[split:]) + [SEP_ID]
            special[i] = (0, split + 1, total - 1)
        else:
            row += list(body) + [SEP_ID]
            special[i] = (0, total - 1)
        input_ids[i, : len(row)] = row
    h5lite.write(
        path,
        {
            "input_ids": input_ids,
            "special_token_positions": special,
            "next_sentence_labels": nsl,
        },
    )


def make_dataset(
    directory: str,
    num_shards: int = 2,
    samples_per_shard: int = 64,
    seq_len: int = 128,
    vocab_size: int = 30522,
    nsp: bool = True,
    seed: int = 0,
) -> List[str]:
    os.makedirs(directory, exist_ok=True)
    paths = []
    for s in range(num_shards):
        path = os.path.join(directory, f"train_{s}.hdf5")
        make_shard(
            path, samples_per_shard, seq_len, vocab_size, nsp=nsp, seed=seed + s
        )
        paths.append(path)
    return paths
