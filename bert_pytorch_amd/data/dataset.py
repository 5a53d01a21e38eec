"""Sharded pretraining dataset + chunked checkpointable DistributedSampler.

Re-designed from the reference's ``ShardedPretrainingDataset``
(src/dataset.py:9-338) and chunked ``DistributedSampler``
(src/dataset.py:341-428) with the same external behavior:

* multi-file HDF5 shards, at most 2 resident in RAM, the next shard
  prefetched by a background thread while the current one is consumed
  sequentially;
* RoBERTa-style dynamic masking (80/10/10) computed on CPU workers from
  the new {input_ids, special_token_positions, next_sentence_labels}
  schema, plus the legacy NVIDIA pre-masked schema;
* a rank-chunked sequential sampler whose position checkpoints into the
  training state dict.

Deliberate fixes over the reference: in-file index uses
``idx - file_sample_start_idx`` (not the accidental negative-index
trick), masked rows are copied before mutation (the reference mutates
the cached shard in place), and mask positions are sampled WITHOUT
replacement so exactly ``mask_count`` distinct tokens are masked.
"""

from __future__ import annotations

import math
import os
import threading
import warnings
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch
import torch.utils.data

from .h5lite import H5LiteFile


class ShardedPretrainingDataset(torch.utils.data.Dataset):
    def __init__(
        self,
        files: Sequence[str] | str,
        mask_token_index: int,
        max_pred_per_seq: int,
        masked_lm_prob: float,
        vocab_size: int,
        original_token_prob: float = 0.1,
        random_token_prob: float = 0.1,
        shuffle: bool = False,
        seed: Optional[int] = None,
    ):
        if original_token_prob + random_token_prob > 1:
            raise ValueError("original_token_prob + random_token_prob > 1")
        if shuffle:
            raise ValueError(
                "shuffle is not supported; pre-shuffle samples in the shards"
            )
        if isinstance(files, str):
            files = [files]
        files = sorted(files)  # all ranks must agree on order
        self.files, self.file_idxs = self._verify_and_count_samples(files)

        self.mask_token_index = mask_token_index
        self.max_pred_per_seq = max_pred_per_seq
        self.masked_lm_prob = masked_lm_prob
        self.vocab_size = vocab_size
        self.original_token_prob = original_token_prob
        self.random_token_prob = random_token_prob
        self.seed = seed
        self.epoch = 0
        self._rng = np.random.default_rng(seed)

        self.file_idx: Optional[int] = None
        self.next_file_idx: Optional[int] = None
        self.file_sample_start_idx = -1
        self.file_sample_end_idx = -1
        self.data: Optional[Dict[str, np.ndarray]] = None
        self._next_data: Optional[Dict[str, np.ndarray]] = None
        self._next_thread: Optional[threading.Thread] = None

    # -- public API ------------------------------------------------------
    def set_epoch(self, epoch: int) -> None:
        self.epoch = epoch

    def __len__(self) -> int:
        return self.file_idxs[-1][1]

    def __getitem__(self, idx: int) -> List[np.ndarray]:
        if self.data is None:
            self.next_file_idx = self._file_for_sample(idx)
            self._next_thread = self._async_load(self.next_file_idx)

        if not (self.file_sample_start_idx <= idx < self.file_sample_end_idx):
            # current shard exhausted: join the prefetch, swap, prefetch next
            self.data = None
            self._next_thread.join()
            self.data = self._next_data
            self._next_data = None
            self.file_idx = self.next_file_idx
            self.next_file_idx = (self.next_file_idx + 1) % len(self.files)
            self._next_thread = self._async_load(self.next_file_idx)
            self.file_sample_start_idx = self.file_idxs[self.file_idx][0]
            self.file_sample_end_idx = self.file_idxs[self.file_idx][1]

        if not (self.file_sample_start_idx <= idx < self.file_sample_end_idx):
            raise RuntimeError(
                f"idx {idx} outside current shard "
                f"[{self.file_sample_start_idx}, {self.file_sample_end_idx}): "
                "this dataset requires sequential (chunked-sampler) access"
            )

        local = idx - self.file_sample_start_idx
        input_ids = self.data["input_ids"][local]
        next_sentence_label = self.data["next_sentence_labels"][local]

        if "special_token_positions" in self.data:
            special = self.data["special_token_positions"][local]
            segment_ids = self._segment_ids(input_ids, special)
            input_mask = self._input_mask(input_ids, special)
            masked_input_ids, masked_lm_labels = self._mask_input(
                input_ids, special
            )
        else:  # legacy NVIDIA pre-masked format
            segment_ids = self.data["segment_ids"][local]
            input_mask = self.data["input_mask"][local]
            masked_input_ids = input_ids
            masked_lm_labels = self._premasked_labels(
                input_ids,
                self.data["masked_lm_positions"][local],
                self.data["masked_lm_ids"][local],
            )

        return [
            masked_input_ids.astype(np.int64),
            segment_ids.astype(np.int64),
            input_mask.astype(np.int64),
            masked_lm_labels.astype(np.int64),
            np.asarray(next_sentence_label).astype(np.int64),
        ]

    # -- internals -------------------------------------------------------
    def _file_for_sample(self, idx: int) -> int:
        for i, (start, end) in enumerate(self.file_idxs):
            if start <= idx < end:
                return i
        raise ValueError(f"idx {idx} exceeds dataset size {len(self)}")

    def _async_load(self, file_idx: int) -> threading.Thread:
        def load(path: str) -> None:
            with H5LiteFile(path) as f:
                self._next_data = {k: np.asarray(f[k]) for k in f.keys()}

        th = threading.Thread(target=load, args=(self.files[file_idx],), daemon=True)
        th.start()
        return th

    @staticmethod
    def _segment_ids(input_ids: np.ndarray, special: np.ndarray) -> np.ndarray:
        """[CLS] a.. [SEP] (b.. [SEP]): second segment (if any) gets 1."""
        segment_ids = np.zeros_like(input_ids)
        if len(special) == 3:
            segment_ids[special[1] + 1 : special[2] + 1] = 1
        return segment_ids

    @staticmethod
    def _input_mask(input_ids: np.ndarray, special: np.ndarray) -> np.ndarray:
        input_mask = np.zeros_like(input_ids)
        input_mask[: special[-1] + 1] = 1
        return input_mask

    def _mask_input(
        self, input_ids: np.ndarray, special: np.ndarray
    ) -> Tuple[np.ndarray, np.ndarray]:
        """Dynamic 80/10/10 masking over non-special, non-pad positions."""
        input_ids = input_ids.copy()  # never mutate the cached shard
        masked_lm_labels = np.full_like(input_ids, -1)
        special_set = set(int(s) for s in special)
        candidates = np.array(
            [i for i in range(int(special[-1])) if i not in special_set],
            dtype=np.int64,
        )
        if candidates.size == 0:
            return input_ids, masked_lm_labels
        mask_count = min(
            self.max_pred_per_seq,
            max(1, int(candidates.size * self.masked_lm_prob)),
        )
        mask_indices = self._rng.choice(candidates, mask_count, replace=False)
        masked_lm_labels[mask_indices] = input_ids[mask_indices]
        draws = self._rng.random(mask_count)
        for idx, draw in zip(mask_indices, draws):
            if draw < self.original_token_prob:
                continue  # keep original token
            if draw < self.original_token_prob + self.random_token_prob:
                input_ids[idx] = self._rng.integers(0, self.vocab_size - 1)
            else:
                input_ids[idx] = self.mask_token_index
        return input_ids, masked_lm_labels

    @staticmethod
    def _premasked_labels(
        input_ids: np.ndarray,
        masked_lm_positions: np.ndarray,
        masked_lm_ids: np.ndarray,
    ) -> np.ndarray:
        labels = np.full_like(input_ids, -1)
        count = len(masked_lm_positions)
        padded = np.nonzero(masked_lm_positions == 0)[0]
        if padded.size:
            count = padded[0]
        labels[masked_lm_positions[:count]] = masked_lm_ids[:count]
        return labels

    @staticmethod
    def _verify_and_count_samples(
        files: Sequence[str],
    ) -> Tuple[List[str], List[Tuple[int, int]]]:
        current = 0
        ok_files, idxs = [], []
        keys = ["input_ids", "next_sentence_labels"]
        for path in files:
            if not os.path.isfile(path):
                warnings.warn(f"file not found, skipping: {path}")
                continue
            try:
                with H5LiteFile(path) as f:
                    counts = [len(f[k]) for k in keys]
            except Exception as e:  # noqa: BLE001
                warnings.warn(f"unreadable shard {path} ({e}); skipping")
                continue
            if len(set(counts)) != 1:
                warnings.warn(f"per-key sample counts differ in {path}; skipping")
                continue
            ok_files.append(path)
            idxs.append((current, current + counts[0]))
            current += counts[0]
        if not ok_files:
            raise RuntimeError("no valid data shards found")
        return ok_files, idxs


class DistributedSampler(torch.utils.data.distributed.DistributedSampler):
    """Rank-chunked sequential sampler with checkpointable position.

    Each rank draws a CONTIGUOUS chunk of the (padded) index space so a
    rank walks its shard files in order — the double-buffered dataset
    then needs one live shard + one prefetch per rank. The iteration
    position (``index``) round-trips through state_dict for mid-epoch
    resume. (Reference: src/dataset.py:341-428.)
    """

    def __init__(self, dataset, num_replicas=None, rank=None, **kwargs):
        kwargs["shuffle"] = False
        super().__init__(dataset, num_replicas, rank, **kwargs)
        if hasattr(self.dataset, "seed"):
            self.dataset.seed = self.seed

        indices = list(range(len(self.dataset)))
        if not self.drop_last:
            padding = self.total_size - len(indices)
            if padding <= len(indices):
                indices += indices[:padding]
            else:
                indices += (indices * math.ceil(padding / len(indices)))[:padding]
        else:
            indices = indices[: self.total_size]
        assert len(indices) == self.total_size
        self.global_indices = indices
        self.index = 0

    def __len__(self) -> int:
        return self.num_samples

    def __iter__(self):
        return self

    def __next__(self) -> int:
        if self.index == self.num_samples:
            self.index = 0
            raise StopIteration
        value = self.global_indices[self.index + self.rank * self.num_samples]
        self.index += 1
        return value

    def state_dict(self) -> dict:
        return {
            "epoch": self.epoch,
            "seed": self.seed,
            "num_replicas": self.num_replicas,
            "total_size": self.total_size,
            "index": self.index,
        }

    def load_state_dict(self, state: dict) -> None:
        if state["total_size"] != self.total_size:
            warnings.warn(
                "dataset size changed since checkpoint; sampler state reset"
            )
            return
        if state["num_replicas"] != self.num_replicas:
            warnings.warn(
                "world size changed since checkpoint; sampler state reset"
            )
            return
        self.epoch = state["epoch"]
        self.seed = state["seed"]
        self.index = state["index"]

    def set_epoch(self, epoch: int) -> None:
        if hasattr(self.dataset, "set_epoch"):
            self.dataset.set_epoch(epoch)
        self.epoch = epoch
