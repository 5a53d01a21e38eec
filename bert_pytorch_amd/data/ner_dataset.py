"""NER (CoNLL-format) dataset: per-word label propagation over
wordpieces (reference: src/ner_dataset.py:13-85)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Tuple

import torch


@dataclass
class Sample:
    words: List[str]
    labels: List[str]

    def encoded(
        self, tokenizer, label_to_id: Dict[str, int], max_seq_len: int = 128,
        pad_label_id: int = -100,
    ) -> Tuple[List[str], List[str], List[int], List[int], List[int]]:
        tokens: List[str] = []
        label_ids: List[int] = []
        for word, label in zip(self.words, self.labels):
            subtokens = tokenizer.encode(word, add_special_tokens=False).tokens
            if not subtokens:
                continue
            tokens.extend(subtokens)
            # label on the first wordpiece; padding label on continuations
            label_ids.append(label_to_id[label])
            label_ids.extend([pad_label_id] * (len(subtokens) - 1))
        tokens = tokens[: max_seq_len - 2]
        label_ids = label_ids[: max_seq_len - 2]
        tokens = ["[CLS]"] + tokens + ["[SEP]"]
        label_ids = [pad_label_id] + label_ids + [pad_label_id]
        ids = tokenizer.convert_tokens_to_ids(tokens)
        mask = [1] * len(ids)
        while len(ids) < max_seq_len:
            ids.append(0)
            mask.append(0)
            label_ids.append(pad_label_id)
        return self.words, self.labels, ids, label_ids, mask


def read_conll(path: str, word_col: int = 0, label_col: int = -1) -> List[Sample]:
    samples: List[Sample] = []
    words: List[str] = []
    labels: List[str] = []
    with open(path, "r", encoding="utf-8") as f:
        for line in f:
            line = line.strip()
            if not line or line.startswith("-DOCSTART-"):
                if words:
                    samples.append(Sample(words, labels))
                    words, labels = [], []
                continue
            parts = line.split()
            words.append(parts[word_col])
            labels.append(parts[label_col])
    if words:
        samples.append(Sample(words, labels))
    return samples


class NERDataset(torch.utils.data.Dataset):
    def __init__(self, path: str, tokenizer, max_seq_len: int = 128,
                 labels: List[str] | None = None, pad_label_id: int = -100):
        self.samples = read_conll(path)
        self.tokenizer = tokenizer
        self.max_seq_len = max_seq_len
        self.pad_label_id = pad_label_id
        if labels is None:
            labels = sorted({l for s in self.samples for l in s.labels})
        self.labels = labels
        self.label_to_id = {l: i for i, l in enumerate(labels)}

    @property
    def num_labels(self) -> int:
        return len(self.labels)

    def __len__(self) -> int:
        return len(self.samples)

    def __getitem__(self, idx: int):
        _, _, ids, label_ids, mask = self.samples[idx].encoded(
            self.tokenizer, self.label_to_id, self.max_seq_len,
            self.pad_label_id,
        )
        return (
            torch.tensor(ids, dtype=torch.long),
            torch.tensor(mask, dtype=torch.long),
            torch.tensor(label_ids, dtype=torch.long),
        )
