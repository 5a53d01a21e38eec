#!/usr/bin/env python3
"""Encode one-sentence-per-line text shards into pretraining HDF5 shards.

MI355X-native equivalent of the reference's utils/encode_data.py
(TrainingSample packing :12-35, create_samples_from_document :65-167,
write_samples_to_hdf5 :183-210, mp.Pool fan-out :306-307). Writes the
same schema through the in-repo h5lite writer (no libhdf5 needed):

    input_ids                int32 [N, S]  (already padded with 0)
    special_token_positions  int32 [N, 3]  ([CLS], first [SEP], last [SEP])
                                   [N, 2]  when NSP is disabled ([CLS], [SEP])
    next_sentence_labels     int8  [N]     (1 = random next, 0 = true next)

Input text format (utils/format.py output): one sentence per line, blank
line between articles/documents.

Usage:
    python utils/encode_data.py --input_dir shards/ --output_dir hdf5/ \
        --tokenizer wordpiece --vocab_file vocab.txt \
        --max_seq_len 128 --nsp_probability 0.5 --short_seq_prob 0.1
"""

from __future__ import annotations

import argparse
import multiprocessing as mp
import os
import random
import sys
from dataclasses import dataclass, field
from pathlib import Path
from typing import List, Optional

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from bert_pytorch_amd.data import h5lite  # noqa: E402


@dataclass
class TrainingSample:
    """One packed pretraining sequence (reference: encode_data.py:12-35)."""

    tokens_a: List[int]
    tokens_b: List[int] = field(default_factory=list)
    is_random_next: bool = False

    def encode(self, cls_id: int, sep_id: int, max_seq_len: int, nsp: bool):
        ids = [cls_id] + self.tokens_a + [sep_id]
        special = [0, len(ids) - 1]
        if nsp:
            ids += self.tokens_b + [sep_id]
            special = [0, special[1], len(ids) - 1]
        ids = ids[:max_seq_len]
        special = [min(s, max_seq_len - 1) for s in special]
        ids += [0] * (max_seq_len - len(ids))
        return ids, special, int(self.is_random_next)


def read_documents(path: str, tokenizer) -> List[List[List[int]]]:
    """Parse a formatted shard into documents of tokenized sentences."""
    documents: List[List[List[int]]] = [[]]
    with open(path, encoding="utf-8") as f:
        for line in f:
            line = line.strip()
            if not line:
                if documents[-1]:
                    documents.append([])
                continue
            ids = tokenizer.encode(line, add_special_tokens=False).ids
            if ids:
                documents[-1].append(ids)
    if documents and not documents[-1]:
        documents.pop()
    return documents


def create_samples_from_document(
    documents: List[List[List[int]]],
    doc_idx: int,
    max_seq_len: int,
    nsp_probability: float,
    short_seq_prob: float,
    rng: random.Random,
) -> List[TrainingSample]:
    """Greedy chunk packing with NSP pair construction (reference:
    encode_data.py:65-167). With NSP: pack sentences to a target length,
    split at a random sentence boundary into A/B, and with probability
    ``nsp_probability`` replace B with a span from a random other
    document. Without NSP (RoBERTa): pack contiguous full-length
    sequences."""
    document = documents[doc_idx]
    nsp = nsp_probability > 0
    # [CLS] + [SEP] (+ [SEP] for pairs)
    max_tokens = max_seq_len - (3 if nsp else 2)
    target_len = max_tokens
    if rng.random() < short_seq_prob:
        target_len = rng.randint(2, max_tokens)

    samples: List[TrainingSample] = []
    chunk: List[List[int]] = []
    chunk_len = 0
    i = 0
    while i < len(document):
        sentence = document[i]
        chunk.append(sentence)
        chunk_len += len(sentence)
        last = i == len(document) - 1
        if last or chunk_len >= target_len:
            if not nsp:
                flat = [t for s in chunk for t in s]
                for start in range(0, len(flat), max_tokens):
                    part = flat[start : start + max_tokens]
                    if len(part) >= 2 or (start == 0 and part):
                        samples.append(TrainingSample(tokens_a=part))
            else:
                a_end = 1
                if len(chunk) > 1:
                    a_end = rng.randint(1, len(chunk) - 1)
                tokens_a = [t for s in chunk[:a_end] for t in s]
                is_random = False
                if len(chunk) == a_end or rng.random() < nsp_probability:
                    # random next from another document
                    is_random = True
                    target_b = target_len - len(tokens_a)
                    for _ in range(10):
                        rand_doc_idx = rng.randrange(len(documents))
                        if rand_doc_idx != doc_idx and documents[rand_doc_idx]:
                            break
                    rand_doc = documents[rand_doc_idx]
                    tokens_b = []
                    start = rng.randrange(len(rand_doc))
                    for s in rand_doc[start:]:
                        tokens_b.extend(s)
                        if len(tokens_b) >= target_b:
                            break
                    # unused true-next sentences go back into the stream
                    i -= len(chunk) - a_end
                else:
                    tokens_b = [t for s in chunk[a_end:] for t in s]
                _truncate_pair(tokens_a, tokens_b, max_tokens, rng)
                if tokens_a and tokens_b:
                    samples.append(
                        TrainingSample(tokens_a, tokens_b, is_random)
                    )
            chunk = []
            chunk_len = 0
            target_len = max_tokens
            if rng.random() < short_seq_prob:
                target_len = rng.randint(2, max_tokens)
        i += 1
    return samples


def _truncate_pair(
    tokens_a: List[int], tokens_b: List[int], max_tokens: int,
    rng: random.Random,
) -> None:
    """Trim the longer side, front or back at random (reference style)."""
    while len(tokens_a) + len(tokens_b) > max_tokens:
        longer = tokens_a if len(tokens_a) >= len(tokens_b) else tokens_b
        if rng.random() < 0.5:
            longer.pop(0)
        else:
            longer.pop()


def write_samples_to_hdf5(
    path: str, samples: List[TrainingSample], cls_id: int, sep_id: int,
    max_seq_len: int, nsp: bool,
) -> int:
    """Reference schema (encode_data.py:204-210), gzip'd via h5lite."""
    n = len(samples)
    width = 3 if nsp else 2
    input_ids = np.zeros((n, max_seq_len), dtype=np.int32)
    special = np.zeros((n, width), dtype=np.int32)
    nsl = np.zeros((n,), dtype=np.int8)
    for j, s in enumerate(samples):
        ids, sp, lab = s.encode(cls_id, sep_id, max_seq_len, nsp)
        input_ids[j] = ids
        special[j] = sp
        nsl[j] = lab
    h5lite.write(
        path,
        {
            "input_ids": input_ids,
            "special_token_positions": special,
            "next_sentence_labels": nsl,
        },
    )
    return n


def _make_tokenizer(args):
    from bert_pytorch_amd.data.tokenization import (
        get_bpe_tokenizer,
        get_wordpiece_tokenizer,
    )

    if args.tokenizer == "wordpiece":
        return get_wordpiece_tokenizer(args.vocab_file, lowercase=args.lowercase)
    return get_bpe_tokenizer(args.vocab_file, args.merges_file)


def encode_shard(job) -> int:
    args, in_path, out_path, seed = job
    tokenizer = _make_tokenizer(args)
    rng = random.Random(seed)
    documents = read_documents(in_path, tokenizer)
    if not documents:
        return 0
    samples: List[TrainingSample] = []
    for doc_idx in range(len(documents)):
        samples.extend(
            create_samples_from_document(
                documents, doc_idx, args.max_seq_len,
                args.nsp_probability, args.short_seq_prob, rng,
            )
        )
    rng.shuffle(samples)  # intra-shard shuffle: the runtime sampler is
    # sequential by design (reference: encode_data.py:179)
    nsp = args.nsp_probability > 0
    cls_id = tokenizer.token_to_id(args.cls_token) or 101
    sep_id = tokenizer.token_to_id(args.sep_token) or 102
    return write_samples_to_hdf5(
        out_path, samples, cls_id, sep_id, args.max_seq_len, nsp
    )


def parse_args(argv: Optional[List[str]] = None):
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--input_dir", required=True)
    p.add_argument("--output_dir", required=True)
    p.add_argument("--tokenizer", choices=["wordpiece", "bpe"],
                   default="wordpiece")
    p.add_argument("--vocab_file", required=True)
    p.add_argument("--merges_file", default=None)
    p.add_argument("--lowercase", action="store_true")
    p.add_argument("--max_seq_len", type=int, default=128)
    p.add_argument("--nsp_probability", type=float, default=0.5,
                   help="0 disables NSP (RoBERTa layout)")
    p.add_argument("--short_seq_prob", type=float, default=0.1)
    p.add_argument("--cls_token", default="[CLS]")
    p.add_argument("--sep_token", default="[SEP]")
    p.add_argument("--processes", type=int, default=os.cpu_count())
    p.add_argument("--seed", type=int, default=1234)
    return p.parse_args(argv)


def main(argv: Optional[List[str]] = None) -> None:
    args = parse_args(argv)
    os.makedirs(args.output_dir, exist_ok=True)
    shards = sorted(Path(args.input_dir).glob("*.txt"))
    if not shards:
        raise SystemExit(f"no *.txt shards under {args.input_dir}")
    jobs = [
        (args, str(s), os.path.join(args.output_dir, f"train_{i}.hdf5"),
         args.seed + i)
        for i, s in enumerate(shards)
    ]
    if args.processes > 1 and len(jobs) > 1:
        with mp.Pool(min(args.processes, len(jobs))) as pool:
            counts = pool.map(encode_shard, jobs)
    else:
        counts = [encode_shard(j) for j in jobs]
    total = sum(counts)
    print(f"encoded {total} samples into {len(jobs)} shards -> {args.output_dir}")


if __name__ == "__main__":
    main()
