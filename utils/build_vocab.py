#!/usr/bin/env python3
"""Train a WordPiece or byte-level BPE vocabulary from text shards.

Equivalent of the reference's utils/build_vocab.py (:7-80), which
delegated to the HuggingFace Rust trainers; here training runs through
the in-repo C++ BPE trainer (csrc/tok/tokenizer.cpp train_bpe, greedy
frequency merges) with a pure-Python fallback. As in the reference,
special tokens are forced to the front of the vocab with [PAD] = 0
(reference :69-75).

WordPiece is derived from the BPE merges: each word's final
segmentation contributes its initial piece verbatim and its
continuation pieces with the ``##`` prefix (frequency-merge
approximation of the likelihood-based WordPiece trainer).

Outputs:
    wordpiece:  <out>/vocab.txt
    bpe:        <out>/vocab.json + <out>/merges.txt
"""

from __future__ import annotations

import argparse
import collections
import glob
import json
import os
import sys
from pathlib import Path
from typing import Dict, List, Optional, Tuple

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from bert_pytorch_amd.data.tokenization import (  # noqa: E402
    BasicTokenizer,
    _bytes_to_unicode,
    _load_cpp,
)

SPECIALS_WORDPIECE = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]"]
SPECIALS_BPE = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]"]


def count_words(paths: List[str], lowercase: bool,
                max_lines: int = 0) -> collections.Counter:
    basic = BasicTokenizer(do_lower_case=lowercase)
    counts: collections.Counter = collections.Counter()
    seen = 0
    for path in paths:
        with open(path, encoding="utf-8") as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                counts.update(basic.tokenize(line))
                seen += 1
                if max_lines and seen >= max_lines:
                    return counts
    return counts


def _train_merges(words: List[str], counts: List[int],
                  num_merges: int) -> List[str]:
    cpp = _load_cpp()
    if cpp is not None:
        return cpp.tok_train_bpe(words, counts, num_merges)
    # pure-Python fallback: identical greedy frequency merges
    seqs = [list(w) for w in words]
    merges: List[str] = []
    for _ in range(num_merges):
        pair_counts: collections.Counter = collections.Counter()
        for seq, c in zip(seqs, counts):
            for i in range(len(seq) - 1):
                pair_counts[(seq[i], seq[i + 1])] += c
        if not pair_counts:
            break
        (left, right), best = pair_counts.most_common(1)[0]
        if best < 2:
            break
        merges.append(f"{left} {right}")
        merged = left + right
        for seq in seqs:
            i = 0
            while i < len(seq) - 1:
                if seq[i] == left and seq[i + 1] == right:
                    seq[i] = merged
                    del seq[i + 1]
                else:
                    i += 1
    return merges


def _apply_merges(word: str, merges: List[Tuple[str, str]]) -> List[str]:
    seq = list(word)
    ranks = {m: i for i, m in enumerate(merges)}
    while len(seq) > 1:
        best_rank, best_i = None, -1
        for i in range(len(seq) - 1):
            r = ranks.get((seq[i], seq[i + 1]))
            if r is not None and (best_rank is None or r < best_rank):
                best_rank, best_i = r, i
        if best_rank is None:
            break
        seq[best_i] = seq[best_i] + seq[best_i + 1]
        del seq[best_i + 1]
    return seq


def train_wordpiece(
    word_counts: collections.Counter, vocab_size: int, num_merges: int,
) -> List[str]:
    words = list(word_counts)
    counts = [word_counts[w] for w in words]
    merge_lines = _train_merges(words, counts, num_merges)
    merges = [tuple(m.split(" ", 1)) for m in merge_lines]

    piece_counts: collections.Counter = collections.Counter()
    for w, c in word_counts.items():
        pieces = _apply_merges(w, merges)
        for i, p in enumerate(pieces):
            piece_counts[p if i == 0 else "##" + p] += c
    # alphabet coverage so no word becomes [UNK] purely by character
    for w, c in word_counts.items():
        for i, ch in enumerate(w):
            piece_counts[ch if i == 0 else "##" + ch] += 0

    vocab = list(SPECIALS_WORDPIECE)
    taken = set(vocab)
    for piece, _ in piece_counts.most_common():
        if len(vocab) >= vocab_size:
            break
        if piece not in taken:
            vocab.append(piece)
            taken.add(piece)
    return vocab


def train_byte_bpe(
    word_counts: collections.Counter, vocab_size: int,
) -> Tuple[Dict[str, int], List[str]]:
    b2u = _bytes_to_unicode()
    mapped: collections.Counter = collections.Counter()
    for w, c in word_counts.items():
        mapped["".join(b2u[b] for b in w.encode("utf-8"))] += c
    alphabet = sorted(set(b2u.values()))
    num_merges = max(0, vocab_size - len(alphabet) - len(SPECIALS_BPE))
    words = list(mapped)
    counts = [mapped[w] for w in words]
    merge_lines = _train_merges(words, counts, num_merges)

    vocab: Dict[str, int] = {}
    for tok in SPECIALS_BPE:
        vocab[tok] = len(vocab)
    for ch in alphabet:
        vocab[ch] = len(vocab)
    for m in merge_lines:
        left, right = m.split(" ", 1)
        merged = left + right
        if merged not in vocab:
            vocab[merged] = len(vocab)
    return vocab, merge_lines


def main(argv: Optional[List[str]] = None) -> None:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--input_glob", required=True)
    p.add_argument("--output_dir", required=True)
    p.add_argument("--tokenizer", choices=["wordpiece", "bpe"],
                   default="wordpiece")
    p.add_argument("--vocab_size", type=int, default=30522)
    p.add_argument("--num_merges", type=int, default=0,
                   help="BPE merge budget; default vocab_size-driven")
    p.add_argument("--lowercase", action="store_true")
    p.add_argument("--max_lines", type=int, default=0,
                   help="cap lines read (0 = all)")
    args = p.parse_args(argv)

    paths = sorted(glob.glob(args.input_glob, recursive=True))
    if not paths:
        raise SystemExit(f"no files match {args.input_glob}")
    word_counts = count_words(paths, args.lowercase, args.max_lines)
    os.makedirs(args.output_dir, exist_ok=True)

    if args.tokenizer == "wordpiece":
        num_merges = args.num_merges or args.vocab_size
        vocab = train_wordpiece(word_counts, args.vocab_size, num_merges)
        out = os.path.join(args.output_dir, "vocab.txt")
        with open(out, "w", encoding="utf-8") as f:
            f.write("\n".join(vocab) + "\n")
        print(f"wordpiece vocab: {len(vocab)} tokens -> {out}")
    else:
        vocab, merges = train_byte_bpe(word_counts, args.vocab_size)
        vp = os.path.join(args.output_dir, "vocab.json")
        mp_ = os.path.join(args.output_dir, "merges.txt")
        with open(vp, "w", encoding="utf-8") as f:
            json.dump(vocab, f, ensure_ascii=False)
        with open(mp_, "w", encoding="utf-8") as f:
            f.write("#version: 0.2\n")
            f.write("\n".join(merges) + "\n")
        print(f"byte-level BPE vocab: {len(vocab)} tokens -> {vp}, {mp_}")


if __name__ == "__main__":
    main()
