#!/usr/bin/env python3
"""Generate synthetic finetune data at real task shapes (no network).

Writes, under --out:
  vocab.txt   BERT-style WordPiece vocab (30,522 tokens by default)
  squad_train.json / squad_predict.json   SQuAD v1.1-schema QA data
  train.txt / valid.txt                   CoNLL-2003-style NER data

Text is drawn from whole vocab words so WordPiece tokenization is
non-degenerate; answers are real spans of the context with correct
char offsets (exercising run_squad.py's answer alignment the same way
real data does). Weights are random-init in the evidence runs, so task
metrics are meaningless — these files exist to measure the finetune
runners' training/inference sequences-per-second on MI355X at the
reference's operating points (SQuAD seq 384 / NER seq 128).
"""

from __future__ import annotations

import argparse
import json
import os
import random

SPECIALS = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]"]


def write_vocab(path: str, size: int, rng: random.Random) -> list:
    words = []
    alphabet = "abcdefghijklmnopqrstuvwxyz"
    seen = set(SPECIALS)
    while len(words) < size - len(SPECIALS) - 1000:
        w = "".join(rng.choice(alphabet) for _ in range(rng.randint(3, 9)))
        if w not in seen:
            seen.add(w)
            words.append(w)
    subs = []
    while len(subs) < 1000:
        s = "##" + "".join(rng.choice(alphabet) for _ in range(rng.randint(2, 5)))
        if s not in seen:
            seen.add(s)
            subs.append(s)
    vocab = SPECIALS + words + subs
    with open(path, "w") as f:
        f.write("\n".join(vocab) + "\n")
    return words


def write_squad(out_dir: str, words: list, rng: random.Random,
                n_train: int, n_predict: int) -> None:
    def make_paragraph(n_qas: int, qa_start: int):
        ctx_words = [rng.choice(words) for _ in range(170)]
        context = " ".join(ctx_words)
        qas = []
        for q in range(n_qas):
            a_idx = rng.randint(0, len(ctx_words) - 4)
            answer = " ".join(ctx_words[a_idx : a_idx + 3])
            start = len(" ".join(ctx_words[:a_idx]))
            if a_idx:
                start += 1
            qas.append({
                "id": f"q{qa_start + q}",
                "question": " ".join(rng.choice(words) for _ in range(9)) + "?",
                "answers": [{"text": answer, "answer_start": start}],
            })
        return {"context": context, "qas": qas}

    for name, n in (("squad_train.json", n_train), ("squad_predict.json", n_predict)):
        paras, made = [], 0
        while made < n:
            k = min(2, n - made)
            paras.append(make_paragraph(k, made))
            made += k
        doc = {"version": "1.1",
               "data": [{"title": "synthetic", "paragraphs": paras}]}
        with open(os.path.join(out_dir, name), "w") as f:
            json.dump(doc, f)


NER_TAGS = ["O", "O", "O", "O", "B-PER", "I-PER", "B-ORG", "I-ORG",
            "B-LOC", "I-LOC", "B-MISC", "I-MISC"]


def write_conll(out_dir: str, words: list, rng: random.Random,
                n_train: int, n_valid: int) -> None:
    for name, n in (("train.txt", n_train), ("valid.txt", n_valid)):
        with open(os.path.join(out_dir, name), "w") as f:
            for _ in range(n):
                for _ in range(rng.randint(8, 16)):
                    f.write(f"{rng.choice(words)} X X {rng.choice(NER_TAGS)}\n")
                f.write("\n")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", required=True)
    p.add_argument("--vocab_size", type=int, default=30522)
    p.add_argument("--squad_train", type=int, default=1024)
    p.add_argument("--squad_predict", type=int, default=256)
    p.add_argument("--ner_train", type=int, default=2000)
    p.add_argument("--ner_valid", type=int, default=400)
    p.add_argument("--seed", type=int, default=7)
    args = p.parse_args()
    os.makedirs(args.out, exist_ok=True)
    rng = random.Random(args.seed)
    words = write_vocab(os.path.join(args.out, "vocab.txt"),
                        args.vocab_size, rng)
    write_squad(args.out, words, rng, args.squad_train, args.squad_predict)
    write_conll(args.out, words, rng, args.ner_train, args.ner_valid)
    print(f"wrote synthetic finetune data to {args.out}")


if __name__ == "__main__":
    main()
