#!/usr/bin/env python3
"""Randomly sample articles to a target sentence count, then shard.

Equivalent of the reference's utils/sample_and_shard.py (:38-125):
used to build smaller training subsets (e.g. a 10%-of-wiki corpus)
without biasing toward any input file. Articles are sampled whole.
"""

from __future__ import annotations

import argparse
import glob
import random
import sys
from pathlib import Path
from typing import List, Optional

sys.path.insert(0, str(Path(__file__).resolve().parent))
from shard import iter_articles  # noqa: E402


def sample_articles(
    paths: List[str], target_sentences: int, rng: random.Random
) -> List[List[str]]:
    """Reservoir-sample whole articles until roughly target_sentences
    total sentences are kept (single pass, O(target) memory)."""
    kept: List[List[str]] = []
    kept_sentences = 0
    seen = 0
    for article in iter_articles(paths):
        seen += 1
        if kept_sentences < target_sentences:
            kept.append(article)
            kept_sentences += len(article)
            continue
        j = rng.randrange(seen)
        if j < len(kept):
            kept_sentences += len(article) - len(kept[j])
            kept[j] = article
    rng.shuffle(kept)
    # trim overshoot
    out: List[List[str]] = []
    total = 0
    for article in kept:
        if total >= target_sentences:
            break
        out.append(article)
        total += len(article)
    return out


def main(argv: Optional[List[str]] = None) -> None:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--input_glob", required=True)
    p.add_argument("--output_dir", required=True)
    p.add_argument("--target_sentences", type=int, required=True)
    p.add_argument("--shards", type=int, default=256)
    p.add_argument("--seed", type=int, default=1234)
    args = p.parse_args(argv)

    paths = sorted(glob.glob(args.input_glob, recursive=True))
    if not paths:
        raise SystemExit(f"no files match {args.input_glob}")
    rng = random.Random(args.seed)
    articles = sample_articles(paths, args.target_sentences, rng)

    out = Path(args.output_dir)
    out.mkdir(parents=True, exist_ok=True)
    files = [
        open(out / f"shard_{i:04d}.txt", "w", encoding="utf-8")
        for i in range(args.shards)
    ]
    try:
        for i, article in enumerate(articles):
            f = files[i % args.shards]
            for s in article:
                f.write(s + "\n")
            f.write("\n")
    finally:
        for f in files:
            f.close()
    total = sum(len(a) for a in articles)
    print(f"sampled {len(articles)} articles / {total} sentences "
          f"into {args.shards} shards -> {args.output_dir}")


if __name__ == "__main__":
    main()
