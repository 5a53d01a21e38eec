#!/usr/bin/env python3
"""Re-shard formatted text at article boundaries into ~N-byte shards.

Equivalent of the reference's utils/shard.py (shard() :6-27): reads
one-sentence-per-line files with blank-line article separators and
writes shards of approximately --shard_size bytes, never splitting an
article across shards.
"""

from __future__ import annotations

import argparse
import glob
from pathlib import Path
from typing import Iterator, List, Optional


def iter_articles(paths: List[str]) -> Iterator[List[str]]:
    for path in paths:
        with open(path, encoding="utf-8") as f:
            buf: List[str] = []
            for line in f:
                line = line.rstrip("\n")
                if not line.strip():
                    if buf:
                        yield buf
                        buf = []
                else:
                    buf.append(line)
            if buf:
                yield buf


def shard(paths: List[str], output_dir: str, shard_size: int) -> int:
    out = Path(output_dir)
    out.mkdir(parents=True, exist_ok=True)
    idx, written = 0, 0
    f = open(out / f"shard_{idx:04d}.txt", "w", encoding="utf-8")
    for article in iter_articles(paths):
        if written >= shard_size:
            f.close()
            idx += 1
            written = 0
            f = open(out / f"shard_{idx:04d}.txt", "w", encoding="utf-8")
        blob = "\n".join(article) + "\n\n"
        f.write(blob)
        written += len(blob.encode("utf-8"))
    f.close()
    return idx + 1


def main(argv: Optional[List[str]] = None) -> None:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--input_glob", required=True)
    p.add_argument("--output_dir", required=True)
    p.add_argument("--shard_size", type=int, default=64 * 1024 * 1024,
                   help="approximate shard size in bytes")
    args = p.parse_args(argv)
    paths = sorted(glob.glob(args.input_glob, recursive=True))
    if not paths:
        raise SystemExit(f"no files match {args.input_glob}")
    n = shard(paths, args.output_dir, args.shard_size)
    print(f"wrote {n} shards -> {args.output_dir}")


if __name__ == "__main__":
    main()
