#!/usr/bin/env python3
"""Format raw corpora into one-sentence-per-line text shards.

MI355X-native equivalent of the reference's utils/format.py
(Formatter.format :28-63, WikiCorpusFormatter :97-118,
BooksCorpusFormatter :126-143). Output format consumed by
utils/encode_data.py: one sentence per line, one blank line between
articles/documents, round-robin sharded across ``--shards`` files.

The sentence splitter is in-repo (the reference used nltk punkt, which
needs a downloaded model): a rule-based splitter with an abbreviation
list, good enough for wiki/books prose.
"""

from __future__ import annotations

import argparse
import glob
import json
import multiprocessing as mp
import os
import re
from typing import Iterable, Iterator, List, Optional

_ABBREV = {
    "mr", "mrs", "ms", "dr", "prof", "sr", "jr", "st", "vs", "etc", "eg",
    "ie", "cf", "al", "inc", "ltd", "co", "corp", "no", "vol", "pp", "ed",
    "fig", "approx", "dept", "est", "min", "max", "jan", "feb", "mar",
    "apr", "jun", "jul", "aug", "sep", "sept", "oct", "nov", "dec", "u.s",
    "u.k", "a.m", "p.m", "e.g", "i.e",
}

_BOUNDARY = re.compile(r"([.!?][\"')\]]*)\s+(?=[\"'(\[]*[A-Z0-9])")


def split_sentences(text: str) -> List[str]:
    """Rule-based sentence splitting with abbreviation protection."""
    text = " ".join(text.split())
    if not text:
        return []
    pieces: List[str] = []
    start = 0
    for m in _BOUNDARY.finditer(text):
        candidate = text[start : m.end(1)]
        last_word = candidate.rstrip(".!?\"')]").rsplit(" ", 1)[-1].lower()
        if last_word in _ABBREV or (len(last_word) == 1 and last_word.isalpha()):
            continue  # abbreviation / initial, not a boundary
        pieces.append(candidate.strip())
        start = m.end()
    tail = text[start:].strip()
    if tail:
        pieces.append(tail)
    return [p for p in pieces if p]


class Formatter:
    """Base: iterate articles from input files, write sharded output
    (reference: format.py:28-63)."""

    def __init__(self, output_dir: str, shards: int = 256,
                 min_sentences: int = 1):
        self.output_dir = output_dir
        self.shards = shards
        self.min_sentences = min_sentences

    def articles(self, path: str) -> Iterator[List[str]]:
        raise NotImplementedError

    def format(self, input_paths: List[str], processes: int = 1) -> int:
        os.makedirs(self.output_dir, exist_ok=True)
        outs = [
            open(os.path.join(self.output_dir, f"shard_{i:04d}.txt"),
                 "w", encoding="utf-8")
            for i in range(self.shards)
        ]
        n_articles = 0
        try:
            if processes > 1 and len(input_paths) > 1:
                with mp.Pool(processes) as pool:
                    it: Iterable[List[List[str]]] = pool.imap(
                        self._collect, input_paths
                    )
                    for file_articles in it:
                        for art in file_articles:
                            self._write(outs, n_articles, art)
                            n_articles += 1
            else:
                for path in input_paths:
                    for art in self.articles(path):
                        if len(art) >= self.min_sentences:
                            self._write(outs, n_articles, art)
                            n_articles += 1
        finally:
            for f in outs:
                f.close()
        return n_articles

    def _collect(self, path: str) -> List[List[str]]:
        return [a for a in self.articles(path) if len(a) >= self.min_sentences]

    def _write(self, outs, idx: int, article: List[str]) -> None:
        f = outs[idx % self.shards]
        for sentence in article:
            f.write(sentence + "\n")
        f.write("\n")


class WikiCorpusFormatter(Formatter):
    """wikiextractor output (``<doc ...>`` blocks or ``--json`` lines)
    -> articles (reference: format.py:97-118)."""

    def articles(self, path: str) -> Iterator[List[str]]:
        with open(path, encoding="utf-8") as f:
            first = f.read(1)
            f.seek(0)
            if first == "{":  # wikiextractor --json
                for line in f:
                    line = line.strip()
                    if not line:
                        continue
                    obj = json.loads(line)
                    sents = split_sentences(obj.get("text", ""))
                    if sents:
                        yield sents
            else:  # <doc id=...>text</doc>
                buf: List[str] = []
                in_doc = False
                for line in f:
                    line = line.strip()
                    if line.startswith("<doc"):
                        in_doc, buf = True, []
                    elif line.startswith("</doc"):
                        in_doc = False
                        sents = split_sentences(" ".join(buf[1:]))  # drop title
                        if sents:
                            yield sents
                    elif in_doc and line:
                        buf.append(line)


class BooksCorpusFormatter(Formatter):
    """One book per .txt file -> one article per book
    (reference: format.py:126-143)."""

    def articles(self, path: str) -> Iterator[List[str]]:
        with open(path, encoding="utf-8", errors="ignore") as f:
            sents = split_sentences(f.read())
        if sents:
            yield sents


FORMATTERS = {"wikicorpus": WikiCorpusFormatter, "bookscorpus": BooksCorpusFormatter}


def parse_args(argv: Optional[List[str]] = None):
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--dataset", choices=sorted(FORMATTERS), required=True)
    p.add_argument("--input_glob", required=True,
                   help="glob of raw input files (wikiextractor output / book txts)")
    p.add_argument("--output_dir", required=True)
    p.add_argument("--shards", type=int, default=256)
    p.add_argument("--min_sentences", type=int, default=3)
    p.add_argument("--processes", type=int, default=os.cpu_count())
    return p.parse_args(argv)


def main(argv: Optional[List[str]] = None) -> None:
    args = parse_args(argv)
    paths = sorted(glob.glob(args.input_glob, recursive=True))
    if not paths:
        raise SystemExit(f"no files match {args.input_glob}")
    fmt = FORMATTERS[args.dataset](args.output_dir, args.shards,
                                   args.min_sentences)
    n = fmt.format(paths, args.processes)
    print(f"formatted {n} articles into {args.shards} shards -> {args.output_dir}")


if __name__ == "__main__":
    main()
