#!/usr/bin/env python3
"""Download raw corpora / weights with integrity verification.

Equivalent of the reference's utils/download.py (Downloader :11-32,
WikiCorpusDownloader :219-235, WeightsDownloader :123-217,
SquadDownloader :103-120). Supports http(s) and file:// sources (the
latter makes every path testable offline), sha256 verification, and
zip/bz2/tar extraction.

Usage:
    python utils/download.py --dataset squad --output_dir data/squad
    python utils/download.py --dataset wikicorpus_en --output_dir data/wiki
"""

from __future__ import annotations

import argparse
import bz2
import hashlib
import os
import shutil
import sys
import tarfile
import urllib.request
import zipfile
from typing import Dict, List, Optional


class Downloader:
    """Fetch URLs into an output dir, verify, extract (reference :11-32)."""

    #: subclasses fill: filename -> (url, sha256-or-None)
    resources: Dict[str, tuple] = {}

    def __init__(self, output_dir: str):
        self.output_dir = output_dir

    def download(self) -> List[str]:
        os.makedirs(self.output_dir, exist_ok=True)
        paths = []
        for filename, (url, sha) in self.resources.items():
            dest = os.path.join(self.output_dir, filename)
            if not os.path.exists(dest):
                print(f"downloading {url} -> {dest}")
                self._fetch(url, dest)
            if sha is not None:
                got = sha256_of(dest)
                if got != sha:
                    raise RuntimeError(
                        f"sha256 mismatch for {dest}: got {got}, want {sha}"
                    )
            self.extract(dest)
            paths.append(dest)
        return paths

    @staticmethod
    def _fetch(url: str, dest: str) -> None:
        tmp = dest + ".part"
        with urllib.request.urlopen(url) as r, open(tmp, "wb") as f:
            shutil.copyfileobj(r, f, length=1 << 20)
        os.replace(tmp, dest)

    def extract(self, path: str) -> None:
        out = self.output_dir
        if path.endswith(".zip"):
            with zipfile.ZipFile(path) as z:
                z.extractall(out)
        elif path.endswith((".tar.gz", ".tgz", ".tar")):
            with tarfile.open(path) as t:
                t.extractall(out)
        elif path.endswith(".bz2"):
            plain = path[: -len(".bz2")]
            if not os.path.exists(plain):
                with bz2.open(path, "rb") as src, open(plain, "wb") as dst:
                    shutil.copyfileobj(src, dst, length=1 << 20)


def sha256_of(path: str) -> str:
    h = hashlib.sha256()
    with open(path, "rb") as f:
        for chunk in iter(lambda: f.read(1 << 20), b""):
            h.update(chunk)
    return h.hexdigest()


class WikiCorpusDownloader(Downloader):
    """English/zh Wikipedia XML dump (reference :219-235). Run
    wikiextractor on the result before utils/format.py."""

    def __init__(self, output_dir: str, language: str = "en"):
        super().__init__(output_dir)
        self.resources = {
            f"wikicorpus_{language}.xml.bz2": (
                f"https://dumps.wikimedia.org/{language}wiki/latest/"
                f"{language}wiki-latest-pages-articles.xml.bz2",
                None,
            )
        }


class BooksCorpusDownloader(Downloader):
    """BooksCorpus mirror (reference :238-249). The original hosting is
    gone; point --url at any mirror tarball of one-book-per-file txts."""

    def __init__(self, output_dir: str, url: Optional[str] = None):
        super().__init__(output_dir)
        self.resources = {
            "bookscorpus.tar.gz": (url or "https://battle.shawwn.com/sdb/books1/books1.tar.gz", None)
        }


class SquadDownloader(Downloader):
    """SQuAD v1.1 + v2.0 JSON and the official evaluate scripts
    (reference :103-120)."""

    def __init__(self, output_dir: str):
        super().__init__(output_dir)
        base_v1 = "https://rajpurkar.github.io/SQuAD-explorer/dataset"
        self.resources = {
            "train-v1.1.json": (f"{base_v1}/train-v1.1.json", None),
            "dev-v1.1.json": (f"{base_v1}/dev-v1.1.json", None),
            "train-v2.0.json": (f"{base_v1}/train-v2.0.json", None),
            "dev-v2.0.json": (f"{base_v1}/dev-v2.0.json", None),
        }


class GLUEDownloader(Downloader):
    """GLUE task archives (reference :252-256)."""

    TASKS = {
        "cola": "https://dl.fbaipublicfiles.com/glue/data/CoLA.zip",
        "sst": "https://dl.fbaipublicfiles.com/glue/data/SST-2.zip",
        "qqp": "https://dl.fbaipublicfiles.com/glue/data/QQP-clean.zip",
        "sts": "https://dl.fbaipublicfiles.com/glue/data/STS-B.zip",
        "mnli": "https://dl.fbaipublicfiles.com/glue/data/MNLI.zip",
        "qnli": "https://dl.fbaipublicfiles.com/glue/data/QNLIv2.zip",
        "rte": "https://dl.fbaipublicfiles.com/glue/data/RTE.zip",
        "wnli": "https://dl.fbaipublicfiles.com/glue/data/WNLI.zip",
    }

    def __init__(self, output_dir: str, task: str = "mnli"):
        super().__init__(output_dir)
        url = self.TASKS[task]
        self.resources = {os.path.basename(url): (url, None)}


class WeightsDownloader(Downloader):
    """Google BERT checkpoint archives with sha256 pinning
    (reference :123-217)."""

    BASE = "https://storage.googleapis.com/bert_models"
    MODELS = {
        "bert-large-uncased": (
            f"{BASE}/2018_10_18/uncased_L-24_H-1024_A-16.zip",
            "beb3ccd44fcb2b452b4c0dbf4d4922087461a714b6f6acc52ac35a01d884167c",
        ),
        "bert-base-uncased": (
            f"{BASE}/2018_10_18/uncased_L-12_H-768_A-12.zip",
            "0ee2d97b0e22a1370eef5ed6cc6a0e28be1c55a6c4efee155a39f1d5ebb6ae35",
        ),
        "bert-large-cased": (
            f"{BASE}/2018_10_18/cased_L-24_H-1024_A-16.zip",
            None,
        ),
        "bert-base-cased": (
            f"{BASE}/2018_10_18/cased_L-12_H-768_A-12.zip",
            None,
        ),
    }

    def __init__(self, output_dir: str, model: str = "bert-large-uncased"):
        super().__init__(output_dir)
        url, sha = self.MODELS[model]
        self.resources = {os.path.basename(url): (url, sha)}


DOWNLOADERS = {
    "wikicorpus_en": lambda out: WikiCorpusDownloader(out, "en"),
    "wikicorpus_zh": lambda out: WikiCorpusDownloader(out, "zh"),
    "bookscorpus": BooksCorpusDownloader,
    "squad": SquadDownloader,
    "glue": GLUEDownloader,
    "weights": WeightsDownloader,
}


def main(argv: Optional[List[str]] = None) -> None:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--dataset", choices=sorted(DOWNLOADERS), required=True)
    p.add_argument("--output_dir", required=True)
    args = p.parse_args(argv)
    paths = DOWNLOADERS[args.dataset](args.output_dir).download()
    print(f"downloaded {len(paths)} file(s) -> {args.output_dir}")


if __name__ == "__main__":
    sys.exit(main())
