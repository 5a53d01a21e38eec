"""Offline data pipeline tests (utils/): format -> shard -> vocab ->
encode -> runtime dataset round trip, all on CPU with tiny synthetic
corpora. Reference pipeline being reproduced: utils/{format,shard,
sample_and_shard,build_vocab,encode_data}.py + scripts/create_datasets.sh.
"""

import json
import os
import random
import subprocess
import sys
from pathlib import Path

import numpy as np
import pytest

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT / "utils"))

from bert_pytorch_amd.data import h5lite  # noqa: E402
from bert_pytorch_amd.data.dataset import ShardedPretrainingDataset  # noqa: E402

import build_vocab  # noqa: E402
import encode_data  # noqa: E402
import format as format_mod  # noqa: E402
import sample_and_shard  # noqa: E402
import shard as shard_mod  # noqa: E402

CORPUS = [
    [  # article 1
        "The quick brown fox jumps over the lazy dog.",
        "Dogs are loyal companions and foxes are wild animals.",
        "The fox returned to the forest at dusk.",
        "Night fell over the quiet forest.",
    ],
    [  # article 2
        "Computers process information using binary arithmetic.",
        "Modern processors contain billions of transistors.",
        "Memory bandwidth often limits application performance.",
        "Parallel computing divides work across many processors.",
    ],
    [  # article 3
        "Rivers flow from mountains toward the sea.",
        "The river delta hosts many species of birds.",
        "Seasonal floods deposit rich sediment on the plains.",
    ],
]


def _write_formatted(path: Path) -> None:
    with open(path, "w", encoding="utf-8") as f:
        for article in CORPUS:
            for s in article:
                f.write(s + "\n")
            f.write("\n")


def test_sentence_splitter():
    text = ("Dr. Smith visited Washington. He arrived at 3 p.m. on Monday. "
            "The meeting, i.e. the annual review, went well!")
    sents = format_mod.split_sentences(text)
    assert len(sents) == 3
    assert sents[0].startswith("Dr. Smith")
    assert sents[0].endswith("Washington.")


def test_wiki_formatter_doc_and_json(tmp_path):
    raw = tmp_path / "wiki_00"
    raw.write_text(
        '<doc id="1" title="A">\nTitle A\nOne sentence here. Another one '
        "follows.\n</doc>\n"
        '<doc id="2" title="B">\nTitle B\nSecond document text.\n</doc>\n',
        encoding="utf-8",
    )
    fmt = format_mod.WikiCorpusFormatter(str(tmp_path / "out"), shards=2,
                                         min_sentences=1)
    n = fmt.format([str(raw)])
    assert n == 2
    joined = "".join(
        (tmp_path / "out" / f"shard_{i:04d}.txt").read_text() for i in range(2)
    )
    assert "One sentence here." in joined
    assert "Title A" not in joined  # titles dropped

    raw_json = tmp_path / "wiki_json"
    raw_json.write_text(
        json.dumps({"text": "Json article sentence one. And sentence two."})
        + "\n",
        encoding="utf-8",
    )
    fmt2 = format_mod.WikiCorpusFormatter(str(tmp_path / "out2"), shards=1,
                                          min_sentences=1)
    assert fmt2.format([str(raw_json)]) == 1


def test_shard_preserves_article_boundaries(tmp_path):
    src = tmp_path / "formatted.txt"
    _write_formatted(src)
    n = shard_mod.shard([str(src)], str(tmp_path / "shards"), shard_size=120)
    assert n >= 2
    total_articles = 0
    for p in sorted((tmp_path / "shards").glob("*.txt")):
        arts = list(shard_mod.iter_articles([str(p)]))
        total_articles += len(arts)
        # no article may be split: every article from CORPUS appears whole
        for a in arts:
            assert a in CORPUS
    assert total_articles == len(CORPUS)


def test_sample_and_shard(tmp_path):
    src = tmp_path / "formatted.txt"
    _write_formatted(src)
    rng = random.Random(0)
    sampled = sample_and_shard.sample_articles([str(src)], 7, rng)
    total = sum(len(a) for a in sampled)
    assert total >= 7
    for a in sampled:
        assert a in CORPUS


def test_build_vocab_wordpiece_and_encode_roundtrip(tmp_path):
    formatted = tmp_path / "formatted"
    formatted.mkdir()
    _write_formatted(formatted / "shard_0000.txt")

    # vocab training
    build_vocab.main([
        "--input_glob", str(formatted / "*.txt"),
        "--output_dir", str(tmp_path / "vocab"),
        "--tokenizer", "wordpiece", "--vocab_size", "400",
        "--lowercase",
    ])
    vocab_file = tmp_path / "vocab" / "vocab.txt"
    vocab = vocab_file.read_text().split("\n")
    assert vocab[0] == "[PAD]"  # [PAD] forced to id 0
    assert "[MASK]" in vocab[:5]

    # tokenizer loads it and round-trips a word
    from bert_pytorch_amd.data.tokenization import get_wordpiece_tokenizer

    tok = get_wordpiece_tokenizer(str(vocab_file), lowercase=True)
    enc = tok.encode("the quick brown fox", add_special_tokens=False)
    assert enc.ids
    assert all(i != tok.token_to_id("[UNK]") for i in enc.ids)

    # encode to HDF5 (NSP on)
    encode_data.main([
        "--input_dir", str(formatted),
        "--output_dir", str(tmp_path / "hdf5"),
        "--tokenizer", "wordpiece", "--lowercase",
        "--vocab_file", str(vocab_file),
        "--max_seq_len", "64", "--nsp_probability", "0.5",
        "--processes", "1",
    ])
    files = sorted((tmp_path / "hdf5").glob("*.hdf5"))
    assert files
    with h5lite.H5LiteFile(str(files[0])) as f:
        ids = np.asarray(f["input_ids"])
        special = np.asarray(f["special_token_positions"])
        nsl = np.asarray(f["next_sentence_labels"])
    assert ids.dtype == np.int32 and ids.shape[1] == 64
    assert special.shape[1] == 3  # [CLS], sep1, sep2
    assert nsl.dtype == np.int8
    cls_id = tok.token_to_id("[CLS]")
    sep_id = tok.token_to_id("[SEP]")
    for row, sp in zip(ids, special):
        assert row[sp[0]] == cls_id
        assert row[sp[1]] == sep_id
        assert row[sp[2]] == sep_id
        assert sp[0] < sp[1] < sp[2]

    # the runtime dataset consumes the shards
    ds = ShardedPretrainingDataset(
        [str(p) for p in files],
        mask_token_index=tok.token_to_id("[MASK]"),
        max_pred_per_seq=10, masked_lm_prob=0.15,
        vocab_size=tok.vocab_size(), seed=0,
    )
    assert len(ds) == len(ids)
    sample = ds[0]
    assert len(sample) == 5
    masked, seg, mask, labels, _nsp = sample
    assert masked.shape == (64,)
    assert (labels >= 0).sum() >= 1  # something got masked


def test_encode_roberta_no_nsp(tmp_path):
    formatted = tmp_path / "formatted"
    formatted.mkdir()
    _write_formatted(formatted / "shard_0000.txt")
    build_vocab.main([
        "--input_glob", str(formatted / "*.txt"),
        "--output_dir", str(tmp_path / "vocab"),
        "--tokenizer", "wordpiece", "--vocab_size", "400", "--lowercase",
    ])
    encode_data.main([
        "--input_dir", str(formatted),
        "--output_dir", str(tmp_path / "hdf5"),
        "--tokenizer", "wordpiece", "--lowercase",
        "--vocab_file", str(tmp_path / "vocab" / "vocab.txt"),
        "--max_seq_len", "64", "--nsp_probability", "0",
        "--processes", "1",
    ])
    files = sorted((tmp_path / "hdf5").glob("*.hdf5"))
    with h5lite.H5LiteFile(str(files[0])) as f:
        special = np.asarray(f["special_token_positions"])
        nsl = np.asarray(f["next_sentence_labels"])
    assert special.shape[1] == 2  # no NSP: [CLS], [SEP] only
    assert (nsl == 0).all()


def test_build_vocab_bpe(tmp_path):
    formatted = tmp_path / "formatted"
    formatted.mkdir()
    _write_formatted(formatted / "shard_0000.txt")
    build_vocab.main([
        "--input_glob", str(formatted / "*.txt"),
        "--output_dir", str(tmp_path / "vocab"),
        "--tokenizer", "bpe", "--vocab_size", "360",
    ])
    vocab = json.loads((tmp_path / "vocab" / "vocab.json").read_text())
    merges = (tmp_path / "vocab" / "merges.txt").read_text().splitlines()
    assert vocab["[PAD]"] == 0
    assert merges[0].startswith("#version")
    from bert_pytorch_amd.data.tokenization import get_bpe_tokenizer

    tok = get_bpe_tokenizer(
        str(tmp_path / "vocab" / "vocab.json"),
        str(tmp_path / "vocab" / "merges.txt"),
    )
    enc = tok.encode("the quick brown fox", add_special_tokens=False)
    assert enc.ids


def test_download_file_url_and_sha(tmp_path):
    import download as download_mod

    src = tmp_path / "payload.txt"
    src.write_text("hello corpus")
    sha = download_mod.sha256_of(str(src))

    class Local(download_mod.Downloader):
        resources = {"payload.txt": (src.as_uri(), sha)}

    d = Local(str(tmp_path / "out"))
    paths = d.download()
    assert (tmp_path / "out" / "payload.txt").read_text() == "hello corpus"

    class Bad(download_mod.Downloader):
        resources = {"payload2.txt": (src.as_uri(), "0" * 64)}

    with pytest.raises(RuntimeError, match="sha256 mismatch"):
        Bad(str(tmp_path / "out2")).download()


def test_create_datasets_script_vocab_encode(tmp_path):
    """scripts/create_datasets.sh --vocab --encode on a pre-formatted dir."""
    data = tmp_path / "data"
    (data / "formatted").mkdir(parents=True)
    _write_formatted(data / "formatted" / "shard_0000.txt")
    r = subprocess.run(
        ["bash", str(ROOT / "scripts" / "create_datasets.sh"),
         "--data-dir", str(data), "--vocab", "--encode"],
        capture_output=True, text=True, cwd=str(ROOT), timeout=300,
    )
    assert r.returncode == 0, r.stderr
    assert (data / "vocab" / "vocab.txt").exists()
    assert list((data / "hdf5" / "seq128_nsp5").glob("*.hdf5"))
    assert list((data / "hdf5" / "seq512_nsp5").glob("*.hdf5"))
