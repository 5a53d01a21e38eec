"""Conformance of the in-repo tokenizer (C++ core + Python layer,
SURVEY §2.2 N9) against the HuggingFace Rust ``tokenizers`` library —
the exact dependency the reference imports (src/tokenization.py:4) —
configured as the canonical BERT pipeline (BertNormalizer +
BertPreTokenizer + WordPiece). ``tokenizers`` is in this image's
wheelhouse; only a local vocab file is needed (no network)."""

import pytest

tokenizers = pytest.importorskip("tokenizers")

from bert_pytorch_amd.data.tokenization import (  # noqa: E402
    BertTokenizer as OurBertTokenizer,
)

VOCAB = [
    "[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]",
    "the", "th", "##e", "##ere", "there", "a", "##b", "##c", "ab",
    "un", "##aff", "##able", "##wanted", "want", "##ed", "runn", "##ing",
    "hello", "world", "!", ",", ".", "-", "'", "s", "##s",
    "1", "2", "##9", "##8", "19", "##99",
    "你", "好", "吗",  # CJK chars
    "caf", "cafe", "na", "##ive", "##ve", "x", "##x",
]

TEXTS = [
    "Hello world!",
    "there there THERE ThErE",
    "unaffable unwanted running",
    "café naive naïve",  # precomposed + combining accents
    "你好吗? hello,world",  # CJK + ascii punctuation runs
    "it's John's   book  ",
    "1999 19 99 2-1",
    "  　spaced out",  # unicode whitespace
    "x" * 99,    # just under max_input_chars_per_word
    "x" * 101,   # just over -> [UNK] (reference cap 100)
    "control\x00char\x1ftext",
    "emoji \U0001f600 mix",
    "",
]


@pytest.fixture(scope="module")
def vocab_file(tmp_path_factory):
    p = tmp_path_factory.mktemp("vocab") / "vocab.txt"
    p.write_text("\n".join(VOCAB) + "\n", encoding="utf-8")
    return str(p)


def _hf_bert_pipeline(vocab_file, lower):
    from tokenizers import Tokenizer
    from tokenizers.models import WordPiece
    from tokenizers.normalizers import BertNormalizer
    from tokenizers.pre_tokenizers import BertPreTokenizer

    vocab = {}
    with open(vocab_file, encoding="utf-8") as f:
        for i, line in enumerate(f):
            tok = line.rstrip("\n")
            if tok:
                vocab[tok] = i
    t = Tokenizer(
        WordPiece(vocab, unk_token="[UNK]", max_input_chars_per_word=100)
    )
    # strip_accents follows lowercase, the Google-BERT default the
    # reference's BasicTokenizer implements
    t.normalizer = BertNormalizer(
        lowercase=lower, handle_chinese_chars=True, strip_accents=None,
        clean_text=True,
    )
    t.pre_tokenizer = BertPreTokenizer()
    return t


@pytest.mark.parametrize("lower", [True, False])
def test_tokenize_matches_hf_rust(vocab_file, lower):
    ours = OurBertTokenizer(vocab_file, do_lower_case=lower)
    hf = _hf_bert_pipeline(vocab_file, lower)
    for text in TEXTS:
        got = ours.tokenize(text)
        want = hf.encode(text, add_special_tokens=False).tokens
        assert got == want, (text, got, want)


def test_ids_match_hf_rust(vocab_file):
    ours = OurBertTokenizer(vocab_file)
    hf = _hf_bert_pipeline(vocab_file, True)
    for text in TEXTS:
        enc = hf.encode(text, add_special_tokens=False)
        assert ours.convert_tokens_to_ids(ours.tokenize(text)) == enc.ids, text


def test_cpp_wordpiece_class_matches_hf_rust(vocab_file):
    """The C++-backed WordPieceTokenizer (the pretraining encode path)
    agrees with the Rust pipeline too."""
    from bert_pytorch_amd.data.tokenization import WordPieceTokenizer

    ours = WordPieceTokenizer(vocab_file, lowercase=True)
    hf = _hf_bert_pipeline(vocab_file, True)
    for text in TEXTS:
        got = ours.tokenize(text)
        want = hf.encode(text, add_special_tokens=False).tokens
        assert got == want, (text, got, want)


def test_byte_bpe_matches_hf_rust(tmp_path):
    """RoBERTa-style byte-level BPE: train vocab+merges with the Rust
    library, encode with both it and the in-repo tokenizer (GPT-2
    regex pre-tokenization + C++ merge core)."""
    from tokenizers import Tokenizer
    from tokenizers.models import BPE
    from tokenizers.pre_tokenizers import ByteLevel
    from tokenizers.decoders import ByteLevel as ByteLevelDecoder
    from tokenizers.trainers import BpeTrainer

    from bert_pytorch_amd.data.tokenization import ByteLevelBPETokenizer

    corpus = [
        "the quick brown fox jumps over the lazy dog",
        "it's John's book, isn't it? Yes -- really!",
        "numbers 1999 and 42 mix with words2000 tokens",
        "Hello, world! hello WORLD... the theater there",
    ] * 8
    corpus_file = tmp_path / "corpus.txt"
    corpus_file.write_text("\n".join(corpus), encoding="utf-8")

    hf = Tokenizer(BPE(unk_token=None))
    hf.pre_tokenizer = ByteLevel(add_prefix_space=False)
    hf.decoder = ByteLevelDecoder()
    trainer = BpeTrainer(
        vocab_size=400, min_frequency=2, special_tokens=["<s>", "</s>"],
        initial_alphabet=ByteLevel.alphabet(),
    )
    hf.train([str(corpus_file)], trainer)

    import json

    model_dir = tmp_path / "bpe"
    model_dir.mkdir()
    blob = json.loads(hf.to_str())
    (model_dir / "vocab.json").write_text(
        json.dumps(blob["model"]["vocab"]), encoding="utf-8"
    )
    merge_lines = [
        m if isinstance(m, str) else " ".join(m)
        for m in blob["model"]["merges"]
    ]
    (model_dir / "merges.txt").write_text(
        "\n".join(merge_lines) + "\n", encoding="utf-8"
    )

    ours = ByteLevelBPETokenizer(
        str(model_dir / "vocab.json"), str(model_dir / "merges.txt")
    )
    texts = corpus[:4] + [
        "unseen words zigzag quickly!",
        "  leading spaces and   runs",
        "punct,comma.dot-dash's",
        "",
    ]
    for text in texts:
        want = hf.encode(text).tokens
        got = ours.tokenize(text)
        assert got == want, (text, got, want)
        assert [ours.token_to_id(t) for t in got] == hf.encode(text).ids
