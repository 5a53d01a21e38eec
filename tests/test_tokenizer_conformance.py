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
