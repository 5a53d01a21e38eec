"""Tokenization: C++-core WordPiece & byte-level BPE + pure-Python
legacy tokenizers.

Replaces the reference's HuggingFace Rust `tokenizers` dependency
(src/tokenization.py:4,42-57; SURVEY.md §2.2 N9 — Rust is unavailable
here, so the hot encode loops live in csrc/tok/tokenizer.cpp) while
keeping the call surface the runners use: ``encode(text,
add_special_tokens=...)`` returning an object with ``.tokens``/``.ids``,
``token_to_id``, plus the legacy ``BasicTokenizer``/
``WordpieceTokenizer``/``BertTokenizer`` the SQuAD answer-text alignment
needs (reference: src/tokenization.py:60-277).

The C++ core is optional at import time: a pure-Python WordPiece
fallback keeps CPU-only environments working.
"""

from __future__ import annotations

import unicodedata
from typing import Dict, List, Optional, Sequence


def _load_cpp():
    try:
        import torch  # noqa: F401  (libc10 must be resident first)
        from bert_pytorch_amd import _C  # noqa: PLC0415

        return _C
    except ImportError:
        return None


# ---------------------------------------------------------------------------
# character classes (BERT conventions)
# ---------------------------------------------------------------------------
def _is_whitespace(ch: str) -> bool:
    if ch in (" ", "\t", "\n", "\r"):
        return True
    return unicodedata.category(ch) == "Zs"


def _is_control(ch: str) -> bool:
    if ch in ("\t", "\n", "\r"):
        return False
    return unicodedata.category(ch).startswith("C")


def _is_punctuation(ch: str) -> bool:
    cp = ord(ch)
    if (33 <= cp <= 47) or (58 <= cp <= 64) or (91 <= cp <= 96) or (123 <= cp <= 126):
        return True
    return unicodedata.category(ch).startswith("P")


def _is_cjk(cp: int) -> bool:
    return (
        0x4E00 <= cp <= 0x9FFF
        or 0x3400 <= cp <= 0x4DBF
        or 0x20000 <= cp <= 0x2A6DF
        or 0x2A700 <= cp <= 0x2B73F
        or 0x2B740 <= cp <= 0x2B81F
        or 0x2B820 <= cp <= 0x2CEAF
        or 0xF900 <= cp <= 0xFAFF
        or 0x2F800 <= cp <= 0x2FA1F
    )


class BasicTokenizer:
    """Whitespace/punctuation/CJK splitting with optional lowercasing
    and accent stripping (reference semantics: src/tokenization.py:60-173)."""

    NEVER_SPLIT = ("[UNK]", "[SEP]", "[PAD]", "[CLS]", "[MASK]")

    def __init__(self, do_lower_case: bool = True, never_split=None):
        self.do_lower_case = do_lower_case
        self.never_split = (
            tuple(never_split) if never_split is not None else self.NEVER_SPLIT
        )

    def tokenize(self, text: str) -> List[str]:
        text = self._clean(text)
        text = self._pad_cjk(text)
        tokens = text.strip().split()
        out: List[str] = []
        for token in tokens:
            # special tokens pass through verbatim (reference:
            # src/tokenization.py:64-65,74) so e.g. a literal "[UNK]" in
            # raw SQuAD text is not lowercased/punct-split
            if token in self.never_split:
                out.append(token)
                continue
            if self.do_lower_case:
                token = token.lower()
                token = self._strip_accents(token)
            out.extend(self._split_punct(token))
        return " ".join(out).strip().split()

    @staticmethod
    def _clean(text: str) -> str:
        return "".join(
            " " if _is_whitespace(c) else c
            for c in text
            if ord(c) != 0 and ord(c) != 0xFFFD and not _is_control(c)
        )

    @staticmethod
    def _pad_cjk(text: str) -> str:
        return "".join(f" {c} " if _is_cjk(ord(c)) else c for c in text)

    @staticmethod
    def _strip_accents(text: str) -> str:
        return "".join(
            c for c in unicodedata.normalize("NFD", text)
            if unicodedata.category(c) != "Mn"
        )

    @staticmethod
    def _split_punct(token: str) -> List[str]:
        out: List[List[str]] = []
        start_new = True
        for ch in token:
            if _is_punctuation(ch):
                out.append([ch])
                start_new = True
            else:
                if start_new:
                    out.append([])
                    start_new = False
                out[-1].append(ch)
        return ["".join(x) for x in out]


class WordpieceTokenizer:
    """Pure-Python greedy longest-match WordPiece (legacy API and CPU
    fallback; reference: src/tokenization.py:176-229)."""

    def __init__(self, vocab: Dict[str, int], unk_token: str = "[UNK]",
                 max_input_chars_per_word: int = 100):
        self.vocab = vocab
        self.unk_token = unk_token
        self.max_input_chars_per_word = max_input_chars_per_word

    def tokenize(self, text: str) -> List[str]:
        output: List[str] = []
        for word in text.strip().split():
            if len(word) > self.max_input_chars_per_word:
                output.append(self.unk_token)
                continue
            start, sub, bad = 0, [], False
            while start < len(word):
                end = len(word)
                cur = None
                while start < end:
                    piece = word[start:end]
                    if start > 0:
                        piece = "##" + piece
                    if piece in self.vocab:
                        cur = piece
                        break
                    end -= 1
                if cur is None:
                    bad = True
                    break
                sub.append(cur)
                start = end
            output.extend([self.unk_token] if bad else sub)
        return output


class Encoding:
    __slots__ = ("tokens", "ids")

    def __init__(self, tokens: List[str], ids: List[int]):
        self.tokens = tokens
        self.ids = ids


def load_vocab(vocab_file: str) -> Dict[str, int]:
    vocab: Dict[str, int] = {}
    with open(vocab_file, "r", encoding="utf-8") as f:
        for i, line in enumerate(f):
            token = line.rstrip("\n")
            if token:
                vocab[token] = i
    return vocab


class WordPieceTokenizer:
    """HF-BertWordPieceTokenizer-shaped API over the C++ core."""

    def __init__(self, vocab_file: str, lowercase: bool = True,
                 unk_token: str = "[UNK]", cls_token: str = "[CLS]",
                 sep_token: str = "[SEP]"):
        self.vocab = load_vocab(vocab_file)
        self.ids_to_tokens = {i: t for t, i in self.vocab.items()}
        self.basic = BasicTokenizer(do_lower_case=lowercase)
        self.unk_token, self.cls_token, self.sep_token = (
            unk_token, cls_token, sep_token,
        )
        self._cpp = _load_cpp()
        self._handle = None
        if self._cpp is not None:
            tokens = [self.ids_to_tokens[i] for i in range(len(self.vocab))]
            self._handle = self._cpp.tok_create_wordpiece(tokens, unk_token)
        self._py_wp = WordpieceTokenizer(self.vocab, unk_token)

    def vocab_size(self) -> int:
        return len(self.vocab)

    def get_vocab(self) -> Dict[str, int]:
        return dict(self.vocab)

    def token_to_id(self, token: str) -> Optional[int]:
        return self.vocab.get(token)

    def id_to_token(self, idx: int) -> Optional[str]:
        return self.ids_to_tokens.get(idx)

    def tokenize(self, text: str) -> List[str]:
        words = self.basic.tokenize(text)
        if self._handle is not None:
            tokens, _ = self._cpp.tok_encode_wordpiece(self._handle, words)
            return list(tokens)
        out: List[str] = []
        for w in words:
            out.extend(self._py_wp.tokenize(w))
        return out

    def encode(self, text: str, add_special_tokens: bool = True) -> Encoding:
        tokens = self.tokenize(text)
        if add_special_tokens:
            tokens = [self.cls_token] + tokens + [self.sep_token]
        ids = [self.vocab.get(t, self.vocab.get(self.unk_token, 0)) for t in tokens]
        return Encoding(tokens, ids)

    def convert_tokens_to_ids(self, tokens: Sequence[str]) -> List[int]:
        unk = self.vocab.get(self.unk_token, 0)
        return [self.vocab.get(t, unk) for t in tokens]

    def convert_ids_to_tokens(self, ids: Sequence[int]) -> List[str]:
        return [self.ids_to_tokens.get(i, self.unk_token) for i in ids]


# byte-level mapping (GPT-2 style): bytes -> printable unicode chars
def _bytes_to_unicode() -> Dict[int, str]:
    bs = (
        list(range(ord("!"), ord("~") + 1))
        + list(range(ord("\xa1"), ord("\xac") + 1))
        + list(range(ord("\xae"), ord("\xff") + 1))
    )
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return dict(zip(bs, [chr(c) for c in cs]))


_BYTE_ENC = _bytes_to_unicode()
_BYTE_DEC = {v: k for k, v in _BYTE_ENC.items()}


class ByteLevelBPETokenizer:
    """Byte-level BPE over the C++ merge core (RoBERTa-style)."""

    def __init__(self, vocab_file: str, merges_file: str,
                 lowercase: bool = False):
        import json  # noqa: PLC0415

        with open(vocab_file, "r", encoding="utf-8") as f:
            first = f.read(1)
            f.seek(0)
            if first == "{":
                self.vocab = json.load(f)
            else:
                self.vocab = {line.rstrip("\n"): i for i, line in enumerate(f)
                              if line.rstrip("\n")}
        self.ids_to_tokens = {i: t for t, i in self.vocab.items()}
        with open(merges_file, "r", encoding="utf-8") as f:
            merges = [ln.rstrip("\n") for ln in f
                      if ln.strip() and not ln.startswith("#version")]
        self.lowercase = lowercase
        self._cpp = _load_cpp()
        self._handle = None
        self._merge_ranks = {tuple(m.split(" ")): i for i, m in enumerate(merges)}
        if self._cpp is not None:
            tokens = [self.ids_to_tokens.get(i, "") for i in range(len(self.vocab))]
            self._handle = self._cpp.tok_create_bpe(tokens, merges)

    def vocab_size(self) -> int:
        return len(self.vocab)

    def get_vocab(self) -> Dict[str, int]:
        return dict(self.vocab)

    def token_to_id(self, token: str) -> Optional[int]:
        return self.vocab.get(token)

    def id_to_token(self, idx: int) -> Optional[str]:
        return self.ids_to_tokens.get(idx)

    # GPT-2 / RoBERTa pre-tokenization pattern (contractions, letter
    # runs, digit runs, punctuation runs — each optionally taking one
    # leading space). Needs \p classes, so the `regex` module; without
    # it a simplified whitespace-prefix split is used (documented
    # divergence: punctuation is then folded into words).
    _GPT2_PAT = (
        r"""'s|'t|'re|'ve|'m|'ll|'d|"""
        r""" ?\p{L}+| ?\p{N}+| ?[^\s\p{L}\p{N}]+|\s+(?!\S)|\s+"""
    )

    def _pretokenize(self, text: str) -> List[str]:
        if self.lowercase:
            text = text.lower()
        try:
            import regex  # noqa: PLC0415

            words = regex.findall(self._GPT2_PAT, text)
        except ImportError:  # pragma: no cover - regex ships in-image
            words = []
            current = ""
            for ch in text:
                if ch == " ":
                    if current:
                        words.append(current)
                    current = " "
                else:
                    current += ch
            if current:
                words.append(current)
        return [
            "".join(_BYTE_ENC[b] for b in w.encode("utf-8")) for w in words
        ]

    def tokenize(self, text: str) -> List[str]:
        pretokens = self._pretokenize(text)
        if self._handle is not None:
            tokens, _ = self._cpp.tok_encode_bpe(self._handle, pretokens)
            return list(tokens)
        return [t for pt in pretokens for t in self._py_bpe(pt)]

    def _py_bpe(self, word: str) -> List[str]:
        parts = list(word)
        while len(parts) > 1:
            ranked = [
                (self._merge_ranks.get((parts[i], parts[i + 1]), 1 << 30), i)
                for i in range(len(parts) - 1)
            ]
            rank, i = min(ranked)
            if rank == 1 << 30:
                break
            parts[i : i + 2] = [parts[i] + parts[i + 1]]
        return parts

    def encode(self, text: str, add_special_tokens: bool = True) -> Encoding:
        tokens = self.tokenize(text)
        if add_special_tokens and "<s>" in self.vocab:
            tokens = ["<s>"] + tokens + ["</s>"]
        ids = [self.vocab.get(t, self.vocab.get("<unk>", 0)) for t in tokens]
        return Encoding(tokens, ids)


def get_wordpiece_tokenizer(vocab_file: str, lowercase: bool = True,
                            **kw) -> WordPieceTokenizer:
    """Reference factory (src/tokenization.py:42-48)."""
    return WordPieceTokenizer(vocab_file, lowercase=lowercase, **kw)


def get_bpe_tokenizer(vocab_file: str, merges_file: str,
                      lowercase: bool = False, **kw) -> ByteLevelBPETokenizer:
    """Reference factory (src/tokenization.py:51-57)."""
    return ByteLevelBPETokenizer(vocab_file, merges_file, lowercase=lowercase, **kw)


class BertTokenizer:
    """Legacy full tokenizer (basic + wordpiece), used by the SQuAD
    answer alignment (reference: src/tokenization.py:232-277)."""

    def __init__(self, vocab_file: str, do_lower_case: bool = True,
                 max_len: Optional[int] = None):
        self.vocab = load_vocab(vocab_file)
        self.ids_to_tokens = {i: t for t, i in self.vocab.items()}
        self.basic_tokenizer = BasicTokenizer(do_lower_case=do_lower_case)
        self.wordpiece_tokenizer = WordpieceTokenizer(self.vocab)
        self.max_len = max_len or int(1e12)

    def tokenize(self, text: str) -> List[str]:
        tokens: List[str] = []
        for word in self.basic_tokenizer.tokenize(text):
            tokens.extend(self.wordpiece_tokenizer.tokenize(word))
        return tokens

    def convert_tokens_to_ids(self, tokens: Sequence[str]) -> List[int]:
        unk = self.vocab.get("[UNK]", 0)
        return [self.vocab.get(t, unk) for t in tokens]

    def convert_ids_to_tokens(self, ids: Sequence[int]) -> List[str]:
        return [self.ids_to_tokens[i] for i in ids]
