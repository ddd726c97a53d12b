"""Tokenizer for the local embedding model.

Two modes:
- SentencePiece/HF tokenizer.json when a file is provided (the `tokenizers`
  wheel is available offline) — real bge-m3 vocab compatibility.
- HashTokenizer fallback: deterministic word/subword hashing into the
  XLM-R vocab space. No network access means no pretrained vocab file in
  this environment; the hash tokenizer preserves shapes, determinism and
  throughput characteristics (the benchmark contract).
"""

from __future__ import annotations

import re
from typing import List, Sequence, Tuple

_WORD_RE = re.compile(r"\w+|[^\w\s]")

BOS = 0
PAD = 1
EOS = 2
UNK = 3
RESERVED = 4


class HashTokenizer:
    def __init__(self, vocab_size: int = 250002, max_tokens: int = 512):
        self.vocab_size = vocab_size
        self.max_tokens = max_tokens

    def encode(self, text: str) -> List[int]:
        ids = [BOS]
        for w in _WORD_RE.findall(text.lower())[: self.max_tokens - 2]:
            h = 0
            for ch in w:
                h = (h * 1000003 + ord(ch)) & 0xFFFFFFFF
            ids.append(RESERVED + h % (self.vocab_size - RESERVED))
        ids.append(EOS)
        return ids

    def encode_batch(self, texts: Sequence[str]) -> Tuple[List[List[int]], List[List[int]]]:
        encoded = [self.encode(t) for t in texts]
        maxlen = max((len(e) for e in encoded), default=2)
        ids, mask = [], []
        for e in encoded:
            pad = maxlen - len(e)
            ids.append(e + [PAD] * pad)
            mask.append([1] * len(e) + [0] * pad)
        return ids, mask


class HFTokenizer:
    """Wraps a tokenizers.Tokenizer json file (real bge-m3 vocab)."""

    def __init__(self, path: str, max_tokens: int = 512):
        from tokenizers import Tokenizer
        self.tok = Tokenizer.from_file(path)
        self.max_tokens = max_tokens

    def encode(self, text: str) -> List[int]:
        return self.tok.encode(text).ids[: self.max_tokens]

    def encode_batch(self, texts: Sequence[str]):
        encs = self.tok.encode_batch(list(texts))
        encoded = [e.ids[: self.max_tokens] for e in encs]
        maxlen = max((len(e) for e in encoded), default=1)
        ids, mask = [], []
        for e in encoded:
            pad = maxlen - len(e)
            ids.append(e + [PAD] * pad)
            mask.append([1] * len(e) + [0] * pad)
        return ids, mask


DEFAULT_VOCAB = __file__.rsplit("/", 1)[0] + "/vocab/nornic_bpe.json"


def default_tokenizer(vocab_size: int = 250002, max_tokens: int = 512):
    """The embed queue's tokenizer: NORNICDB_TOKENIZER env path (e.g. a
    real bge-m3 tokenizer.json) > the shipped trained BPE artifact
    (scripts/train_tokenizer.py; XLM-R special-token layout, ids fit the
    bge-m3 embedding table) > hash fallback (no artifact present)."""
    import os
    path = os.environ.get("NORNICDB_TOKENIZER") or DEFAULT_VOCAB
    if os.path.exists(path):
        try:
            return HFTokenizer(path, max_tokens)
        except Exception:
            pass
    return HashTokenizer(vocab_size, max_tokens)
