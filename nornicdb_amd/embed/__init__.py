"""Embedding providers.

Parity: reference pkg/embed (Embedder interface embed.go:71-84, Ollama
:215, OpenAI :486, local GGUF via llama.cpp, LRU CachedEmbedder, chunked
auto-embed). The local provider here is the from-scratch MI355X bge-m3
forward (nornicdb_amd.models.bge_m3 + HIP kernels) instead of llama.cpp;
HTTP providers (Ollama/OpenAI-compatible) keep wire parity but are
network-gated in this environment.
"""

from __future__ import annotations

import hashlib
import json
import threading
from collections import OrderedDict
from typing import List, Optional, Sequence

import numpy as np


class Embedder:
    """reference pkg/embed/embed.go:71-84."""

    dims: int = 1024

    def embed(self, text: str) -> np.ndarray:
        return self.embed_batch([text])[0]

    def embed_batch(self, texts: Sequence[str]) -> np.ndarray:
        raise NotImplementedError

    def embed_query(self, text: str) -> np.ndarray:
        return self.embed(text)


class MockEmbedder(Embedder):
    """Deterministic hash-based embedder for tests (no model load)."""

    def __init__(self, dims: int = 64):
        self.dims = dims

    def embed_batch(self, texts: Sequence[str]) -> np.ndarray:
        out = np.zeros((len(texts), self.dims), np.float32)
        for i, t in enumerate(texts):
            seed = int.from_bytes(hashlib.sha256(t.encode()).digest()[:8], "little")
            rng = np.random.default_rng(seed)
            v = rng.standard_normal(self.dims).astype(np.float32)
            out[i] = v / np.linalg.norm(v)
        return out


class BgeM3Embedder(Embedder):
    """bge-m3 (XLM-R-large) forward on MI355X via nornicdb_amd.models.

    Replaces reference pkg/localllm llama.cpp embedding path
    (llama.go:104-180: tokenize -> encode -> mean pool -> L2 norm).
    Weights are random-init unless a state_dict path is given (no network
    in this environment; shape/throughput parity is the target).
    """

    def __init__(self, dims: int = 1024, device: Optional[str] = None,
                 max_tokens: int = 512, weights_path: Optional[str] = None,
                 layers: int = 24):
        import torch
        from ..models import BgeM3Config, BgeM3Encoder
        from .tokenizer import default_tokenizer

        self.dims = dims
        self.max_tokens = max_tokens
        self.device = device or ("cuda" if torch.cuda.is_available() else "cpu")
        cfg = BgeM3Config(num_layers=layers)
        dtype = torch.bfloat16 if self.device.startswith("cuda") else torch.float32
        self.model = BgeM3Encoder(cfg).init_small().to(self.device, dtype).eval()
        if weights_path:
            sd = torch.load(weights_path, map_location=self.device)
            self.model.load_state_dict(sd)
        self.tokenizer = default_tokenizer(cfg.vocab_size, max_tokens)
        self._torch = torch

    def embed_batch(self, texts: Sequence[str]) -> np.ndarray:
        torch = self._torch
        ids, mask = self.tokenizer.encode_batch(texts)
        ids = torch.as_tensor(ids, device=self.device)
        mask = torch.as_tensor(mask, device=self.device)
        with torch.no_grad():
            out = self.model(ids, mask)
        return out.float().cpu().numpy()


class CachedEmbedder(Embedder):
    """LRU cache wrapper (reference pkg/embed/cached_embedder.go)."""

    def __init__(self, inner: Embedder, capacity: int = 10000):
        self.inner = inner
        self.dims = inner.dims
        self.capacity = capacity
        self._lock = threading.Lock()
        self._cache: OrderedDict[str, np.ndarray] = OrderedDict()
        self.hits = 0
        self.misses = 0

    def embed_batch(self, texts: Sequence[str]) -> np.ndarray:
        out: List[Optional[np.ndarray]] = [None] * len(texts)
        missing, missing_idx = [], []
        with self._lock:
            for i, t in enumerate(texts):
                v = self._cache.get(t)
                if v is not None:
                    self._cache.move_to_end(t)
                    out[i] = v
                    self.hits += 1
                else:
                    missing.append(t)
                    missing_idx.append(i)
                    self.misses += 1
        if missing:
            vecs = self.inner.embed_batch(missing)
            with self._lock:
                for t, v, i in zip(missing, vecs, missing_idx):
                    self._cache[t] = v
                    out[i] = v
                while len(self._cache) > self.capacity:
                    self._cache.popitem(last=False)
        return np.stack(out)


def chunk_text(text: str, chunk_tokens: int = 512, overlap: int = 50) -> List[str]:
    """Word-boundary chunking (reference pkg/nornicdb/db.go:981-982:
    512-token chunks, 50 overlap)."""
    words = text.split()
    if len(words) <= chunk_tokens:
        return [text] if text else []
    chunks = []
    step = max(1, chunk_tokens - overlap)
    for s in range(0, len(words), step):
        chunks.append(" ".join(words[s:s + chunk_tokens]))
        if s + chunk_tokens >= len(words):
            break
    return chunks


def create_embedder(provider: str = "mock", **kw) -> Embedder:
    """Factory (reference pkg/embed/embed.go:816)."""
    if provider in ("mock", "test"):
        return MockEmbedder(**kw)
    if provider in ("local", "bge-m3", "native"):
        return BgeM3Embedder(**kw)
    if provider in ("ollama", "openai"):
        from .http_providers import HTTPEmbedder
        return HTTPEmbedder(provider, **kw)
    raise ValueError(f"unknown embedder provider {provider}")
