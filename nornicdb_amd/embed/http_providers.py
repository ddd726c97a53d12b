"""HTTP embedding providers: Ollama and OpenAI-compatible endpoints.

Parity: reference pkg/embed/embed.go OllamaEmbedder (:215) and
OpenAIEmbedder (:486). Kept for wire compatibility; this environment has
no egress, so these are exercised via injected transports in tests.
"""

from __future__ import annotations

import json
from typing import Callable, Optional, Sequence

import numpy as np

from . import Embedder


class HTTPEmbedder(Embedder):
    def __init__(self, provider: str, base_url: str = None, model: str = None,
                 api_key: str = None, dims: int = 1024,
                 transport: Optional[Callable] = None, timeout: float = 30.0):
        self.provider = provider
        self.dims = dims
        self.timeout = timeout
        self.api_key = api_key
        if provider == "ollama":
            self.base_url = base_url or "http://localhost:11434"
            self.model = model or "bge-m3"
        else:
            self.base_url = base_url or "https://api.openai.com"
            self.model = model or "text-embedding-3-small"
        self._transport = transport or self._default_transport

    def _default_transport(self, url: str, payload: dict, headers: dict) -> dict:
        import urllib.request
        req = urllib.request.Request(
            url, data=json.dumps(payload).encode(),
            headers={"Content-Type": "application/json", **headers})
        with urllib.request.urlopen(req, timeout=self.timeout) as resp:
            return json.loads(resp.read())

    def embed_batch(self, texts: Sequence[str]) -> np.ndarray:
        headers = {}
        if self.provider == "ollama":
            url = f"{self.base_url}/api/embed"
            data = self._transport(url, {"model": self.model, "input": list(texts)},
                                   headers)
            vecs = data["embeddings"]
        else:
            if self.api_key:
                headers["Authorization"] = f"Bearer {self.api_key}"
            url = f"{self.base_url}/v1/embeddings"
            data = self._transport(url, {"model": self.model, "input": list(texts)},
                                   headers)
            vecs = [d["embedding"] for d in data["data"]]
        out = np.asarray(vecs, dtype=np.float32)
        norms = np.linalg.norm(out, axis=1, keepdims=True)
        return out / np.clip(norms, 1e-12, None)
