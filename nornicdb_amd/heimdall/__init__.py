"""Heimdall: the embedded AI assistant.

Parity: reference pkg/heimdall — Manager with Generate/GenerateStream/Chat
(scheduler.go:178-238), HeimdallPlugin subsystem interface with lifecycle,
health and metrics (plugin.go:97-207), full-DB metrics aggregation for the
SLM (metrics.go:18), and the Bifrost HTTP+SSE bridge (mounted in
nornicdb_amd/server/http.py).
"""

from __future__ import annotations

import json
import queue
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, Iterator, List, Optional

from ..embed.tokenizer import HashTokenizer
from ..models.heimdall import HeimdallConfig, HeimdallModel


class HeimdallPlugin:
    """Subsystem plugin interface (reference plugin.go:97-207)."""

    name = "plugin"

    def start(self, manager: "HeimdallManager") -> None: ...
    def stop(self) -> None: ...
    def health(self) -> Dict[str, Any]:
        return {"status": "ok"}
    def metrics(self) -> Dict[str, float]:
        return {}


@dataclass
class ChatMessage:
    role: str
    content: str


class HeimdallManager:
    """Loads the decoder, serializes generation requests, aggregates
    database metrics into the system context."""

    def __init__(self, db=None, config: HeimdallConfig = None,
                 device: str = None, max_tokens_default: int = 64):
        import torch
        self.db = db
        self.device = device or ("cuda" if torch.cuda.is_available() else "cpu")
        cfg = config or (HeimdallConfig() if self.device.startswith("cuda")
                         else HeimdallConfig.tiny())
        dtype = torch.bfloat16 if self.device.startswith("cuda") else torch.float32
        self.model = HeimdallModel(cfg).init_small().to(self.device, dtype).eval()
        self.tokenizer = HashTokenizer(cfg.vocab_size, cfg.max_position)
        self.max_tokens_default = max_tokens_default
        self._lock = threading.Lock()  # one generation at a time (scheduler)
        self.plugins: List[HeimdallPlugin] = []
        self.stats = {"generations": 0, "tokens_generated": 0,
                      "total_latency_s": 0.0}
        self._torch = torch
        self._graphed = None
        if self.device.startswith("cuda"):
            # preference: fused cooperative decoder (one launch per token,
            # csrc/decode_fused.hip) > hipGraph replay > eager
            try:
                from ..models.heimdall import FusedDecoder
                self._graphed = FusedDecoder(self.model,
                                             max_len=min(cfg.max_position,
                                                         4096))
            except Exception:
                try:
                    from ..models.heimdall import GraphedDecoder
                    self._graphed = GraphedDecoder(
                        self.model, max_len=min(cfg.max_position,
                                                2048)).capture()
                except Exception:
                    self._graphed = None  # eager fallback

    # ---- plugins ----
    def register_plugin(self, plugin: HeimdallPlugin):
        self.plugins.append(plugin)
        plugin.start(self)

    def plugin_health(self) -> Dict[str, Any]:
        return {p.name: p.health() for p in self.plugins}

    # ---- metrics aggregation for the SLM (reference metrics.go) ----
    def db_metrics(self) -> Dict[str, Any]:
        if self.db is None:
            return {}
        eng = self.db.engine
        m = {"nodes": eng.node_count(), "relationships": eng.edge_count(),
             "pending_embeddings": len(eng.pending_embeddings()),
             "vector_index_size": len(self.db.search.emb),
             "fulltext_docs": len(self.db.search.fulltext)}
        for p in self.plugins:
            m.update({f"{p.name}.{k}": v for k, v in p.metrics().items()})
        return m

    def _decode_tokens(self, ids: List[int]) -> str:
        # random-init weights produce arbitrary ids; render deterministically
        return " ".join(f"<{i}>" for i in ids)

    # ---- generation API ----
    def generate(self, prompt: str, max_tokens: int = None,
                 temperature: float = 0.8) -> str:
        toks = list(self.tokenizer.encode(prompt))
        t = self._torch.as_tensor([toks], device=self.device)
        t0 = time.time()
        with self._lock:
            gen = self._graphed.generate if self._graphed is not None \
                else self.model.generate
            out = gen(t, max_new_tokens=max_tokens or self.max_tokens_default,
                      temperature=temperature)
        dt = time.time() - t0
        self.stats["generations"] += 1
        self.stats["tokens_generated"] += len(out)
        self.stats["total_latency_s"] += dt
        return self._decode_tokens(out)

    def generate_stream(self, prompt: str, max_tokens: int = None,
                        temperature: float = 0.8) -> Iterator[str]:
        toks = list(self.tokenizer.encode(prompt))
        t = self._torch.as_tensor([toks], device=self.device)
        q: "queue.Queue" = queue.Queue()
        DONE = object()

        def worker():
            with self._lock:
                gen = self._graphed.generate if self._graphed is not None \
                    else self.model.generate
                gen(t, max_new_tokens=max_tokens or self.max_tokens_default,
                    temperature=temperature, stream_cb=lambda tok: q.put(tok))
            q.put(DONE)

        th = threading.Thread(target=worker, daemon=True)
        th.start()
        n = 0
        while True:
            item = q.get()
            if item is DONE:
                break
            n += 1
            yield f"<{item}>"
        self.stats["generations"] += 1
        self.stats["tokens_generated"] += n

    def chat(self, messages: List[ChatMessage], max_tokens: int = None) -> str:
        """Chat with database context injected (reference scheduler Chat)."""
        ctx = json.dumps(self.db_metrics()) if self.db else "{}"
        prompt_parts = [f"[system] You are Heimdall, the NornicDB assistant. "
                        f"Database metrics: {ctx}"]
        for m in messages:
            prompt_parts.append(f"[{m.role}] {m.content}")
        prompt_parts.append("[assistant]")
        return self.generate("\n".join(prompt_parts), max_tokens=max_tokens)

    def tokens_per_second(self) -> float:
        t = self.stats["total_latency_s"]
        return self.stats["tokens_generated"] / t if t > 0 else 0.0
