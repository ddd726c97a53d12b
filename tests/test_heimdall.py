"""Heimdall assistant tests: decoder KV-cache correctness, generation,
manager scheduling/metrics, Bifrost endpoints."""

import pytest
import torch

from nornicdb_amd.heimdall import ChatMessage, HeimdallManager, HeimdallPlugin
from nornicdb_amd.models.heimdall import HeimdallConfig, HeimdallModel


class TestDecoder:
    def test_kv_cache_matches_full_forward(self):
        torch.manual_seed(0)
        cfg = HeimdallConfig.tiny()
        m = HeimdallModel(cfg)
        for p in m.parameters():
            if p.dim() > 1:
                torch.nn.init.normal_(p, 0, 0.05)
        m.eval()
        ids = torch.randint(0, cfg.vocab_size, (1, 10))
        with torch.no_grad():
            full, _ = m(ids)
            # incremental: prefill 6, then 4 single steps
            caches = [(None, None)] * cfg.num_layers
            l1, caches = m(ids[:, :6], kv_caches=caches, pos0=0)
            outs = [l1]
            for t in range(6, 10):
                lt, caches = m(ids[:, t:t+1], kv_caches=caches, pos0=t)
                outs.append(lt)
            inc = torch.cat(outs, dim=1)
        assert torch.allclose(full, inc, atol=1e-4), (full - inc).abs().max()

    def test_generate_deterministic_greedy(self):
        torch.manual_seed(1)
        m = HeimdallModel(HeimdallConfig.tiny()).init_small().eval()
        ids = torch.randint(0, 512, (1, 5))
        a = m.generate(ids.clone(), max_new_tokens=8, temperature=0)
        b = m.generate(ids.clone(), max_new_tokens=8, temperature=0)
        assert a == b and len(a) == 8


class TestManager:
    def test_generate_and_stats(self):
        h = HeimdallManager(config=HeimdallConfig.tiny(), device="cpu")
        out = h.generate("hello heimdall", max_tokens=5)
        assert out.count("<") == 5
        assert h.stats["generations"] == 1
        assert h.stats["tokens_generated"] == 5
        assert h.tokens_per_second() > 0

    def test_stream(self):
        h = HeimdallManager(config=HeimdallConfig.tiny(), device="cpu")
        toks = list(h.generate_stream("stream me", max_tokens=4))
        assert len(toks) == 4

    def test_chat_includes_db_metrics(self):
        from nornicdb_amd.db import open_db
        from nornicdb_amd.embed import MockEmbedder
        mgr = open_db(embedder=MockEmbedder(8), dims=8)
        db = mgr.get()
        db.store("a memory")
        h = HeimdallManager(db, config=HeimdallConfig.tiny(), device="cpu")
        m = h.db_metrics()
        assert m["nodes"] == 1
        out = h.chat([ChatMessage("user", "how many nodes?")], max_tokens=3)
        assert out
        mgr.close()

    def test_plugins(self):
        class P(HeimdallPlugin):
            name = "test"
            started = False
            def start(self, mgr): self.started = True
            def metrics(self): return {"x": 1.0}

        h = HeimdallManager(config=HeimdallConfig.tiny(), device="cpu")
        p = P()
        h.register_plugin(p)
        assert p.started
        assert h.plugin_health()["test"]["status"] == "ok"


class TestBifrostHTTP:
    def test_endpoints(self):
        from fastapi.testclient import TestClient
        from nornicdb_amd.db import open_db
        from nornicdb_amd.embed import MockEmbedder
        from nornicdb_amd.server import create_app
        mgr = open_db(embedder=MockEmbedder(8), dims=8)
        with TestClient(create_app(mgr)) as c:
            r = c.post("/bifrost/generate", json={"prompt": "hi", "max_tokens": 3})
            assert r.status_code == 200 and r.json()["text"]
            r = c.post("/bifrost/chat", json={"messages": [
                {"role": "user", "content": "hello"}], "max_tokens": 2})
            assert r.status_code == 200
            r = c.get("/bifrost/metrics")
            assert "generation" in r.json()
            r = c.post("/bifrost/stream", json={"prompt": "s", "max_tokens": 2})
            assert "data:" in r.text and "[DONE]" in r.text
        mgr.close()


@pytest.mark.gpu
def test_graphed_decode_matches_eager():
    from nornicdb_amd.models.heimdall import GraphedDecoder
    torch.manual_seed(5)
    cfg = HeimdallConfig.tiny(max_position=128)
    m = HeimdallModel(cfg).init_small().cuda().eval()
    for p in m.parameters():
        if p.dim() > 1:
            torch.nn.init.normal_(p, 0, 0.05)
    ids = torch.randint(0, cfg.vocab_size, (1, 6), device="cuda")
    eager = m.generate(ids.clone(), max_new_tokens=10, temperature=0)
    gd = GraphedDecoder(m, max_len=64).capture()
    graphed = gd.generate(ids.clone(), max_new_tokens=10, temperature=0)
    assert eager == graphed, (eager, graphed)
