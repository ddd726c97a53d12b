"""Eval harness + CLI tests."""

import json
import subprocess
import sys

import pytest

from nornicdb_amd.search.eval import (EvalCase, EvalHarness, mrr, ndcg_at_k,
                                      precision_at_k, recall_at_k, diversity)


class TestMetrics:
    def test_precision_recall(self):
        got = ["a", "b", "c", "d"]
        rel = ["a", "c", "x"]
        assert precision_at_k(got, rel, 4) == 0.5
        assert recall_at_k(got, rel, 4) == pytest.approx(2 / 3)

    def test_mrr(self):
        assert mrr(["x", "a"], ["a"]) == 0.5
        assert mrr(["x", "y"], ["a"]) == 0.0

    def test_ndcg(self):
        assert ndcg_at_k(["a", "b"], ["a", "b"], 2) == pytest.approx(1.0)
        perfect = ndcg_at_k(["a", "x"], ["a"], 2)
        assert perfect == pytest.approx(1.0)
        worse = ndcg_at_k(["x", "a"], ["a"], 2)
        assert worse < 1.0

    def test_diversity(self):
        import numpy as np
        same = [np.array([1.0, 0]), np.array([1.0, 0])]
        assert diversity(same) == pytest.approx(0.0, abs=1e-6)
        orth = [np.array([1.0, 0]), np.array([0, 1.0])]
        assert diversity(orth) == pytest.approx(1.0, abs=1e-6)

    def test_harness(self):
        docs = {"q1": ["a", "b", "z"], "q2": ["z", "c"]}
        h = EvalHarness(lambda q, k: docs[q][:k])
        report = h.run([EvalCase("q1", ["a", "b"], k=3),
                        EvalCase("q2", ["c"], k=2)])
        assert report["cases"] == 2
        assert report["mrr"] == pytest.approx((1.0 + 0.5) / 2)


class TestCLI:
    def test_init_import_decay(self, tmp_path):
        # one subprocess for all three commands: each interpreter start
        # pays the torch import (~10 s), so chaining keeps the suite fast
        import os
        env = {**os.environ, "PYTHONPATH": "."}
        d = str(tmp_path / "data")
        f = tmp_path / "imp.json"
        f.write_text(json.dumps({
            "nodes": [{"id": "a", "labels": ["P"], "properties": {"x": 1}},
                      {"id": "b", "labels": ["P"], "properties": {}}],
            "relationships": [{"id": "e1", "type": "R", "start": "a", "end": "b"}],
        }))
        script = (
            "from nornicdb_amd.__main__ import main\n"
            f"main(['init', '--data-dir', {d!r}])\n"
            f"main(['import', '--data-dir', {d!r}, '--file', {str(f)!r}])\n"
            f"main(['decay', '--data-dir', {d!r}])\n")
        r = subprocess.run([sys.executable, "-c", script],
                           capture_output=True, text=True, env=env,
                           timeout=600)
        assert r.returncode == 0, r.stderr
        assert "initialized" in r.stdout
        assert "imported 2 nodes, 1 relationships" in r.stdout
        assert "scored" in r.stdout
