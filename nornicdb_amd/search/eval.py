"""IR evaluation harness: Precision@K, Recall@K, MRR, NDCG, diversity.

Parity: reference pkg/eval/harness.go + cmd/eval (JSON test cases ->
metrics report).
"""

from __future__ import annotations

import json
import math
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Sequence


@dataclass
class EvalCase:
    query: str
    relevant: List[str]            # relevant doc ids (graded by position)
    k: int = 10


def precision_at_k(retrieved: Sequence[str], relevant: Sequence[str], k: int) -> float:
    if k == 0:
        return 0.0
    hits = sum(1 for r in retrieved[:k] if r in set(relevant))
    return hits / k


def recall_at_k(retrieved: Sequence[str], relevant: Sequence[str], k: int) -> float:
    if not relevant:
        return 0.0
    hits = sum(1 for r in retrieved[:k] if r in set(relevant))
    return hits / len(relevant)


def mrr(retrieved: Sequence[str], relevant: Sequence[str]) -> float:
    rel = set(relevant)
    for i, r in enumerate(retrieved, 1):
        if r in rel:
            return 1.0 / i
    return 0.0


def ndcg_at_k(retrieved: Sequence[str], relevant: Sequence[str], k: int) -> float:
    """Binary-graded NDCG (relevance 1 for listed ids)."""
    rel = set(relevant)
    dcg = sum(1.0 / math.log2(i + 1) for i, r in enumerate(retrieved[:k], 1)
              if r in rel)
    ideal = sum(1.0 / math.log2(i + 1)
                for i in range(1, min(len(relevant), k) + 1))
    return dcg / ideal if ideal > 0 else 0.0


def diversity(retrieved_vectors) -> float:
    """1 - mean pairwise cosine similarity of result vectors."""
    import numpy as np
    if len(retrieved_vectors) < 2:
        return 0.0
    v = np.asarray(retrieved_vectors, dtype=np.float32)
    v = v / np.clip(np.linalg.norm(v, axis=1, keepdims=True), 1e-12, None)
    sims = v @ v.T
    n = len(v)
    total = (sims.sum() - n) / (n * (n - 1))
    return float(1.0 - total)


class EvalHarness:
    def __init__(self, search_fn: Callable[[str, int], List[str]]):
        """search_fn(query, k) -> ranked doc ids."""
        self.search_fn = search_fn

    def run(self, cases: List[EvalCase]) -> Dict[str, float]:
        agg = {"precision@k": 0.0, "recall@k": 0.0, "mrr": 0.0, "ndcg@k": 0.0}
        per_case = []
        for c in cases:
            got = self.search_fn(c.query, c.k)
            m = {
                "precision@k": precision_at_k(got, c.relevant, c.k),
                "recall@k": recall_at_k(got, c.relevant, c.k),
                "mrr": mrr(got, c.relevant),
                "ndcg@k": ndcg_at_k(got, c.relevant, c.k),
            }
            per_case.append({"query": c.query, **m})
            for k in agg:
                agg[k] += m[k]
        n = max(len(cases), 1)
        report = {k: v / n for k, v in agg.items()}
        report["cases"] = len(cases)
        self.per_case = per_case
        return report

    @staticmethod
    def load_cases(path: str) -> List[EvalCase]:
        with open(path) as f:
            data = json.load(f)
        return [EvalCase(query=c["query"], relevant=c["relevant"],
                         k=c.get("k", 10)) for c in data]
