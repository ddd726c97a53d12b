"""BM25 full-text index.

Parity: reference pkg/search/fulltext_index.go (:20,:125,:205; tokenizer +
stopwords :249-287). k1=1.2, b=0.75.
"""

from __future__ import annotations

import math
import re
import threading
from collections import Counter, defaultdict
from typing import Dict, List, Tuple

_TOKEN_RE = re.compile(r"[a-z0-9]+")

STOPWORDS = frozenset("""
a an and are as at be but by for from has have he her his i in is it its my
not of on or she that the their them they this to was we were will with you
your
""".split())


def tokenize(text: str) -> List[str]:
    return [t for t in _TOKEN_RE.findall(text.lower())
            if t not in STOPWORDS and len(t) > 1]


class FulltextIndex:
    def __init__(self, k1: float = 1.2, b: float = 0.75):
        self.k1 = k1
        self.b = b
        self._lock = threading.RLock()
        self._postings: Dict[str, Dict[str, int]] = defaultdict(dict)  # term -> doc -> tf
        self._doc_len: Dict[str, int] = {}
        self._total_len = 0

    def __len__(self):
        return len(self._doc_len)

    def index(self, doc_id: str, text: str) -> None:
        with self._lock:
            self.remove(doc_id)
            toks = tokenize(text or "")
            if not toks:
                return
            tf = Counter(toks)
            for term, c in tf.items():
                self._postings[term][doc_id] = c
            self._doc_len[doc_id] = len(toks)
            self._total_len += len(toks)

    def remove(self, doc_id: str) -> None:
        with self._lock:
            old = self._doc_len.pop(doc_id, None)
            if old is None:
                return
            self._total_len -= old
            for term in list(self._postings):
                self._postings[term].pop(doc_id, None)
                if not self._postings[term]:
                    del self._postings[term]

    def search(self, query: str, k: int = 10) -> List[Tuple[str, float]]:
        with self._lock:
            n_docs = len(self._doc_len)
            if n_docs == 0:
                return []
            avg_len = self._total_len / n_docs
            scores: Dict[str, float] = defaultdict(float)
            for term in tokenize(query):
                posting = self._postings.get(term)
                if not posting:
                    continue
                df = len(posting)
                idf = math.log(1 + (n_docs - df + 0.5) / (df + 0.5))
                for doc, tf in posting.items():
                    dl = self._doc_len[doc]
                    s = idf * (tf * (self.k1 + 1)) / (
                        tf + self.k1 * (1 - self.b + self.b * dl / avg_len))
                    scores[doc] += s
            best = sorted(scores.items(), key=lambda kv: -kv[1])
            return best[:k]
