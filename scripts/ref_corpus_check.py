"""Run the Cypher queries extracted from the REFERENCE's own test suite
against this engine (parity smoke). Extraction: backtick/quoted strings
starting with a clause keyword from /root/reference/pkg/cypher/*_test.go.

Usage: python scripts/ref_corpus_check.py [queries.txt]
"""
import re
import sys

sys.path.insert(0, __import__('os').path.dirname(
    __import__('os').path.dirname(__import__('os').path.abspath(__file__))))

from nornicdb_amd.db import DatabaseManager
from nornicdb_amd.storage.memory import MemoryEngine


def main(path=None):
    import os
    if path is None:
        path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                            "ref_queries.txt")
    qs = [l for l in open(path).read().splitlines() if l.strip()]
    # extraction kept Go escape sequences literally; restore the real
    # whitespace the reference executed (whitespaceVariations tests)
    qs = [q.replace("\\n", "\n").replace("\\t", "\t").replace("\\r", "\r")
            .replace("\\\\", "\\")
          if "\\" in q else q for q in qs]
    db = DatabaseManager(MemoryEngine()).get()
    ok = pf = rf = 0
    for q in qs:
        params = {p: 1 for p in set(re.findall(r"\$(\w+)", q))}
        for p in list(params):
            lp = p.lower()
            if any(k in lp for k in ("name", "text", "id", "query", "label",
                                     "type", "content", "title", "key",
                                     "prop")):
                params[p] = "x"
            if "list" in lp or "ids" in lp or "tags" in lp:
                params[p] = []
            if "props" in lp or "map" in lp or "properties" in lp:
                params[p] = {}
            if "vector" in lp or "embedding" in lp:
                params[p] = [0.0] * 4
        try:
            db.cypher(q, params)
            ok += 1
        except Exception as e:
            if "Syntax" in type(e).__name__:
                pf += 1
            else:
                rf += 1
    print(f"{ok}/{len(qs)} clean ({pf} parse, {rf} runtime)")


if __name__ == "__main__":
    main(*sys.argv[1:])
