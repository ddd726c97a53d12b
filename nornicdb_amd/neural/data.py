"""Training datasets for Heimdall fine-tuning.

Parity: reference neural/scripts dataset generators (training pairs from
database content) + JSONL instruction data loading (neural/train.py).
"""

from __future__ import annotations

import json
import random
from typing import Iterator, List, Optional, Tuple

import torch

from ..embed.tokenizer import HashTokenizer


class InstructionDataset:
    """JSONL records {"prompt": ..., "completion": ...} tokenized with the
    Heimdall tokenizer. Loss is masked over the prompt (completion-only)."""

    def __init__(self, records: List[dict], tokenizer: HashTokenizer,
                 max_len: int = 512):
        self.records = records
        self.tok = tokenizer
        self.max_len = max_len

    @classmethod
    def from_jsonl(cls, path: str, tokenizer: HashTokenizer,
                   max_len: int = 512) -> "InstructionDataset":
        recs = []
        with open(path) as f:
            for line in f:
                if line.strip():
                    recs.append(json.loads(line))
        return cls(recs, tokenizer, max_len)

    def __len__(self):
        return len(self.records)

    def encode(self, rec: dict) -> Tuple[List[int], int]:
        prompt = list(self.tok.encode(f"[user] {rec['prompt']}\n[assistant] "))
        completion = list(self.tok.encode(str(rec["completion"])))
        ids = (prompt + completion)[: self.max_len]
        return ids, min(len(prompt), len(ids))

    def batches(self, batch_size: int, shuffle: bool = True,
                seed: int = 0, device: str = "cpu"
                ) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        """Yields (token_ids [B, S], labels [B, S]) with -100 on prompt
        positions and padding."""
        order = list(range(len(self.records)))
        if shuffle:
            random.Random(seed).shuffle(order)
        for i in range(0, len(order), batch_size):
            chunk = [self.encode(self.records[j])
                     for j in order[i:i + batch_size]]
            s = max(len(ids) for ids, _ in chunk)
            toks = torch.zeros(len(chunk), s, dtype=torch.long)
            labels = torch.full((len(chunk), s), -100, dtype=torch.long)
            for b, (ids, plen) in enumerate(chunk):
                toks[b, :len(ids)] = torch.as_tensor(ids)
                if len(ids) > plen:
                    labels[b, plen:len(ids)] = torch.as_tensor(ids[plen:])
            yield toks.to(device), labels.to(device)


def generate_dataset_from_db(db, limit: int = 1000) -> List[dict]:
    """Build Q/A training pairs from stored graph content (reference
    neural/scripts generators: node recall + relationship questions)."""
    recs = []
    eng = db.engine
    for n in eng.all_nodes():
        content = n.properties.get("content") or n.properties.get("name")
        if not content:
            continue
        title = n.properties.get("title") or n.properties.get("name") or n.id
        recs.append({"prompt": f"What do you know about {title}?",
                     "completion": str(content)})
        if len(recs) >= limit:
            return recs
    for e in eng.all_edges():
        try:
            a = eng.get_node(e.start_node)
            b = eng.get_node(e.end_node)
        except Exception:
            continue
        an = a.properties.get("name") or a.id
        bn = b.properties.get("name") or b.id
        recs.append({"prompt": f"How is {an} related to {bn}?",
                     "completion": f"{an} {e.type.lower().replace('_', ' ')} {bn}."})
        if len(recs) >= limit:
            break
    return recs
