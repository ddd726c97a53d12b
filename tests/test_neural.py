"""Neural training subsystem: LoRA adapters, trainer, dataset generation,
merged export (reference neural/ train.py + export_to_gguf.py +
training/trainer.py — SURVEY.md §2)."""

import json

import pytest
import torch

from nornicdb_amd.embed.tokenizer import HashTokenizer
from nornicdb_amd.models.heimdall import HeimdallConfig, HeimdallModel
from nornicdb_amd.neural import (InstructionDataset, LoRATrainer, TrainConfig,
                                 export_merged, generate_dataset_from_db,
                                 inject_lora, load_merged, merge_lora)


def tiny_model():
    return HeimdallModel(HeimdallConfig.tiny()).init_small()


def tiny_dataset(tok, n=16):
    recs = [{"prompt": f"question {i}", "completion": f"answer {i % 4}"}
            for i in range(n)]
    return InstructionDataset(recs, tok, max_len=32)


class TestLoRA:
    def test_inject_freezes_base(self):
        m = tiny_model()
        adapted = inject_lora(m, r=4)
        assert len(adapted) == 7 * len(m.layers)  # 7 projections per layer
        trainable = [n for n, p in m.named_parameters() if p.requires_grad]
        assert trainable and all("lora_" in n for n in trainable)

    def test_zero_init_preserves_output(self):
        torch.manual_seed(0)
        m = tiny_model().eval()
        x = torch.randint(0, 100, (1, 8))
        with torch.no_grad():
            before, _ = m(x)
        inject_lora(m, r=4)
        with torch.no_grad():
            after, _ = m(x)
        # lora_B starts at zero -> identical function
        assert torch.allclose(before, after, atol=1e-6)

    def test_merge_equivalence(self):
        torch.manual_seed(0)
        m = tiny_model().eval()
        inject_lora(m, r=4)
        # perturb adapters so the merge is non-trivial
        for n, p in m.named_parameters():
            if "lora_B" in n:
                torch.nn.init.normal_(p, std=0.02)
        x = torch.randint(0, 100, (1, 8))
        with torch.no_grad():
            lora_out, _ = m(x)
        n_merged = merge_lora(m)
        assert n_merged == 7 * len(m.layers)
        with torch.no_grad():
            merged_out, _ = m(x)
        assert torch.allclose(lora_out, merged_out, atol=1e-4)


class TestTrainer:
    def test_loss_decreases(self):
        torch.manual_seed(0)
        m = tiny_model()
        tok = HashTokenizer(m.cfg.vocab_size, m.cfg.max_position)
        ds = tiny_dataset(tok, 16)
        tr = LoRATrainer(m, TrainConfig(batch_size=4, epochs=30, lr=5e-3,
                                        log_every=1), device="cpu")
        hist = tr.train(ds)
        assert len(hist) > 10
        first = sum(h["loss"] for h in hist[:5]) / 5
        last = sum(h["loss"] for h in hist[-5:]) / 5
        assert last < first  # memorizes the tiny set

    def test_checkpoint_roundtrip(self, tmp_path):
        torch.manual_seed(0)
        m = tiny_model()
        tok = HashTokenizer(m.cfg.vocab_size, m.cfg.max_position)
        tr = LoRATrainer(m, TrainConfig(batch_size=4, epochs=1), device="cpu")
        tr.train(tiny_dataset(tok, 8))
        p = str(tmp_path / "adapter.pt")
        tr.save_adapter(p)
        m2 = tiny_model()
        tr2 = LoRATrainer(m2, TrainConfig(), device="cpu")
        ckpt = tr2.load_adapter(p)
        assert ckpt["step"] == tr.step
        sd1 = {k: v for k, v in m.state_dict().items() if "lora" in k}
        sd2 = {k: v for k, v in m2.state_dict().items() if "lora" in k}
        for k in sd1:
            assert torch.allclose(sd1[k], sd2[k])


class TestExport:
    def test_export_load_roundtrip(self, tmp_path):
        torch.manual_seed(0)
        m = tiny_model().eval()
        inject_lora(m, r=4)
        for n, p in m.named_parameters():
            if "lora_B" in n:
                torch.nn.init.normal_(p, std=0.02)
        merge_lora(m)
        export_merged(m, str(tmp_path))
        m2 = load_merged(str(tmp_path)).eval()
        x = torch.randint(0, 100, (1, 8))
        with torch.no_grad():
            a, _ = m(x)
            b, _ = m2(x)
        assert torch.allclose(a, b, atol=1e-5)


class TestDatasetGeneration:
    def test_from_db(self):
        from nornicdb_amd.db import NornicDB
        from nornicdb_amd.storage.memory import MemoryEngine
        db = NornicDB(MemoryEngine(), auto_embed=False)
        db.cypher("CREATE (a:Doc {name: 'graphs', content: 'Graphs store "
                  "nodes and edges.'})-[:MENTIONS]->"
                  "(b:Doc {name: 'vectors', content: 'Vectors embed text.'})")
        recs = generate_dataset_from_db(db)
        assert any("graphs" in r["prompt"] for r in recs)
        assert any("mentions" in r["completion"] for r in recs)
        tok = HashTokenizer(1000, 128)
        ds = InstructionDataset(recs, tok, max_len=64)
        toks, labels = next(ds.batches(2, device="cpu"))
        assert toks.shape == labels.shape
        assert (labels == -100).any()  # prompt masked


@pytest.mark.gpu
class TestNeuralGPU:
    def test_lora_train_step_bf16(self):
        torch.manual_seed(0)
        m = HeimdallModel(HeimdallConfig(num_layers=4)).init_small()
        tok = HashTokenizer(m.cfg.vocab_size, m.cfg.max_position)
        tr = LoRATrainer(m, TrainConfig(batch_size=2, max_steps=4,
                                        log_every=1), device="cuda")
        hist = tr.train(tiny_dataset(tok, 8))
        assert len(hist) == 4
        assert all(h["loss"] == h["loss"] for h in hist)  # no NaNs
        merged = tr.merge()
        out = merged.generate(
            torch.randint(0, 1000, (1, 4), device="cuda"), max_new_tokens=4)
        assert len(out) == 4
