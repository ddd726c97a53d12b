"""LoRA fine-tuning loop for Heimdall.

Parity: reference neural/training/trainer.py (LoRA/QLoRA trainer with
warmup + cosine schedule, gradient accumulation, checkpointing).
MI355X-native: bf16 compute on ROCm, fused via PyTorch kernels;
multi-GPU via torch.distributed DDP over RCCL (gloo on CPU for tests).
"""

from __future__ import annotations

import json
import math
import os
import time
from dataclasses import dataclass, field
from typing import List, Optional

import torch

from ..embed.tokenizer import HashTokenizer
from ..models.heimdall import HeimdallConfig, HeimdallModel
from .data import InstructionDataset
from .lora import inject_lora, lora_parameters, lora_state_dict, merge_lora


@dataclass
class TrainConfig:
    lora_r: int = 8
    lora_alpha: float = 16.0
    lora_dropout: float = 0.0
    lr: float = 2e-4
    weight_decay: float = 0.01
    batch_size: int = 4
    grad_accum: int = 1
    epochs: int = 1
    max_steps: int = 0          # 0 = until data exhausted
    warmup_ratio: float = 0.05
    max_grad_norm: float = 1.0
    seed: int = 0
    log_every: int = 10


class LoRATrainer:
    def __init__(self, model: HeimdallModel, cfg: TrainConfig = None,
                 device: str = None):
        self.cfg = cfg or TrainConfig()
        self.device = device or (
            "cuda" if torch.cuda.is_available() else "cpu")
        self.model = model.to(self.device)
        if self.device.startswith("cuda"):
            self.model = self.model.to(torch.bfloat16)
        self.adapted = inject_lora(self.model, self.cfg.lora_r,
                                   self.cfg.lora_alpha, self.cfg.lora_dropout)
        self.opt = torch.optim.AdamW(lora_parameters(self.model),
                                     lr=self.cfg.lr,
                                     weight_decay=self.cfg.weight_decay)
        self.step = 0
        self.history: List[dict] = []
        self._ddp = None
        if torch.distributed.is_available() and \
                torch.distributed.is_initialized():
            self._ddp = torch.nn.parallel.DistributedDataParallel(
                self.model)

    def _lr_at(self, step, total):
        warm = max(1, int(total * self.cfg.warmup_ratio))
        if step < warm:
            return self.cfg.lr * (step + 1) / warm
        t = (step - warm) / max(1, total - warm)
        return self.cfg.lr * 0.5 * (1 + math.cos(math.pi * t))

    def _loss(self, toks, labels):
        net = self._ddp or self.model
        logits, _ = net(toks)
        return torch.nn.functional.cross_entropy(
            logits[:, :-1].reshape(-1, logits.shape[-1]).float(),
            labels[:, 1:].reshape(-1), ignore_index=-100)

    def train(self, dataset: InstructionDataset) -> List[dict]:
        c = self.cfg
        steps_per_epoch = max(1, math.ceil(len(dataset) / c.batch_size))
        total = c.max_steps or steps_per_epoch * c.epochs
        self.model.train()
        accum = 0
        t0 = time.time()
        for epoch in range(c.epochs):
            for toks, labels in dataset.batches(c.batch_size, seed=c.seed + epoch,
                                                device=self.device):
                if self.step >= total:
                    break
                lr = self._lr_at(self.step, total)
                for g in self.opt.param_groups:
                    g["lr"] = lr
                loss = self._loss(toks, labels) / c.grad_accum
                loss.backward()
                accum += 1
                if accum >= c.grad_accum:
                    torch.nn.utils.clip_grad_norm_(
                        lora_parameters(self.model), c.max_grad_norm)
                    self.opt.step()
                    self.opt.zero_grad(set_to_none=True)
                    accum = 0
                self.step += 1
                if self.step % c.log_every == 0 or self.step == total:
                    self.history.append({
                        "step": self.step, "loss": float(loss) * c.grad_accum,
                        "lr": lr, "elapsed_s": time.time() - t0})
            if self.step >= total:
                break
        return self.history

    # ---- checkpointing ----
    def save_adapter(self, path: str):
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        torch.save({"lora": lora_state_dict(self.model),
                    "step": self.step,
                    "config": self.cfg.__dict__,
                    "adapted": self.adapted}, path)

    def load_adapter(self, path: str):
        ckpt = torch.load(path, map_location=self.device, weights_only=False)
        missing, unexpected = self.model.load_state_dict(
            ckpt["lora"], strict=False)
        self.step = ckpt.get("step", 0)
        return ckpt

    def merge(self) -> HeimdallModel:
        """Fold adapters into the base model (in place) and return it."""
        merge_lora(self.model)
        return self.model
