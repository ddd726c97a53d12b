"""LoRA adapters for the Heimdall decoder.

Parity: reference neural/ (LoRA/QLoRA fine-tuning of Heimdall SLMs,
neural/train.py + neural/training/trainer.py). MI355X-native design:
plain PyTorch-ROCm modules in bf16 — adapters wrap the decoder's
nn.Linear projections, train only the low-rank factors, and merge back
into the base weights for zero-overhead inference (the reference
exports GGUF for a llama.cpp runtime; our runtime is HeimdallModel +
hipGraph decode, so the export target is merged safetensors —
see export.py).
"""

from __future__ import annotations

import math
from typing import Dict, Iterable, List

import torch
import torch.nn as nn

# decoder projections eligible for adaptation (models/heimdall.py:58-64)
DEFAULT_TARGETS = ("q_proj", "k_proj", "v_proj", "o_proj",
                   "gate_proj", "up_proj", "down_proj")


class LoRALinear(nn.Module):
    """y = W x + (alpha/r) * B(A x); W frozen, A/B trained."""

    def __init__(self, base: nn.Linear, r: int = 8, alpha: float = 16.0,
                 dropout: float = 0.0):
        super().__init__()
        self.base = base
        for p in self.base.parameters():
            p.requires_grad_(False)
        self.r = r
        self.scaling = alpha / r
        dtype = base.weight.dtype
        dev = base.weight.device
        self.lora_A = nn.Parameter(
            torch.empty(r, base.in_features, dtype=dtype, device=dev))
        self.lora_B = nn.Parameter(
            torch.zeros(base.out_features, r, dtype=dtype, device=dev))
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
        self.dropout = nn.Dropout(dropout) if dropout > 0 else nn.Identity()

    def forward(self, x):
        y = self.base(x)
        return y + self.dropout(x) @ self.lora_A.T @ self.lora_B.T * self.scaling

    def merged_weight(self) -> torch.Tensor:
        return self.base.weight + (self.lora_B @ self.lora_A) * self.scaling


def inject_lora(model: nn.Module, r: int = 8, alpha: float = 16.0,
                dropout: float = 0.0,
                targets: Iterable[str] = DEFAULT_TARGETS) -> List[str]:
    """Replace target nn.Linear submodules with LoRALinear wrappers.
    Returns the list of adapted module paths. All non-LoRA parameters
    are frozen."""
    targets = set(targets)
    adapted = []
    for p in model.parameters():
        p.requires_grad_(False)
    for name, module in model.named_modules():
        leaf = name.rsplit(".", 1)[-1]
        if leaf in targets and isinstance(module, nn.Linear):
            parent = model.get_submodule(name.rsplit(".", 1)[0]) \
                if "." in name else model
            setattr(parent, leaf, LoRALinear(module, r, alpha, dropout))
            adapted.append(name)
    return adapted


def merge_lora(model: nn.Module) -> int:
    """Fold every LoRALinear back into a plain nn.Linear (in place).
    Returns the number of merged modules."""
    merged = 0
    for name, module in list(model.named_modules()):
        if isinstance(module, LoRALinear):
            base = module.base
            with torch.no_grad():
                base.weight.copy_(module.merged_weight())
            parent = model.get_submodule(name.rsplit(".", 1)[0]) \
                if "." in name else model
            setattr(parent, name.rsplit(".", 1)[-1], base)
            merged += 1
    return merged


def lora_parameters(model: nn.Module):
    return [p for p in model.parameters() if p.requires_grad]


def lora_state_dict(model: nn.Module) -> Dict[str, torch.Tensor]:
    return {k: v for k, v in model.state_dict().items()
            if "lora_A" in k or "lora_B" in k}
