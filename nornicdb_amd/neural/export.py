"""Model export/load for trained Heimdall models.

Parity: reference neural/export_to_gguf.py. Design decision: the
reference exports GGUF because its runtime is llama.cpp; OUR decode
runtime is HeimdallModel + hipGraph (models/heimdall.py), so the export
target is merged-weight safetensors plus a JSON config — loadable
straight into HeimdallManager without a format shim.
"""

from __future__ import annotations

import json
import os
from typing import Tuple

import torch

from ..models.heimdall import HeimdallConfig, HeimdallModel


def export_merged(model: HeimdallModel, out_dir: str) -> Tuple[str, str]:
    """Write model.safetensors + config.json; returns both paths."""
    from safetensors.torch import save_file
    os.makedirs(out_dir, exist_ok=True)
    wpath = os.path.join(out_dir, "model.safetensors")
    cpath = os.path.join(out_dir, "config.json")
    state = {k: v.contiguous() for k, v in model.state_dict().items()
             if k != "lm_head.weight"}  # tied to embed.weight
    save_file(state, wpath)
    with open(cpath, "w") as f:
        json.dump(model.cfg.__dict__, f, indent=2)
    return wpath, cpath


def load_merged(out_dir: str, device: str = "cpu") -> HeimdallModel:
    from safetensors.torch import load_file
    with open(os.path.join(out_dir, "config.json")) as f:
        cfg = HeimdallConfig(**json.load(f))
    model = HeimdallModel(cfg)
    state = load_file(os.path.join(out_dir, "model.safetensors"))
    model.load_state_dict(state, strict=False)
    model.lm_head.weight = model.embed.weight  # retie
    return model.to(device)
