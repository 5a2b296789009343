"""Checkpoint IO: load HF checkpoint dirs, save sharded safetensors.

Analog of the reference's `modules/checkpoint.py` (:1-364 —
load_state_dict / save_state_dict_safetensors / prune_state_dict used at
`application_base.py:30-34`).
"""

from __future__ import annotations

import json
import os
from typing import Dict, Optional

import torch

try:
    from safetensors.torch import load_file as st_load_file
    from safetensors.torch import save_file as st_save_file
except ImportError:  # pragma: no cover
    st_load_file = st_save_file = None


def load_state_dict(model_path: str) -> Dict[str, torch.Tensor]:
    """Load a HF model dir: safetensors (indexed or single), .bin, or .pt."""
    index = os.path.join(model_path, "model.safetensors.index.json")
    if os.path.exists(index):
        with open(index) as f:
            idx = json.load(f)
        shards = sorted(set(idx["weight_map"].values()))
        sd: Dict[str, torch.Tensor] = {}
        for s in shards:
            sd.update(st_load_file(os.path.join(model_path, s)))
        return sd
    single = os.path.join(model_path, "model.safetensors")
    if os.path.exists(single):
        return st_load_file(single)
    bin_index = os.path.join(model_path, "pytorch_model.bin.index.json")
    if os.path.exists(bin_index):
        with open(bin_index) as f:
            idx = json.load(f)
        sd = {}
        for s in sorted(set(idx["weight_map"].values())):
            sd.update(torch.load(os.path.join(model_path, s), map_location="cpu", weights_only=True))
        return sd
    for name in ("pytorch_model.bin", "model.pt", "checkpoint.pt"):
        p = os.path.join(model_path, name)
        if os.path.exists(p):
            return torch.load(p, map_location="cpu", weights_only=True)
    raise FileNotFoundError(f"no checkpoint found under {model_path}")


def save_state_dict_safetensors(
    sd: Dict[str, torch.Tensor], path: str, filename: str = "model.safetensors"
):
    os.makedirs(path, exist_ok=True)
    st_save_file({k: v.contiguous() for k, v in sd.items()}, os.path.join(path, filename))


def prune_state_dict(sd: Dict[str, torch.Tensor], keep_prefixes=None) -> Dict[str, torch.Tensor]:
    if not keep_prefixes:
        return {k: v for k, v in sd.items() if v is not None}
    return {k: v for k, v in sd.items() if any(k.startswith(p) for p in keep_prefixes)}
