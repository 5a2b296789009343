"""Checkpoint sharding: full state dict -> per-TP-rank shards.

The MI355X analog of the reference's ModelBuilder weight sharding + preshard
hooks (`models/application_base.py:240-265,389-399`, `gqa.py:679-954`):
each parallel parameter carries ``partition_dim`` metadata (set by the
layers in `parallel.layers`) and an optional ``preshard_fn`` that transforms
the FULL tensor (pad heads, replicate KV heads) so a plain slice along the
partition dim yields rank shards. Output files follow the reference layout:
``weights/tp{rank}_sharded_checkpoint.safetensors``.
"""

from __future__ import annotations

import os
from typing import Dict

import torch
import torch.nn as nn


def _param_meta(model: nn.Module) -> Dict[str, dict]:
    meta = {}
    for name, p in model.named_parameters():
        if getattr(p, "tensor_model_parallel", False):
            meta[name] = {
                "dim": p.partition_dim,
                "preshard_fn": getattr(p, "preshard_fn", None),
                # a param may shard over a SUB-group of the TP world (e.g.
                # the CP prefill projections over tp/cp contiguous ranks):
                # shard index = rank % num_partitions
                "num_partitions": getattr(p, "num_partitions", None),
            }
    return meta


def shard_state_dict_for_rank(
    full_sd: Dict[str, torch.Tensor], model: nn.Module, rank: int, tp_degree: int
) -> Dict[str, torch.Tensor]:
    """Slice the full (native-keyed) state dict for one TP rank."""
    meta = _param_meta(model)
    out: Dict[str, torch.Tensor] = {}
    for name, tensor in full_sd.items():
        m = meta.get(name)
        if m is None:
            out[name] = tensor
            continue
        t = tensor
        parts = m["num_partitions"] or tp_degree
        if m["preshard_fn"] is not None:
            t = m["preshard_fn"](t, parts)
        dim = m["dim"]
        size = t.shape[dim]
        if size % parts != 0:
            raise ValueError(f"{name}: dim {dim} size {size} not divisible by {parts}")
        shard = size // parts
        out[name] = t.narrow(dim, (rank % parts) * shard, shard).contiguous()
    return out


def save_sharded_checkpoints(
    full_sd: Dict[str, torch.Tensor], model: nn.Module, path: str, tp_degree: int
):
    """Write weights/tp{r}_sharded_checkpoint.safetensors for every rank."""
    from safetensors.torch import save_file

    wdir = os.path.join(path, "weights")
    os.makedirs(wdir, exist_ok=True)
    for r in range(tp_degree):
        sd = shard_state_dict_for_rank(full_sd, model, r, tp_degree)
        save_file(
            {k: v.contiguous() for k, v in sd.items()},
            os.path.join(wdir, f"tp{r}_sharded_checkpoint.safetensors"),
        )


def load_sharded_checkpoint(path: str, rank: int) -> Dict[str, torch.Tensor]:
    from safetensors.torch import load_file

    return load_file(os.path.join(path, "weights", f"tp{rank}_sharded_checkpoint.safetensors"))
