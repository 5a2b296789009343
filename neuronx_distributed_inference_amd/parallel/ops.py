"""Distributed reductions over vocab-sharded logits.

Analogs of `nxd_argmax` / `nxd_topk` (ref K9/K10, used from
`modules/generation/sampling.py:306-384` and `models/model_base.py:1166-1172`):
greedy argmax and top-k over logits whose vocab dim is sharded across the TP
group. Local reduce first, then one small all-gather of (value, global-index)
payloads — the message is O(B*k), never O(V).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from neuronx_distributed_inference_amd.parallel import state as ps


def _group_info(group):
    group = group if group is not None else ps.get_tensor_model_parallel_group()
    if group is None or not dist.is_initialized():
        return None, 1, 0
    return group, dist.get_world_size(group=group), dist.get_rank(group=group)


def distributed_argmax(logits: torch.Tensor, dim: int = -1, group=None) -> torch.Tensor:
    """Greedy argmax over the full (sharded) vocab; returns global indices."""
    group, world, rank = _group_info(group)
    local_max, local_idx = logits.max(dim=dim)
    if world == 1:
        return local_idx
    shard = logits.size(dim)
    local_idx = local_idx + rank * shard
    # payload: stack values and indices, all-gather, reduce on every rank
    vals = [torch.empty_like(local_max) for _ in range(world)]
    idxs = [torch.empty_like(local_idx) for _ in range(world)]
    dist.all_gather(vals, local_max.contiguous(), group=group)
    dist.all_gather(idxs, local_idx.contiguous(), group=group)
    vals = torch.stack(vals, dim=0)   # (world, ...)
    idxs = torch.stack(idxs, dim=0)
    best = vals.argmax(dim=0, keepdim=True)
    return idxs.gather(0, best).squeeze(0)


def distributed_topk(logits: torch.Tensor, k: int, dim: int = -1, group=None):
    """Top-k (values, global indices) over the sharded vocab dim."""
    group, world, rank = _group_info(group)
    k_local = min(k, logits.size(dim))
    vals, idxs = logits.topk(k_local, dim=dim)
    if world == 1:
        return vals, idxs
    shard = logits.size(dim)
    idxs = idxs + rank * shard
    gv = [torch.empty_like(vals) for _ in range(world)]
    gi = [torch.empty_like(idxs) for _ in range(world)]
    dist.all_gather(gv, vals.contiguous(), group=group)
    dist.all_gather(gi, idxs.contiguous(), group=group)
    all_vals = torch.cat(gv, dim=dim)
    all_idxs = torch.cat(gi, dim=dim)
    top_vals, pos = all_vals.topk(min(k, all_vals.size(dim)), dim=dim)
    top_idxs = all_idxs.gather(dim, pos)
    return top_vals, top_idxs
