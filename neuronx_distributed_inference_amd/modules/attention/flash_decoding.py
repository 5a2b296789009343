"""Flash decoding: decode-time KV cache sharded along S across the ranks
that hold the same (replicated) KV head.

SURVEY §2.8 long-context mechanism 3 (ref flash_decoding strided KV cache
+ cross-rank LSE combine). MI355X design: the cache rows are STRIDED over
the kv-shard group (global position p lives on sub-rank p % fd at local
slot p // fd), which keeps the shards balanced as the sequence grows; each
rank runs the split-KV HIP decode kernel over its shard for the WHOLE
group's query heads (q all-gathered — a few KB) and the softmax partials
are exchanged head-wise with one all-to-all, then LSE-merged locally.

RCCL cost per layer step: all-gather of q (B*Hq_local*D*fd elems) + an
all-to-all of the same volume in (out, lse) — both tiny next to the KV
read the sharding saves.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.parallel import state as ps


def shard_write_positions(positions: torch.Tensor, fd: int, sub: int) -> torch.Tensor:
    """Global scatter positions -> this shard's local slots.

    Position p belongs to sub-rank p % fd at slot p // fd; everything else
    (and pads, p < 0) maps to -1 which the HIP scatter skips."""
    mine = (positions % fd == sub) & (positions >= 0)
    return torch.where(mine, positions // fd, torch.full_like(positions, -1))


def shard_query_positions(positions: torch.Tensor, fd: int, sub: int) -> torch.Tensor:
    """Per-query effective position INSIDE this shard: the local slot of the
    newest owned row with global position <= p, or -1 if none (empty)."""
    loc = (positions - sub).div(fd, rounding_mode="floor")
    return torch.where((positions >= sub) & (positions >= 0), loc,
                       torch.full_like(positions, -1))


def flash_decoding_attention(
    q: torch.Tensor,                # (B, Hq_local, Sq, D) this rank's heads
    k_cache: torch.Tensor,          # (Bc, Hkv_local, Smax/fd, D) local shard
    v_cache: torch.Tensor,
    positions: torch.Tensor,        # (B, Sq) GLOBAL per-query positions
    seq_ids: Optional[torch.Tensor],
    scale: float,
    group=None,
) -> torch.Tensor:
    group = group if group is not None else ps.get_kv_shard_group()
    fd = dist.get_world_size(group=group)
    sub = dist.get_rank(group=group)
    B, Hl, Sq, D = q.shape

    # every rank computes the whole group's heads over its own S-shard
    q_all = [torch.empty_like(q) for _ in range(fd)]
    dist.all_gather(q_all, q.contiguous(), group=group)
    q_grp = torch.cat(q_all, dim=1)  # (B, fd*Hl, Sq, D)

    pos_local = shard_query_positions(positions, fd, sub)
    out, lse = ops.attention_decode_lse(
        q_grp, k_cache, v_cache, pos_local, seq_ids=seq_ids, scale=scale,
    )

    # head-wise exchange: shard j's partial for rank i's heads -> rank i
    outs = [t.contiguous() for t in out.chunk(fd, dim=1)]
    lses = [t.contiguous() for t in lse.chunk(fd, dim=1)]
    out_rx = [torch.empty_like(outs[0]) for _ in range(fd)]
    lse_rx = [torch.empty_like(lses[0]) for _ in range(fd)]
    if dist.get_backend(group) == "gloo":
        # gloo has no all_to_all: emulate with per-chunk all_gather
        for j in range(fd):
            buf_o = [torch.empty_like(outs[0]) for _ in range(fd)]
            buf_l = [torch.empty_like(lses[0]) for _ in range(fd)]
            dist.all_gather(buf_o, outs[j], group=group)
            dist.all_gather(buf_l, lses[j], group=group)
            if j == sub:
                out_rx, lse_rx = buf_o, buf_l
    else:
        dist.all_to_all(out_rx, outs, group=group)
        dist.all_to_all(lse_rx, lses, group=group)

    return ops.combine_lse_partials(out_rx, lse_rx)
