"""Reassemble the full KV cache from per-rank shards (debug utility).

Analog of the reference's cross-shard KV reconstruction
(`utils/kv_cache_reconstruct_utils.py:57-218`): given each rank's cache
tensor for one layer, undo the parallel layout so numerics can be
compared against a single-rank run. Handles the three layouts this
framework shards KV over:

- TP: KV heads split contiguously across ranks; when tp > num_kv_heads
  each head is replicated tp/num_kv_heads times (GQA replication) and
  the duplicates are collapsed.
- attention-DP: batch rows split across dp subgroups (rank r of the dp
  split holds rows [r*B/dp, (r+1)*B/dp)).
- flash-decoding kv-shard: sequence positions strided across fd
  sub-ranks (position p lives on sub-rank p % fd at slot p // fd).
"""

from __future__ import annotations

from typing import List, Sequence

import torch


def merge_tp_heads(shards: Sequence[torch.Tensor], num_kv_heads: int) -> torch.Tensor:
    """(tp x (B, Hl, S, D)) -> (B, num_kv_heads, S, D). Collapses GQA
    replication when tp > num_kv_heads."""
    tp = len(shards)
    if tp <= num_kv_heads:
        return torch.cat(list(shards), dim=1)[:, :num_kv_heads]
    # replicated: ranks r covering the same head group hold identical rows
    rep = tp // num_kv_heads
    return torch.cat([shards[h * rep] for h in range(num_kv_heads)], dim=1)


def merge_dp_batch(shards: Sequence[torch.Tensor]) -> torch.Tensor:
    """(dp x (Bl, H, S, D)) -> (B, H, S, D)."""
    return torch.cat(list(shards), dim=0)


def merge_kv_shard_seq(shards: Sequence[torch.Tensor],
                       total_len: int) -> torch.Tensor:
    """(fd x (B, H, S/fd, D)) -> (B, H, total_len, D): inverse of the
    strided flash-decoding write (pos p -> sub-rank p%fd, slot p//fd)."""
    fd = len(shards)
    B, H, Sl, D = shards[0].shape
    out = shards[0].new_zeros(B, H, fd * Sl, D)
    for r, s in enumerate(shards):
        out[:, :, r::fd] = s
    return out[:, :, :total_len]


def reconstruct_kv_cache(
    rank_caches: List[torch.Tensor],
    num_kv_heads: int,
    tp_degree: int,
    kv_shard_degree: int = 1,
    dp_degree: int = 1,
    total_len: int | None = None,
) -> torch.Tensor:
    """rank_caches[r] = rank r's (B_local, H_local, S_local, D) cache for
    one layer, in global rank order. Returns the unsharded
    (B, num_kv_heads, S, D) cache.

    Mesh convention mirrors parallel/state.py: within the TP window,
    kv-shard groups are CONTIGUOUS blocks of `kv_shard_degree` ranks
    (head-sharding is the outer dim), and attention-DP splits the SAME
    window with roles swapped (dp outer, heads inner).
    """
    world = len(rank_caches)
    assert world == tp_degree * dp_degree, (world, tp_degree, dp_degree)
    per_head = tp_degree // kv_shard_degree  # ranks per head-shard position

    dp_parts = []
    for d in range(dp_degree):
        tp_block = rank_caches[d * tp_degree:(d + 1) * tp_degree]
        head_parts = []
        for h in range(per_head):
            sub = tp_block[h * kv_shard_degree:(h + 1) * kv_shard_degree]
            if kv_shard_degree > 1:
                tl = total_len if total_len is not None else \
                    sub[0].shape[2] * kv_shard_degree
                head_parts.append(merge_kv_shard_seq(sub, tl))
            else:
                head_parts.append(sub[0])
        dp_parts.append(merge_tp_heads(head_parts, num_kv_heads)
                        if per_head > 1 else head_parts[0])
    full = merge_dp_batch(dp_parts) if dp_degree > 1 else dp_parts[0]
    return full[:, :num_kv_heads]
