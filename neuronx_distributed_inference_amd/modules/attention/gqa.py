"""GQA head-sharding math and the Q/K/V + output projections.

Analog of the reference's `modules/attention/gqa.py` (strategies :32-100,
QKV :566-631, O :1024, preshard hooks :137-245,679-954). Two sharding
strategies, picked automatically:

- ``CONVERT_TO_MHA``-free path: when num_kv_heads % tp == 0 each rank keeps
  num_kv_heads/tp distinct KV heads.
- ``REPLICATE_TO_TP_DEGREE``: when tp > num_kv_heads, KV heads are
  replicated tp/num_kv_heads times so every rank owns exactly one; the
  replication happens at checkpoint-shard time via each parameter's
  ``preshard_fn`` (see `utils/checkpoint.shard_state_dict`).
"""

from __future__ import annotations

import enum
from typing import Optional, Tuple

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd.parallel.layers import (
    ColumnParallelLinear,
    RowParallelLinear,
)


class GQA(enum.Enum):
    MHA = "mha"
    SHARD_OVER_TP = "shard"
    REPLICATE_TO_TP_DEGREE = "replicate"


def get_shardable_head_counts(
    tp: int, num_heads: int, num_kv_heads: int
) -> Tuple[int, int, GQA]:
    """Returns (padded_num_heads, effective_num_kv_heads, strategy)."""
    padded_heads = ((num_heads + tp - 1) // tp) * tp
    if num_kv_heads == num_heads:
        return padded_heads, padded_heads, GQA.MHA
    if num_kv_heads % tp == 0:
        return padded_heads, num_kv_heads, GQA.SHARD_OVER_TP
    if tp % num_kv_heads == 0:
        return padded_heads, tp, GQA.REPLICATE_TO_TP_DEGREE
    # fall back: replicate to lcm-compatible count
    rep = (tp + num_kv_heads - 1) // num_kv_heads
    return padded_heads, num_kv_heads * rep, GQA.REPLICATE_TO_TP_DEGREE


def _replicate_head_blocks(w: torch.Tensor, head_dim: int, rep: int) -> torch.Tensor:
    """Repeat each head_dim-sized row block `rep` times (dim 0)."""
    out_rows = w.shape[0]
    heads = out_rows // head_dim
    w = w.view(heads, head_dim, *w.shape[1:])
    w = torch.repeat_interleave(w, rep, dim=0)
    return w.reshape(heads * rep * head_dim, *w.shape[2:])


def _pad_head_blocks(w: torch.Tensor, head_dim: int, target_heads: int) -> torch.Tensor:
    heads = w.shape[0] // head_dim
    if heads == target_heads:
        return w
    pad = torch.zeros(
        (target_heads - heads) * head_dim, *w.shape[1:], dtype=w.dtype, device=w.device
    )
    return torch.cat([w, pad], dim=0)


class GroupQueryAttentionQKV(nn.Module):
    """Column-parallel Q/K/V projections with GQA-aware presharding."""

    def __init__(
        self,
        hidden_size: int,
        num_heads: int,
        num_kv_heads: int,
        head_dim: int,
        tp_degree: int,
        dtype: torch.dtype,
        bias: bool = False,
        fused: bool = False,
        group=None,
    ):
        super().__init__()
        self.hidden_size = hidden_size
        self.head_dim = head_dim
        self.tp_degree = tp_degree
        self.fused = fused
        padded_heads, eff_kv_heads, self.strategy = get_shardable_head_counts(
            tp_degree, num_heads, num_kv_heads
        )
        self.num_heads = padded_heads
        self.num_kv_heads = eff_kv_heads
        self.orig_num_heads = num_heads
        self.orig_num_kv_heads = num_kv_heads
        self.num_heads_per_rank = padded_heads // tp_degree
        self.num_kv_heads_per_rank = max(eff_kv_heads // tp_degree, 1)

        hd = head_dim
        rep = (eff_kv_heads // num_kv_heads
               if self.strategy == GQA.REPLICATE_TO_TP_DEGREE else 1)

        if fused:
            # ONE column-parallel GEMM for q|k|v. The checkpoint carries the
            # full cat([q, k, v]); the preshard hook pads q heads, replicates
            # kv heads and reorders rows rank-major so a plain dim-0 slice
            # yields [q_r | k_r | v_r] per rank.
            out_rows = (padded_heads + 2 * eff_kv_heads) * head_dim
            self.qkv = ColumnParallelLinear(
                hidden_size, out_rows, bias=bias, dtype=dtype,
                tp_degree=tp_degree, group=group,
            )

            def _fuse_preshard(w, tp, hd=hd, rep=rep,
                               nq=num_heads, nkv=num_kv_heads,
                               pq=padded_heads, ekv=eff_kv_heads):
                q, k, v = torch.split(w, [nq * hd, nkv * hd, nkv * hd], dim=0)
                q = _pad_head_blocks(q, hd, pq)
                if rep > 1:
                    k = _replicate_head_blocks(k, hd, rep)
                    v = _replicate_head_blocks(v, hd, rep)
                qpr, kpr = pq * hd // tp, ekv * hd // tp
                parts = []
                for r in range(tp):
                    parts.append(q[r * qpr:(r + 1) * qpr])
                    parts.append(k[r * kpr:(r + 1) * kpr])
                    parts.append(v[r * kpr:(r + 1) * kpr])
                return torch.cat(parts, dim=0)

            self.qkv.weight.preshard_fn = _fuse_preshard
            if bias:
                self.qkv.bias.preshard_fn = _fuse_preshard
            self._split_sizes = (
                self.num_heads_per_rank * head_dim,
                self.num_kv_heads_per_rank * head_dim,
                self.num_kv_heads_per_rank * head_dim,
            )
            return

        self.q_proj = ColumnParallelLinear(
            hidden_size, padded_heads * head_dim, bias=bias, dtype=dtype,
            tp_degree=tp_degree, group=group,
        )
        self.k_proj = ColumnParallelLinear(
            hidden_size, eff_kv_heads * head_dim, bias=bias, dtype=dtype,
            tp_degree=tp_degree, group=group,
        )
        self.v_proj = ColumnParallelLinear(
            hidden_size, eff_kv_heads * head_dim, bias=bias, dtype=dtype,
            tp_degree=tp_degree, group=group,
        )

        # preshard hooks: transform the FULL checkpoint tensor so dim-0
        # slicing yields the right shard (ref gqa.py:679-954)
        if padded_heads != num_heads:
            for p in ([self.q_proj.weight] + ([self.q_proj.bias] if bias else [])):
                p.preshard_fn = lambda w, tp, hd=hd, th=padded_heads: _pad_head_blocks(w, hd, th)
        if self.strategy == GQA.REPLICATE_TO_TP_DEGREE:
            for proj in (self.k_proj, self.v_proj):
                params = [proj.weight] + ([proj.bias] if bias else [])
                for p in params:
                    p.preshard_fn = lambda w, tp, hd=hd, rep=rep: _replicate_head_blocks(w, hd, rep)

    def forward(self, x: torch.Tensor, sequence_parallel: bool = False):
        sp = sequence_parallel
        if self.fused:
            y = self.qkv(x, sequence_parallel=sp)
            return torch.split(y, self._split_sizes, dim=-1)
        return (self.q_proj(x, sequence_parallel=sp),
                self.k_proj(x, sequence_parallel=sp),
                self.v_proj(x, sequence_parallel=sp))


class GroupQueryAttentionO(nn.Module):
    """Row-parallel output projection; the trailing all-reduce is collective
    C1 (ref gqa.py GroupQueryAttention_O)."""

    def __init__(
        self,
        hidden_size: int,
        num_heads: int,
        head_dim: int,
        tp_degree: int,
        dtype: torch.dtype,
        bias: bool = False,
        group=None,
        reduce_output: bool = True,
    ):
        super().__init__()
        padded_heads, _, _ = get_shardable_head_counts(tp_degree, num_heads, num_heads)
        self.o_proj = RowParallelLinear(
            padded_heads * head_dim, hidden_size, bias=bias, dtype=dtype,
            tp_degree=tp_degree, group=group, reduce_output=reduce_output,
        )
        if padded_heads != num_heads:
            hd = head_dim
            # pad along the INPUT dim (dim 1) for o_proj
            def _pad_in(w, tp, hd=hd, th=padded_heads):
                heads = w.shape[1] // hd
                if heads == th:
                    return w
                pad = torch.zeros(w.shape[0], (th - heads) * hd, dtype=w.dtype, device=w.device)
                return torch.cat([w, pad], dim=1)

            self.o_proj.weight.preshard_fn = _pad_in

    def forward(self, x: torch.Tensor, sequence_parallel: bool = False) -> torch.Tensor:
        return self.o_proj(x, sequence_parallel=sequence_parallel)
