"""Collective mappings for tensor / sequence / context parallelism.

Inference-only (no autograd) analogs of the reference's NxD mapping ops
(`_gather_along_dim` / `_reduce_scatter_along_dim`, used from
`models/model_base.py:1019-1023,1509-1514` and
`modules/attention/attention_base.py:548-561,1350-1358`).

On MI355X these lower to RCCL collectives over xGMI. xGMI is point-to-point
(7 links x ~153 GB/s per GPU), so ring collectives are per-link bound; the
decode-path all-reduce is latency-critical at tiny message sizes, which we
attack later with a fused low-latency path — the API here is the seam.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from neuronx_distributed_inference_amd.parallel import state as ps


def _size(group) -> int:
    if group is None or not dist.is_initialized():
        return 1
    return dist.get_world_size(group=group)


def _rank(group) -> int:
    if group is None or not dist.is_initialized():
        return 0
    return dist.get_rank(group=group)


def reduce_from_tensor_parallel_region(x: torch.Tensor, group=None) -> torch.Tensor:
    """All-reduce SUM over the TP group (ref C1: the per-token-latency-critical
    collective, 2x per layer after o-proj and down-proj)."""
    group = group if group is not None else ps.get_tensor_model_parallel_group()
    if _size(group) == 1:
        return x
    dist.all_reduce(x, op=dist.ReduceOp.SUM, group=group)
    return x


def gather_from_tensor_parallel_region(
    x: torch.Tensor, dim: int = -1, group=None
) -> torch.Tensor:
    """All-gather along `dim` (ref C3/C4/C7)."""
    group = group if group is not None else ps.get_tensor_model_parallel_group()
    world = _size(group)
    if world == 1:
        return x
    if dim < 0:
        dim = x.dim() + dim
    out_shards = [torch.empty_like(x) for _ in range(world)]
    dist.all_gather(out_shards, x.contiguous(), group=group)
    return torch.cat(out_shards, dim=dim)


def scatter_to_tensor_parallel_region(x: torch.Tensor, dim: int = -1, group=None) -> torch.Tensor:
    """Keep this rank's shard along `dim` (no communication)."""
    group = group if group is not None else ps.get_tensor_model_parallel_group()
    world = _size(group)
    if world == 1:
        return x
    if dim < 0:
        dim = x.dim() + dim
    size = x.size(dim) // world
    return x.narrow(dim, _rank(group) * size, size).contiguous()


def reduce_scatter_along_dim(x: torch.Tensor, dim: int = 0, group=None) -> torch.Tensor:
    """Reduce-scatter SUM along `dim` (ref C2: enter sequence-parallel region)."""
    group = group if group is not None else ps.get_tensor_model_parallel_group()
    world = _size(group)
    if world == 1:
        return x
    if dim < 0:
        dim = x.dim() + dim
    if dim != 0:
        x = x.movedim(dim, 0)
    x = x.contiguous()
    out = torch.empty(
        (x.size(0) // world,) + tuple(x.shape[1:]), dtype=x.dtype, device=x.device
    )
    dist.reduce_scatter_tensor(out, x, op=dist.ReduceOp.SUM, group=group)
    if dim != 0:
        out = out.movedim(0, dim)
    return out


def gather_along_dim(x: torch.Tensor, dim: int = 0, group=None) -> torch.Tensor:
    """All-gather along `dim`, concatenating shards in rank order (ref C3)."""
    group = group if group is not None else ps.get_tensor_model_parallel_group()
    world = _size(group)
    if world == 1:
        return x
    if dim < 0:
        dim = x.dim() + dim
    if dim != 0:
        x = x.movedim(dim, 0)
    x = x.contiguous()
    out = torch.empty((x.size(0) * world,) + tuple(x.shape[1:]), dtype=x.dtype, device=x.device)
    dist.all_gather_into_tensor(out, x, group=group)
    if dim != 0:
        out = out.movedim(0, dim)
    return out


def all_to_all(x: torch.Tensor, scatter_dim: int, gather_dim: int, group=None) -> torch.Tensor:
    """All-to-all reshard (ref C6 sampling-DP, C9 MoE expert dispatch)."""
    group = group if group is not None else ps.get_tensor_model_parallel_group()
    world = _size(group)
    if world == 1:
        return x
    if dist.get_backend(group) == "gloo":
        # gloo has no alltoall: emulate via all-gather + local select (CPU
        # test path only; RCCL uses the native collective)
        rank = _rank(group)
        gathered = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(gathered, x.contiguous(), group=group)
        picks = [g.chunk(world, dim=scatter_dim)[rank] for g in gathered]
        return torch.cat(picks, dim=gather_dim)
    inputs = [t.contiguous() for t in x.chunk(world, dim=scatter_dim)]
    outputs = [torch.empty_like(inputs[0]) for _ in range(world)]
    dist.all_to_all(outputs, inputs, group=group)
    return torch.cat(outputs, dim=gather_dim)
