"""Process-group state for TP / CP / attention-DP / EP over RCCL.

MI355X scaling model: one OS process per GPU, ``torch.distributed`` with the
"nccl" backend (= RCCL over xGMI on ROCm) on GPU and "gloo" on CPU (tests).
This replaces the reference's SPMD/XLA world where one process drives all
NeuronCores (ref `neuronx_distributed.parallel_layers.parallel_state` used at
`models/model_base.py:150-154`) and the CP/DP mesh factory
(`modules/attention/attention_process_groups.py:11-166`).

Mesh math (generic form of the reference's example at
attention_process_groups.py:81-166): with world = tp and cp dividing tp,
prefill runs in tp/cp-sized TP subgroups [i*s .. i*s+s-1] and CP groups are
the strided complements [j, j+s, j+2s, ...]. Decode attention-DP uses the
same construction with dp in place of cp.

With world_size == 1 (or torch.distributed not initialized) every group is
None and all sizes collapse to 1, so single-GPU runs need no init.
"""

from __future__ import annotations

import logging
import os
from dataclasses import dataclass, field
from typing import List, Optional

import torch
from contextlib import contextmanager

import torch.distributed as dist

logger = logging.getLogger("nxdi_amd")


@dataclass
class _ParallelState:
    initialized: bool = False
    tp_degree: int = 1
    pp_degree: int = 1
    ep_degree: int = 1
    cp_degree: int = 1
    attention_dp_degree: int = 1
    world_group: Optional[object] = None
    tp_group: Optional[object] = None
    ep_group: Optional[object] = None
    moe_tp_group: Optional[object] = None
    cp_group: Optional[object] = None           # context-parallel group (prefill)
    prefill_tp_group: Optional[object] = None   # reduced-TP group inside CP
    dp_group: Optional[object] = None           # attention data-parallel (decode)
    decode_tp_group: Optional[object] = None
    draft_tp_group: Optional[object] = None     # speculative draft sub-communicator
    kv_shard_degree: int = 1
    kv_shard_group: Optional[object] = None     # flash-decoding S-shard group
    tp_mesh: List[List[int]] = field(default_factory=list)


_STATE = _ParallelState()


def _dist_world() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def _dist_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def model_parallel_is_initialized() -> bool:
    return _STATE.initialized


def initialize_model_parallel(
    tensor_model_parallel_size: int = 1,
    pipeline_model_parallel_size: int = 1,
    expert_model_parallel_size: int = 1,
    context_parallel_size: int = 1,
    attention_dp_size: int = 1,
    draft_tp_size: Optional[int] = None,
    kv_shard_size: int = 1,
) -> None:
    """Create all communicators. Must be called identically on every rank."""
    global _STATE
    tp, pp, ep = tensor_model_parallel_size, pipeline_model_parallel_size, expert_model_parallel_size
    world = _dist_world()
    if tp * pp > world and world > 1:
        raise ValueError(f"tp({tp})*pp({pp}) > world({world})")
    st = _ParallelState(
        initialized=True,
        tp_degree=tp,
        pp_degree=pp,
        ep_degree=ep,
        cp_degree=context_parallel_size,
        attention_dp_degree=attention_dp_size,
        kv_shard_degree=kv_shard_size,
    )

    if world == 1 or not dist.is_initialized():
        st.tp_mesh = [[0]]
        _STATE = st
        return

    rank = _dist_rank()
    st.world_group = dist.group.WORLD

    # TP groups: contiguous rank windows of size tp.
    st.tp_mesh = [list(range(i * tp, (i + 1) * tp)) for i in range(world // tp)]
    for mesh in st.tp_mesh:
        g = dist.new_group(mesh)
        if rank in mesh:
            st.tp_group = g

    # EP groups partition each TP group into ep strided sub-groups so that a
    # (moe_tp x moe_ep) grid covers it: moe_tp = tp // ep contiguous, ep strided.
    if ep > 1:
        moe_tp = tp // ep
        for mesh in st.tp_mesh:
            for j in range(moe_tp):
                sub = mesh[j::moe_tp]  # strided: experts across far ranks
                g = dist.new_group(sub)
                if rank in sub:
                    st.ep_group = g
            for j in range(ep):
                sub = mesh[j * moe_tp : (j + 1) * moe_tp]
                g = dist.new_group(sub)
                if rank in sub:
                    st.moe_tp_group = g

    # CP: prefill attention subgroups (ref attention_process_groups.py:81-124).
    if context_parallel_size > 1:
        s = tp // context_parallel_size
        for mesh in st.tp_mesh:
            for i in range(context_parallel_size):
                sub = mesh[i * s : (i + 1) * s]
                g = dist.new_group(sub)
                if rank in sub:
                    st.prefill_tp_group = g
            for j in range(s):
                sub = mesh[j::s]
                g = dist.new_group(sub)
                if rank in sub:
                    st.cp_group = g

    # attention-DP: decode batch-sharding subgroups (ref :125-166).
    if attention_dp_size > 1:
        s = tp // attention_dp_size
        for mesh in st.tp_mesh:
            for i in range(attention_dp_size):
                sub = mesh[i * s : (i + 1) * s]
                g = dist.new_group(sub)
                if rank in sub:
                    st.decode_tp_group = g
            for j in range(s):
                sub = mesh[j::s]
                g = dist.new_group(sub)
                if rank in sub:
                    st.dp_group = g

    # Flash decoding (SURVEY §2.8 long-context mechanism 3): the ranks that
    # hold the SAME replicated KV head (contiguous windows of size
    # kv_shard_size inside each TP group) shard the decode KV cache along S
    # and LSE-combine their partial attentions.
    if kv_shard_size > 1:
        for mesh in st.tp_mesh:
            for i in range(tp // kv_shard_size):
                sub = mesh[i * kv_shard_size : (i + 1) * kv_shard_size]
                g = dist.new_group(sub)
                if rank in sub:
                    st.kv_shard_group = g

    # Speculative-draft group: draft model at smaller TP
    # (ref application_base.py:61-66, utils/distributed.py:30-36).
    # EVERY rank joins one subgroup (the draft's weights are replicated
    # across subgroups), so all ranks compute the draft locally at the
    # reduced degree — no broadcast of draft tokens, and the draft's
    # all-reduces run over draft_tp ranks instead of tp.
    if draft_tp_size and draft_tp_size < tp:
        if tp % draft_tp_size:
            raise ValueError("draft_tp_size must divide tp_degree")
        for mesh in st.tp_mesh:
            for i in range(tp // draft_tp_size):
                sub = mesh[i * draft_tp_size : (i + 1) * draft_tp_size]
                g = dist.new_group(sub)
                if rank in sub:
                    st.draft_tp_group = g

    _STATE = st


def destroy_model_parallel():
    global _STATE
    _STATE = _ParallelState()


# ----------------------------------------------------------------------
def _group_size(group) -> int:
    if group is None or not dist.is_initialized():
        return 1
    return dist.get_world_size(group=group)


def _group_rank(group) -> int:
    if group is None or not dist.is_initialized():
        return 0
    return dist.get_rank(group=group)


def get_world_group():
    return _STATE.world_group


def get_tensor_model_parallel_group():
    if _TP_OVERRIDE:
        return _TP_OVERRIDE[-1]
    return _STATE.tp_group


def get_tensor_model_parallel_size() -> int:
    if not dist.is_initialized():
        return 1
    if _TP_OVERRIDE:
        return dist.get_world_size(group=_TP_OVERRIDE[-1])
    return min(_STATE.tp_degree, _dist_world()) if _STATE.initialized else 1


def get_tensor_model_parallel_rank() -> int:
    if _TP_OVERRIDE:
        return _group_rank(_TP_OVERRIDE[-1])
    return _group_rank(_STATE.tp_group)


def get_context_parallel_group():
    return _STATE.cp_group


def get_context_parallel_size() -> int:
    return _group_size(_STATE.cp_group) if _STATE.cp_group is not None else (
        _STATE.cp_degree if dist.is_initialized() and _STATE.initialized else 1
    )


def get_context_parallel_rank() -> int:
    return _group_rank(_STATE.cp_group)


def get_prefill_tp_group():
    return _STATE.prefill_tp_group


def get_attention_dp_group():
    return _STATE.dp_group


def get_attention_dp_decode_tp_group():
    return _STATE.decode_tp_group


def get_attention_dp_size() -> int:
    return _group_size(_STATE.dp_group) if _STATE.dp_group is not None else (
        _STATE.attention_dp_degree if dist.is_initialized() and _STATE.initialized else 1
    )


def get_attention_dp_rank() -> int:
    return _group_rank(_STATE.dp_group)


def get_expert_parallel_group():
    return _STATE.ep_group


def get_expert_parallel_size() -> int:
    return _group_size(_STATE.ep_group) if _STATE.ep_group is not None else 1


def get_expert_parallel_rank() -> int:
    return _group_rank(_STATE.ep_group)


def get_moe_tp_group():
    return _STATE.moe_tp_group


def get_draft_tp_group():
    return _STATE.draft_tp_group


_TP_OVERRIDE: list = []


@contextmanager
def draft_tp_context():
    """Route the default TP group/size/rank to the speculative-draft
    subgroup for the duration (ref get_tp_group override,
    utils/distributed.py:30-36): the draft model's parallel layers and
    distributed argmax/topk then operate over the reduced-TP group.
    No-op when no draft group was initialized."""
    if _STATE.draft_tp_group is None:
        yield
        return
    _TP_OVERRIDE.append(_STATE.draft_tp_group)
    try:
        yield
    finally:
        _TP_OVERRIDE.pop()


def init_distributed_from_env(backend: Optional[str] = None) -> int:
    """Bootstrap torch.distributed from torchrun env vars; returns local rank.

    Backend: "nccl" (= RCCL) when a GPU is visible, else "gloo".
    """
    if dist.is_initialized():
        return int(os.environ.get("LOCAL_RANK", 0))
    if "RANK" not in os.environ:
        return 0  # single-process
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if backend == "nccl":
        torch.cuda.set_device(local_rank)
        dist.init_process_group(backend="nccl", device_id=torch.device(f"cuda:{local_rank}"))
    else:
        dist.init_process_group(backend="gloo")
    return local_rank


def get_kv_shard_group():
    return _STATE.kv_shard_group


def get_kv_shard_degree() -> int:
    return _STATE.kv_shard_degree


def get_kv_shard_rank() -> int:
    if _STATE.kv_shard_group is None or not dist.is_initialized():
        return 0
    return dist.get_rank(group=_STATE.kv_shard_group)
