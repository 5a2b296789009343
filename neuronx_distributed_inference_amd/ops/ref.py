"""Pure-PyTorch reference implementations of every device op.

These are (a) the CPU execution path (the analog of the reference's
`to_cpu()` gloo fallback, `models/application_base.py:556-628`) and (b) the
numerics oracle for the HIP kernels: every GPU test compares the CDNA4
kernel against these run in fp32.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch


def rms_norm(
    x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6,
    residual: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    """RMSNorm with optional fused residual add.

    Returns (normed, new_residual): when `residual` is given the op computes
    h = x + residual first, returns rmsnorm(h) and h.
    """
    orig_dtype = x.dtype
    xf = x.float()
    if residual is not None:
        xf = xf + residual.float()
        new_residual = xf.to(orig_dtype)
    else:
        new_residual = None
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps)
    out = out * weight.float()
    return out.to(orig_dtype), new_residual


def rope_apply(
    q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Rotary embedding, rotate-half convention (HF Llama).

    q,k: (B, H, S, D); cos,sin: (B, S, D) or (S, D) broadcastable.
    """

    def rot_half(t):
        d = t.shape[-1] // 2
        return torch.cat((-t[..., d:], t[..., :d]), dim=-1)

    if cos.dim() == 2:
        cos = cos.unsqueeze(0)
        sin = sin.unsqueeze(0)
    cos = cos.unsqueeze(1)  # (B,1,S,D)
    sin = sin.unsqueeze(1)
    qf, kf = q.float(), k.float()
    cf, sf = cos.float(), sin.float()
    q_out = qf * cf + rot_half(qf) * sf
    k_out = kf * cf + rot_half(kf) * sf
    return q_out.to(q.dtype), k_out.to(k.dtype)


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    return (torch.nn.functional.silu(gate.float()) * up.float()).to(gate.dtype)


def gelu_tanh_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    return (torch.nn.functional.gelu(gate.float(), approximate="tanh") * up.float()).to(gate.dtype)


def kv_cache_update(
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    k_new: torch.Tensor,
    v_new: torch.Tensor,
    positions: torch.Tensor,
    seq_ids: Optional[torch.Tensor] = None,
) -> None:
    """Scatter-write new K/V rows into a linear (B, Hkv, S, D) cache in place.

    k_new/v_new: (B, Hkv, S_new, D); positions: (B, S_new) target sequence
    positions; seq_ids: (B,) cache-row mapping for continuous batching
    (ref `kv_cache_manager.py:369-586`, `:616`).
    """
    B, Hkv, S_new, D = k_new.shape
    if seq_ids is None:
        seq_ids = torch.arange(B, device=k_new.device)
    max_s = k_cache.shape[2]
    for b in range(B):
        row = int(seq_ids[b])
        if row < 0 or row >= k_cache.shape[0]:
            continue  # garbage row for SPMD-uniform DP writes
        valid = (positions[b] >= 0) & (positions[b] < max_s)
        pos = positions[b][valid].long()
        k_cache[row, :, pos, :] = k_new[b][:, valid].to(k_cache.dtype)
        v_cache[row, :, pos, :] = v_new[b][:, valid].to(v_cache.dtype)


def block_kv_cache_update(
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    k_new: torch.Tensor,
    v_new: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    """Paged cache scatter (ref `block_kv_cache_manager.py:268-375`).

    k_cache/v_cache: (num_blocks, block_size, Hkv, D); k_new/v_new:
    (N_tokens, Hkv, D); slot_mapping: (N_tokens,) flat slot index (or -1).
    """
    bs = k_cache.shape[1]
    flat_k = k_cache.view(-1, *k_cache.shape[2:])
    flat_v = v_cache.view(-1, *v_cache.shape[2:])
    valid = slot_mapping >= 0
    slots = slot_mapping[valid].long()
    flat_k[slots] = k_new[valid].to(flat_k.dtype)
    flat_v[slots] = v_new[valid].to(flat_v.dtype)


def sdpa(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    mask: Optional[torch.Tensor] = None,
    scale: Optional[float] = None,
    sinks: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Masked attention in fp32: q (B,Hq,Sq,D), k/v (B,Hkv,Skv,D); GQA by
    head repetition; optional learned attention sinks (GPT-OSS,
    ref `modules/attention/sink.py`)."""
    B, Hq, Sq, D = q.shape
    Hkv = k.shape[1]
    if Hkv != Hq:
        rep = Hq // Hkv
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    scores = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    if mask is not None:
        if mask.dtype == torch.bool:
            # -inf (not finfo.min): a FULLY-masked row (pad query) must
            # produce zero output, not a uniform average (softmax over a
            # constant-min row is uniform — the kernel/ref divergence the
            # ragged ring test caught)
            scores = scores.masked_fill(~mask, float("-inf"))
        else:
            scores = scores + mask.float()
    if sinks is not None:
        sink_col = sinks.float().view(1, Hq, 1, 1).expand(B, Hq, Sq, 1)
        scores = torch.cat([scores, sink_col], dim=-1)
        probs = torch.softmax(scores, dim=-1)[..., :-1]
    else:
        probs = torch.softmax(scores, dim=-1)
    probs = torch.nan_to_num(probs)  # fully-masked rows -> 0
    out = torch.matmul(probs, v.float())
    return out.to(q.dtype)


def paged_gather_kv(
    k_cache: torch.Tensor, v_cache: torch.Tensor, block_table: torch.Tensor, max_len: int
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Gather per-sequence contiguous (B,Hkv,S,D) K/V from a paged cache via
    `block_table` (B, max_blocks) (ref `block_kv_cache_manager.py:150-182`)."""
    B = block_table.shape[0]
    bs = k_cache.shape[1]
    nblk = (max_len + bs - 1) // bs
    tbl = block_table[:, :nblk].long().clamp(min=0)
    k = k_cache[tbl]  # (B, nblk, bs, Hkv, D)
    v = v_cache[tbl]
    k = k.flatten(1, 2)[:, :max_len].permute(0, 2, 1, 3)
    v = v.flatten(1, 2)[:, :max_len].permute(0, 2, 1, 3)
    return k.contiguous(), v.contiguous()


def topk_topp_sample(
    logits: torch.Tensor,
    top_k: torch.Tensor,
    top_p: torch.Tensor,
    temperature: torch.Tensor,
    rand: torch.Tensor,
    global_topk: int = 256,
) -> torch.Tensor:
    """Per-request top-k -> temperature -> top-p -> inverse-CDF sample
    (ref `modules/generation/sampling.py:285-370`).

    logits: (B, V); top_k/top_p/temperature/rand: (B,). rand in [0,1).
    Returns (B,) sampled token ids. top_k<=0 means greedy.
    """
    B, V = logits.shape
    kcap = min(global_topk, V)
    vals, idxs = logits.float().topk(kcap, dim=-1)  # sorted desc
    out = torch.empty(B, dtype=torch.long, device=logits.device)
    for b in range(B):
        k = int(top_k[b])
        if k == 1:
            out[b] = idxs[b, 0]
            continue
        k = kcap if k <= 0 else min(k, kcap)
        vb = vals[b, :k] / max(float(temperature[b]), 1e-6)
        probs = torch.softmax(vb, dim=-1)
        p = float(top_p[b])
        if p < 1.0:
            csum = probs.cumsum(0)
            keep = (csum - probs) < p  # always keep the first token
            probs = probs * keep
            probs = probs / probs.sum()
        cdf = probs.cumsum(0)
        j = int(torch.searchsorted(cdf, rand[b].float().clamp(max=1 - 1e-7)))
        j = min(j, k - 1)
        out[b] = idxs[b, j]
    return out
