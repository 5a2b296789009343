"""Attention mask construction and head-sharding helpers.

Analog of the reference's mask builders (`models/model_base.py:187-376`) and
attention utils (`modules/attention/utils.py`). Masks are boolean with True
= attend.
"""

from __future__ import annotations

from typing import Optional

import torch


def causal_mask(sq: int, skv: int, device=None) -> torch.Tensor:
    """(1,1,Sq,Skv) causal mask; queries are the LAST sq positions of skv."""
    return torch.ones(sq, skv, dtype=torch.bool, device=device).tril(skv - sq).view(1, 1, sq, skv)


def sliding_window_mask(sq: int, skv: int, window: int, device=None) -> torch.Tensor:
    m = torch.ones(sq, skv, dtype=torch.bool, device=device)
    m = m.tril(skv - sq) & m.triu(skv - sq - window + 1)
    return m.view(1, 1, sq, skv)


def chunked_attention_mask(sq: int, skv: int, chunk: int, device=None) -> torch.Tensor:
    """Llama4 chunked attention: query at position i attends causally only
    within its own chunk [(i//chunk)*chunk, i] (HF Llama4 attention_chunk_size;
    ref llama4 chunked local layers). Queries are the LAST sq of skv."""
    qpos = torch.arange(skv - sq, skv, device=device).view(-1, 1)
    kpos = torch.arange(skv, device=device).view(1, -1)
    m = (kpos <= qpos) & (kpos >= (qpos // chunk) * chunk)
    return m.view(1, 1, sq, skv)


def padded_causal_mask(attention_mask: torch.Tensor, sq: int) -> torch.Tensor:
    """Combine a (B, Skv) padding mask with causality -> (B,1,Sq,Skv)."""
    skv = attention_mask.shape[-1]
    cm = causal_mask(sq, skv, device=attention_mask.device)
    return cm & attention_mask.view(-1, 1, 1, skv).bool()


def speculation_mask(batch: int, spec_len: int, prior_len: int, device=None) -> torch.Tensor:
    """Decode mask for spec_len active tokens: full prior + causal active
    (ref model_base.py:268-310)."""
    prior = torch.ones(spec_len, prior_len, dtype=torch.bool, device=device)
    active = torch.ones(spec_len, spec_len, dtype=torch.bool, device=device).tril()
    return torch.cat([prior, active], dim=-1).view(1, 1, spec_len, prior_len + spec_len).expand(
        batch, 1, spec_len, prior_len + spec_len
    )


def block_diagonal_mask(seq_lens: torch.Tensor, total: int) -> torch.Tensor:
    """Chunked-prefill mask: tokens attend causally within their own segment
    (ref modules/attention/utils.py:347)."""
    device = seq_lens.device
    seg = torch.repeat_interleave(
        torch.arange(len(seq_lens), device=device), seq_lens.long()
    )
    if seg.numel() < total:
        seg = torch.cat([seg, torch.full((total - seg.numel(),), -1, device=device)])
    same = seg.view(-1, 1) == seg.view(1, -1)
    causal = torch.ones(total, total, dtype=torch.bool, device=device).tril()
    return (same & causal).view(1, 1, total, total)


def repeat_kv(x: torch.Tensor, n_rep: int) -> torch.Tensor:
    """(B,Hkv,S,D) -> (B,Hkv*n_rep,S,D) (ref utils.py:219)."""
    if n_rep == 1:
        return x
    return x.repeat_interleave(n_rep, dim=1)
