"""Rotary position embeddings.

Host-side inv_freq setup + on-device cos/sin table lookup. Per the CDNA4
guide, trig is precomputed into a (max_len, D) cos/sin table once (device
`sinf/cosf` per element would turn RoPE memory-bound work VALU-bound); the
apply step is the fused HIP kernel `ops.rope_apply`.

Covers: default RoPE, Llama-3 scaled RoPE (ref `modeling_llama.py:805-871`),
dynamic NTC scaling hook.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn as nn


class RotaryEmbedding(nn.Module):
    def __init__(
        self,
        head_dim: int,
        max_position_embeddings: int = 4096,
        base: float = 10000.0,
        scaling: Optional[dict] = None,
        device=None,
    ):
        super().__init__()
        self.head_dim = head_dim
        self.max_position_embeddings = max_position_embeddings
        self.base = base
        inv_freq = 1.0 / (
            base ** (torch.arange(0, head_dim, 2, dtype=torch.float32) / head_dim)
        )
        self.attention_scaling = 1.0
        if scaling:
            rope_type = scaling.get("rope_type", scaling.get("type", "default"))
            if rope_type == "llama3":
                inv_freq = self._llama3_scale(inv_freq, scaling)
            elif rope_type in ("linear",):
                inv_freq = inv_freq / float(scaling.get("factor", 1.0))
            elif rope_type == "yarn":
                inv_freq, self.attention_scaling = self._yarn(head_dim, base, scaling)
            # "default"/"dynamic": leave as-is (dynamic handled at table build)
        self.register_buffer("inv_freq", inv_freq, persistent=False)
        self._cos_cached: Optional[torch.Tensor] = None
        self._sin_cached: Optional[torch.Tensor] = None
        self._cached_len = 0

    @staticmethod
    def _llama3_scale(inv_freq: torch.Tensor, s: dict) -> torch.Tensor:
        # ref modeling_llama.py:805-871 behavior (public Llama-3.1 rope scaling)
        factor = s.get("factor", 8.0)
        low = s.get("low_freq_factor", 1.0)
        high = s.get("high_freq_factor", 4.0)
        old_ctx = s.get("original_max_position_embeddings", 8192)
        wavelen = 2 * math.pi / inv_freq
        low_wl = old_ctx / low
        high_wl = old_ctx / high
        out = torch.where(wavelen > low_wl, inv_freq / factor, inv_freq)
        smooth = (old_ctx / wavelen - low) / (high - low)
        smoothed = (1 - smooth) / factor * inv_freq + smooth * inv_freq
        is_mid = (wavelen <= low_wl) & (wavelen >= high_wl)
        return torch.where(is_mid, smoothed, out)

    @staticmethod
    def _yarn(dim: int, base: float, s: dict):
        """YaRN NTK-by-parts scaling (GPT-OSS, DeepSeek; matches HF
        _compute_yarn_parameters)."""
        factor = s.get("factor")
        orig = s.get("original_max_position_embeddings", 4096)
        if factor is None:
            factor = 1.0
        beta_fast = s.get("beta_fast") or 32
        beta_slow = s.get("beta_slow") or 1
        mscale = s.get("mscale")
        mscale_all_dim = s.get("mscale_all_dim")

        def get_mscale(scale, m=1):
            return 1.0 if scale <= 1 else 0.1 * m * math.log(scale) + 1.0

        att = s.get("attention_factor")
        if att is None:
            if mscale and mscale_all_dim:
                att = float(get_mscale(factor, mscale) / get_mscale(factor, mscale_all_dim))
            else:
                att = get_mscale(factor)

        def corr_dim(rot):
            return (dim * math.log(orig / (rot * 2 * math.pi))) / (2 * math.log(base))

        low, high = corr_dim(beta_fast), corr_dim(beta_slow)
        if s.get("truncate", True):
            low, high = math.floor(low), math.ceil(high)
        low, high = max(low, 0), min(high, dim - 1)
        if low == high:
            high += 0.001
        pos_freqs = base ** (torch.arange(0, dim, 2, dtype=torch.float32) / dim)
        extrap = 1.0 / pos_freqs
        interp = 1.0 / (factor * pos_freqs)
        ramp = ((torch.arange(dim // 2, dtype=torch.float32) - low) / (high - low)).clamp(0, 1)
        extrap_factor = 1 - ramp
        inv_freq = interp * (1 - extrap_factor) + extrap * extrap_factor
        return inv_freq, float(att)

    def _build_table(self, seqlen: int, device, dtype):
        t = torch.arange(seqlen, dtype=torch.float32, device=device)
        freqs = torch.outer(t, self.inv_freq.to(device))
        emb = torch.cat((freqs, freqs), dim=-1)  # (S, D)
        self._cos_cached = emb.cos() * self.attention_scaling
        self._sin_cached = emb.sin() * self.attention_scaling
        self._cached_len = seqlen

    def ensure_table(self, seqlen: int, device):
        """Build the cos/sin table up to `seqlen` once (host-side, outside
        any hipGraph capture)."""
        seqlen = max(seqlen, self.max_position_embeddings)
        if (
            self._cos_cached is None
            or self._cached_len < seqlen
            or self._cos_cached.device != torch.device(device)
        ):
            self._build_table(seqlen, device, torch.float32)

    def get_cos_sin(
        self, position_ids: torch.Tensor, device=None, dtype=torch.float32
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        """position_ids: (B, S) -> cos/sin (B, S, D) fp32.

        Graph-capture safe: pure tensor indexing, no host sync — the table
        must already cover the positions (ensure_table)."""
        device = device or position_ids.device
        self.ensure_table(1, device)
        pos = position_ids.long().clamp(min=0)
        if self._cached_len > 0:
            pos = pos.clamp(max=self._cached_len - 1)
        return self._cos_cached[pos], self._sin_cached[pos]


def rope_apply_interleaved(q: torch.Tensor, k: torch.Tensor,
                           cos: torch.Tensor, sin: torch.Tensor):
    """Complex-pair (interleaved) RoPE, pairing (2i, 2i+1) lanes — the HF
    Llama4 `apply_rotary_emb` convention (view_as_complex), output kept
    interleaved. cos/sin are the (B,S,D) duplicated-half tables produced by
    RotaryEmbedding.get_cos_sin; pair i's angle sits in the first half."""
    d = cos.shape[-1] // 2
    c = cos[..., :d].unsqueeze(1)  # (B,1,S,d)
    s = sin[..., :d].unsqueeze(1)
    q1, q2 = q[..., 0::2].float(), q[..., 1::2].float()
    k1, k2 = k[..., 0::2].float(), k[..., 1::2].float()
    q_out = torch.stack([q1 * c - q2 * s, q2 * c + q1 * s], dim=-1).flatten(-2)
    k_out = torch.stack([k1 * c - k2 * s, k2 * c + k1 * s], dim=-1).flatten(-2)
    return q_out.to(q.dtype), k_out.to(k.dtype)
