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
        if scaling:
            rope_type = scaling.get("rope_type", scaling.get("type", "default"))
            if rope_type == "llama3":
                inv_freq = self._llama3_scale(inv_freq, scaling)
            elif rope_type in ("linear",):
                inv_freq = inv_freq / float(scaling.get("factor", 1.0))
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

    def _build_table(self, seqlen: int, device, dtype):
        t = torch.arange(seqlen, dtype=torch.float32, device=device)
        freqs = torch.outer(t, self.inv_freq.to(device))
        emb = torch.cat((freqs, freqs), dim=-1)  # (S, D)
        self._cos_cached = emb.cos()
        self._sin_cached = emb.sin()
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
