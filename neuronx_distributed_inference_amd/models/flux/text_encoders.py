"""Flux prompt encoders: CLIP text (pooled vector) + T5 encoder (sequence).

Parity target: the reference's `flux/clip/modeling_clip.py` and
`flux/t5/modeling_t5.py` (SURVEY §2.11 Flux row). These run once per
request (prefill-shaped, library GEMMs); both are HF-checkpoint-compatible
and oracle-tested against `transformers`.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


# ----------------------------------------------------------------------
# CLIP text encoder (pooled projection source)
class _ClipAttention(nn.Module):
    def __init__(self, dim: int, heads: int):
        super().__init__()
        self.heads = heads
        self.head_dim = dim // heads
        self.q_proj = nn.Linear(dim, dim)
        self.k_proj = nn.Linear(dim, dim)
        self.v_proj = nn.Linear(dim, dim)
        self.out_proj = nn.Linear(dim, dim)

    def forward(self, x):
        B, S, _ = x.shape
        sh = lambda t: t.view(B, S, self.heads, self.head_dim).transpose(1, 2)  # noqa: E731
        out = F.scaled_dot_product_attention(
            sh(self.q_proj(x)), sh(self.k_proj(x)), sh(self.v_proj(x)),
            is_causal=True)
        return self.out_proj(out.transpose(1, 2).reshape(B, S, -1))


class _ClipLayer(nn.Module):
    def __init__(self, dim: int, heads: int, inter: int):
        super().__init__()
        self.self_attn = _ClipAttention(dim, heads)
        self.layer_norm1 = nn.LayerNorm(dim)
        self.layer_norm2 = nn.LayerNorm(dim)
        self.mlp = nn.ModuleDict({
            "fc1": nn.Linear(dim, inter), "fc2": nn.Linear(inter, dim),
        })

    def forward(self, x):
        x = x + self.self_attn(self.layer_norm1(x))
        h = self.mlp["fc1"](self.layer_norm2(x))
        h = h * torch.sigmoid(1.702 * h)          # quick_gelu
        return x + self.mlp["fc2"](h)


class ClipTextEncoder(nn.Module):
    """HF CLIPTextModel-compatible (causal transformer, quick-gelu,
    argmax-eos pooling)."""

    def __init__(self, vocab_size: int, dim: int, heads: int, inter: int,
                 layers: int, max_pos: int, eos_token_id: int = 2):
        super().__init__()
        self.eos_token_id = eos_token_id
        self.embeddings = nn.ModuleDict({
            "token_embedding": nn.Embedding(vocab_size, dim),
            "position_embedding": nn.Embedding(max_pos, dim),
        })
        self.encoder_layers = nn.ModuleList(
            _ClipLayer(dim, heads, inter) for _ in range(layers))
        self.final_layer_norm = nn.LayerNorm(dim)

    def forward(self, input_ids: torch.Tensor):
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device)
        h = self.embeddings["token_embedding"](input_ids) + \
            self.embeddings["position_embedding"](pos)
        for layer in self.encoder_layers:
            h = layer(h)
        h = self.final_layer_norm(h)
        # pooled = hidden state at the eos token. HF keeps the LEGACY rule
        # for eos_token_id == 2: the position of the highest token id
        # (original CLIP vocab puts eos at the max id)
        if self.eos_token_id == 2:
            eos_pos = input_ids.argmax(dim=-1)
        else:
            eos_pos = (input_ids == self.eos_token_id).int().argmax(dim=-1)
        pooled = h[torch.arange(B, device=h.device), eos_pos]
        return h, pooled

    @staticmethod
    def convert_hf_to_native_state_dict(sd):
        out = {}
        for k, v in sd.items():
            nk = k
            for pre in ("text_model.",):
                if nk.startswith(pre):
                    nk = nk[len(pre):]
            nk = nk.replace("encoder.layers.", "encoder_layers.")
            out[nk] = v
        return out


# ----------------------------------------------------------------------
# T5 encoder (sequence embedding source)
class T5LayerNorm(nn.Module):
    """No mean subtraction, no bias (T5 convention)."""

    def __init__(self, dim: int, eps: float = 1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x):
        var = x.float().pow(2).mean(-1, keepdim=True)
        return self.weight * (x.float() * torch.rsqrt(var + self.eps)).to(x.dtype)


class _T5Attention(nn.Module):
    def __init__(self, dim: int, heads: int, kv_dim: int,
                 has_bias_table: bool, num_buckets: int, max_distance: int):
        super().__init__()
        self.heads = heads
        self.kv_dim = kv_dim
        self.q = nn.Linear(dim, heads * kv_dim, bias=False)
        self.k = nn.Linear(dim, heads * kv_dim, bias=False)
        self.v = nn.Linear(dim, heads * kv_dim, bias=False)
        self.o = nn.Linear(heads * kv_dim, dim, bias=False)
        self.num_buckets = num_buckets
        self.max_distance = max_distance
        if has_bias_table:
            self.relative_attention_bias = nn.Embedding(num_buckets, heads)

    def _bucket(self, rel: torch.Tensor) -> torch.Tensor:
        """Bidirectional relative-position bucketing (HF T5Attention)."""
        nb = self.num_buckets // 2
        out = (rel > 0).long() * nb
        n = rel.abs()
        max_exact = nb // 2
        large = max_exact + (
            torch.log(n.float() / max_exact) / math.log(self.max_distance / max_exact)
            * (nb - max_exact)
        ).long()
        large = torch.minimum(large, torch.full_like(large, nb - 1))
        return out + torch.where(n < max_exact, n, large)

    def bias(self, S: int, device) -> torch.Tensor:
        ctx = torch.arange(S, device=device)[:, None]
        mem = torch.arange(S, device=device)[None, :]
        buckets = self._bucket(mem - ctx)
        return self.relative_attention_bias(buckets).permute(2, 0, 1).unsqueeze(0)

    def forward(self, x, pos_bias):
        B, S, _ = x.shape
        sh = lambda t: t.view(B, S, self.heads, self.kv_dim).transpose(1, 2)  # noqa: E731
        # T5 uses UNSCALED dot products; the bias is additive
        scores = sh(self.q(x)) @ sh(self.k(x)).transpose(-1, -2) + pos_bias
        probs = torch.softmax(scores.float(), dim=-1).to(x.dtype)
        out = probs @ sh(self.v(x))
        return self.o(out.transpose(1, 2).reshape(B, S, -1))


class _T5Block(nn.Module):
    def __init__(self, dim: int, heads: int, kv_dim: int, ff_dim: int,
                 first: bool, num_buckets: int, max_distance: int):
        super().__init__()
        self.attn = _T5Attention(dim, heads, kv_dim, first, num_buckets,
                                 max_distance)
        self.attn_norm = T5LayerNorm(dim)
        self.wi_0 = nn.Linear(dim, ff_dim, bias=False)
        self.wi_1 = nn.Linear(dim, ff_dim, bias=False)
        self.wo = nn.Linear(ff_dim, dim, bias=False)
        self.ff_norm = T5LayerNorm(dim)

    def forward(self, x, pos_bias):
        x = x + self.attn(self.attn_norm(x), pos_bias)
        h = self.ff_norm(x)
        h = F.gelu(self.wi_0(h), approximate="tanh") * self.wi_1(h)
        return x + self.wo(h)


class T5TextEncoder(nn.Module):
    """HF T5EncoderModel-compatible (gated-gelu variant, e.g. t5-v1_1 /
    the Flux T5-XXL)."""

    def __init__(self, vocab_size: int, dim: int, heads: int, kv_dim: int,
                 ff_dim: int, layers: int, num_buckets: int = 32,
                 max_distance: int = 128):
        super().__init__()
        self.shared = nn.Embedding(vocab_size, dim)
        self.blocks = nn.ModuleList(
            _T5Block(dim, heads, kv_dim, ff_dim, i == 0, num_buckets,
                     max_distance)
            for i in range(layers))
        self.final_layer_norm = T5LayerNorm(dim)

    def forward(self, input_ids: torch.Tensor):
        h = self.shared(input_ids)
        bias = self.blocks[0].attn.bias(input_ids.shape[1], input_ids.device)
        for blk in self.blocks:
            h = blk(h, bias)
        return self.final_layer_norm(h)

    @staticmethod
    def convert_hf_to_native_state_dict(sd):
        out = {}
        for k, v in sd.items():
            nk = k.replace("encoder.block.", "blocks.")
            nk = nk.replace(".layer.0.SelfAttention.", ".attn.")
            nk = nk.replace(".layer.0.layer_norm.", ".attn_norm.")
            nk = nk.replace(".layer.1.DenseReluDense.wi_0.", ".wi_0.")
            nk = nk.replace(".layer.1.DenseReluDense.wi_1.", ".wi_1.")
            nk = nk.replace(".layer.1.DenseReluDense.wo.", ".wo.")
            nk = nk.replace(".layer.1.layer_norm.", ".ff_norm.")
            nk = nk.replace("encoder.final_layer_norm.", "final_layer_norm.")
            nk = nk.replace("encoder.embed_tokens.", "shared.")
            out[nk] = v
        return out
