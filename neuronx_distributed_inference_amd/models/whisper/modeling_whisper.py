"""Whisper encoder-decoder speech-to-text.

Parity target: the reference's `models/whisper/` (encoder-decoder via
`encoder_base.py`). MI355X-native layout:

- the mel encoder (2x Conv1d + GELU + bidirectional transformer) runs once
  per request, eagerly — it is prefill-shaped, library GEMMs;
- the decoder is the serving hot path: TP-sharded attention with a linear
  self-attention KV cache written by the HIP scatter kernel and read by the
  split-KV decode kernel; cross-attention K/V are projected ONCE from the
  encoder output and decode steps run the same HIP decode kernel against
  them (a static full-length "cache");
- LayerNorm pre-norm blocks, learned decoder positions, erf-GELU MLPs
  (HF `activation_function="gelu"` is the exact erf form).

Biases follow the HF checkpoint: q/v/out biased, k un-biased.
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.parallel.layers import (
    ColumnParallelLinear,
    RowParallelLinear,
)


class WhisperInferenceConfig(InferenceConfig):
    def get_required_attributes(self) -> List[str]:
        return ["vocab_size", "num_mel_bins", "d_model", "encoder_layers",
                "decoder_layers", "encoder_attention_heads",
                "decoder_attention_heads", "encoder_ffn_dim", "decoder_ffn_dim",
                "max_source_positions", "max_target_positions"]

    def add_derived_config(self):
        if not hasattr(self, "decoder_start_token_id"):
            self.decoder_start_token_id = 50257
        if not hasattr(self, "eos_token_id"):
            self.eos_token_id = 50256
        # satisfy generic plumbing that inspects decoder geometry
        self.hidden_size = self.d_model
        self.num_attention_heads = self.decoder_attention_heads
        self.num_key_value_heads = self.decoder_attention_heads
        self.num_hidden_layers = self.decoder_layers


class _WhisperAttention(nn.Module):
    """TP multi-head attention with Whisper's bias pattern."""

    def __init__(self, d_model: int, heads: int, tp: int, dtype: torch.dtype):
        super().__init__()
        self.heads = heads
        self.head_dim = d_model // heads
        self.tp = tp
        self.heads_per_rank = heads // tp
        self.scale = 1.0 / math.sqrt(self.head_dim)
        self.q_proj = ColumnParallelLinear(d_model, d_model, bias=True, dtype=dtype, tp_degree=tp)
        self.k_proj = ColumnParallelLinear(d_model, d_model, bias=False, dtype=dtype, tp_degree=tp)
        self.v_proj = ColumnParallelLinear(d_model, d_model, bias=True, dtype=dtype, tp_degree=tp)
        self.out_proj = RowParallelLinear(d_model, d_model, bias=True, dtype=dtype, tp_degree=tp)

    def _shape(self, x: torch.Tensor) -> torch.Tensor:
        B, S, _ = x.shape
        return x.view(B, S, self.heads_per_rank, self.head_dim).transpose(1, 2)

    def project_kv(self, x: torch.Tensor):
        return self._shape(self.k_proj(x)), self._shape(self.v_proj(x))

    def forward_full(self, x: torch.Tensor, causal: bool) -> torch.Tensor:
        """Encoder self-attention (bidirectional) / decoder prefill."""
        B, S, _ = x.shape
        q, k, v = self._shape(self.q_proj(x)), *self.project_kv(x)
        if causal:
            out = ops.attention_prefill(q, k, v, causal=True, scale=self.scale)
        else:
            mask = torch.ones(1, 1, S, S, dtype=torch.bool, device=x.device)
            out = ops.attention_prefill(q, k, v, mask=mask, causal=False, scale=self.scale)
        return self.out_proj(out.transpose(1, 2).reshape(B, S, -1))

    def forward_cached(self, x: torch.Tensor, k_cache: torch.Tensor,
                       v_cache: torch.Tensor, eff_lens: torch.Tensor) -> torch.Tensor:
        """Decode-step attention against a (B,H,Smax,D) cache — both the
        causal self cache and the static cross-attention K/V."""
        B, S, _ = x.shape
        q = self._shape(self.q_proj(x))
        out = ops.attention_decode(q, k_cache, v_cache, eff_lens, scale=self.scale)
        return self.out_proj(out.transpose(1, 2).reshape(B, S, -1))


class _EncoderLayer(nn.Module):
    def __init__(self, config, tp: int, dtype: torch.dtype):
        super().__init__()
        d = config.d_model
        self.self_attn = _WhisperAttention(d, config.encoder_attention_heads, tp, dtype)
        self.self_attn_layer_norm = nn.LayerNorm(d)
        self.fc1 = ColumnParallelLinear(d, config.encoder_ffn_dim, bias=True, dtype=dtype, tp_degree=tp)
        self.fc2 = RowParallelLinear(config.encoder_ffn_dim, d, bias=True, dtype=dtype, tp_degree=tp)
        self.final_layer_norm = nn.LayerNorm(d)

    def forward(self, h: torch.Tensor) -> torch.Tensor:
        h = h + self.self_attn.forward_full(self.self_attn_layer_norm(h), causal=False)
        return h + self.fc2(F.gelu(self.fc1(self.final_layer_norm(h))))


class WhisperEncoder(nn.Module):
    def __init__(self, config, tp: int, dtype: torch.dtype):
        super().__init__()
        d = config.d_model
        self.conv1 = nn.Conv1d(config.num_mel_bins, d, kernel_size=3, padding=1)
        self.conv2 = nn.Conv1d(d, d, kernel_size=3, stride=2, padding=1)
        self.embed_positions = nn.Embedding(config.max_source_positions, d)
        self.layers = nn.ModuleList(
            _EncoderLayer(config, tp, dtype) for _ in range(config.encoder_layers)
        )
        self.layer_norm = nn.LayerNorm(d)

    def forward(self, input_features: torch.Tensor) -> torch.Tensor:
        """input_features: (B, num_mel_bins, T) -> (B, T//2, d_model)."""
        h = F.gelu(self.conv1(input_features))
        h = F.gelu(self.conv2(h)).permute(0, 2, 1)
        h = h + self.embed_positions.weight[: h.shape[1]]
        for layer in self.layers:
            h = layer(h)
        return self.layer_norm(h)


class _DecoderLayer(nn.Module):
    def __init__(self, config, tp: int, dtype: torch.dtype):
        super().__init__()
        d = config.d_model
        self.self_attn = _WhisperAttention(d, config.decoder_attention_heads, tp, dtype)
        self.self_attn_layer_norm = nn.LayerNorm(d)
        self.encoder_attn = _WhisperAttention(d, config.decoder_attention_heads, tp, dtype)
        self.encoder_attn_layer_norm = nn.LayerNorm(d)
        self.fc1 = ColumnParallelLinear(d, config.decoder_ffn_dim, bias=True, dtype=dtype, tp_degree=tp)
        self.fc2 = RowParallelLinear(config.decoder_ffn_dim, d, bias=True, dtype=dtype, tp_degree=tp)
        self.final_layer_norm = nn.LayerNorm(d)


class WhisperDecoder(nn.Module):
    """Decoder with a linear self-attn KV cache (HIP scatter + split-KV
    decode) and per-layer static cross-attention K/V."""

    def __init__(self, config: InferenceConfig):
        super().__init__()
        ec = config.neuron_config
        tp, dtype = ec.tp_degree, ec.torch_dtype
        d = config.d_model
        self.config = config
        self.embed_tokens = nn.Embedding(config.vocab_size, d)
        self.embed_positions = nn.Embedding(config.max_target_positions, d)
        self.layers = nn.ModuleList(
            _DecoderLayer(config, tp, dtype) for _ in range(config.decoder_layers)
        )
        self.layer_norm = nn.LayerNorm(d)
        self._self_kv: List = []
        self._cross_kv: List = []

    def start_request(self, encoder_out: torch.Tensor, max_len: int):
        """Project cross K/V once; allocate the self-attn cache."""
        B = encoder_out.shape[0]
        self._cross_kv = []
        self._self_kv = []
        for layer in self.layers:
            self._cross_kv.append(layer.encoder_attn.project_kv(encoder_out))
            h = layer.self_attn.heads_per_rank
            hd = layer.self_attn.head_dim
            kc = torch.zeros(B, h, max_len, hd, dtype=encoder_out.dtype,
                             device=encoder_out.device)
            self._self_kv.append((kc, torch.zeros_like(kc)))

    def forward(self, ids: torch.Tensor, positions: torch.Tensor,
                enc_len: torch.Tensor) -> torch.Tensor:
        """One decode chunk. ids/positions: (B, S); enc_len: (B,) encoder
        lengths for cross attention. Returns (B, S, d_model)."""
        B, S = ids.shape
        h = self.embed_tokens(ids) + self.embed_positions(positions.clamp(min=0))
        pos32 = positions.to(torch.int32)
        seq_ids = torch.arange(B, device=ids.device, dtype=torch.int32)
        for li, layer in enumerate(self.layers):
            kc, vc = self._self_kv[li]
            x = layer.self_attn_layer_norm(h)
            k_new, v_new = layer.self_attn.project_kv(x)
            ops.kv_cache_update(kc, vc, k_new, v_new, pos32, seq_ids)
            h = h + layer.self_attn.forward_cached(x, kc, vc, pos32)
            x = layer.encoder_attn_layer_norm(h)
            ck, cv = self._cross_kv[li]
            cross_pos = (enc_len.view(B, 1) - 1).expand(B, S).to(torch.int32)
            h = h + layer.encoder_attn.forward_cached(x, ck, cv, cross_pos)
            h = h + layer.fc2(F.gelu(layer.fc1(layer.final_layer_norm(h))))
        return self.layer_norm(h)


class WhisperForConditionalGeneration(nn.Module):
    """compile()/load() compatible speech-to-text model with greedy
    generation (ref whisper model + encoder_base)."""

    _config_cls = WhisperInferenceConfig

    def __init__(self, config: InferenceConfig):
        super().__init__()
        ec = config.neuron_config
        self.config = config
        self.encoder = WhisperEncoder(config, ec.tp_degree, ec.torch_dtype)
        self.decoder = WhisperDecoder(config)
        self.proj_out = nn.Linear(config.d_model, config.vocab_size, bias=False)

    @torch.inference_mode()
    def generate(self, input_features: torch.Tensor,
                 max_new_tokens: int = 32,
                 decoder_input_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
        B = input_features.shape[0]
        dev = input_features.device
        enc = self.encoder(input_features)
        enc_len = torch.full((B,), enc.shape[1], device=dev, dtype=torch.long)
        if decoder_input_ids is None:
            decoder_input_ids = torch.full(
                (B, 1), self.config.decoder_start_token_id, device=dev, dtype=torch.long
            )
        prefix = decoder_input_ids.shape[1]
        self.decoder.start_request(enc, prefix + max_new_tokens)
        ids = decoder_input_ids
        out_tokens = [decoder_input_ids]
        # prefix pass (teacher tokens), then single-token steps
        pos = torch.arange(prefix, device=dev).view(1, -1).expand(B, -1)
        h = self.decoder(ids, pos, enc_len)
        next_tok = self.proj_out(h[:, -1:]).argmax(-1)
        out_tokens.append(next_tok)
        finished = torch.zeros(B, dtype=torch.bool, device=dev)
        for step in range(1, max_new_tokens):
            pos = torch.full((B, 1), prefix + step - 1, device=dev, dtype=torch.long)
            h = self.decoder(next_tok, pos, enc_len)
            next_tok = self.proj_out(h[:, -1:]).argmax(-1)
            next_tok = torch.where(finished.view(B, 1),
                                   torch.full_like(next_tok, self.config.eos_token_id),
                                   next_tok)
            out_tokens.append(next_tok)
            finished |= next_tok.view(-1) == self.config.eos_token_id
            if bool(finished.all()):
                break
        return torch.cat(out_tokens, dim=1)

    @staticmethod
    def convert_hf_to_native_state_dict(
        sd: Dict[str, torch.Tensor], config: InferenceConfig
    ) -> Dict[str, torch.Tensor]:
        out: Dict[str, torch.Tensor] = {}
        for k, v in sd.items():
            nk = k[len("model."):] if k.startswith("model.") else k
            out[nk] = v
        if "proj_out.weight" not in out:
            out["proj_out.weight"] = out["decoder.embed_tokens.weight"]
        dtype = config.neuron_config.torch_dtype
        return {k: v.to(dtype) for k, v in out.items()}
