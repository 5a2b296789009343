"""Gemma3 (text) for MI355X (ref `models/gemma3/`):
sandwich norms (input/post-attention + pre/post-feedforward GemmaRMSNorm),
per-head Gemma qk-norms, scaled word embeddings (sqrt(hidden)), mixed
sliding/full attention layers with per-layer-type RoPE bases, and
query_pre_attn_scalar attention scaling."""

from __future__ import annotations

import math
from typing import Dict, List

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.models.llama.modeling_llama import LlamaForCausalLM
from neuronx_distributed_inference_amd.models.model_base import DecoderModel
from neuronx_distributed_inference_amd.modules.attention.attention_base import AttentionBase
from neuronx_distributed_inference_amd.modules.mlp import GatedMLP
from neuronx_distributed_inference_amd.modules.norm import GemmaRMSNorm
from neuronx_distributed_inference_amd.modules.rope import RotaryEmbedding
from neuronx_distributed_inference_amd.parallel.layers import ParallelEmbedding


class Gemma3InferenceConfig(InferenceConfig):
    attribute_map = {"hidden_activation": "hidden_act"}

    def get_required_attributes(self) -> List[str]:
        return ["hidden_size", "num_attention_heads", "num_hidden_layers",
                "num_key_value_heads", "vocab_size", "intermediate_size", "head_dim"]

    def add_derived_config(self):
        if not hasattr(self, "rms_norm_eps"):
            self.rms_norm_eps = 1e-6
        if not hasattr(self, "hidden_act"):
            self.hidden_act = "gelu_pytorch_tanh"
        if not hasattr(self, "tie_word_embeddings"):
            self.tie_word_embeddings = True
        if not hasattr(self, "query_pre_attn_scalar"):
            self.query_pre_attn_scalar = self.head_dim
        if not hasattr(self, "layer_types"):
            pattern = getattr(self, "sliding_window_pattern", 6)
            self.layer_types = [
                "full_attention" if (i + 1) % pattern == 0 else "sliding_attention"
                for i in range(self.num_hidden_layers)
            ]
        if not hasattr(self, "rope_parameters") or self.rope_parameters is None:
            self.rope_parameters = {
                "sliding_attention": {"rope_type": "default",
                                      "rope_theta": getattr(self, "rope_local_base_freq", 10000.0)},
                "full_attention": {"rope_type": "default",
                                   "rope_theta": getattr(self, "rope_theta", 1000000.0)},
            }


class ScaledParallelEmbedding(ParallelEmbedding):
    """Gemma embeddings are scaled by sqrt(hidden_size)."""

    def __init__(self, *a, embed_scale: float = 1.0, **kw):
        super().__init__(*a, **kw)
        self.embed_scale = embed_scale

    def forward(self, ids, sequence_parallel: bool = False):
        y = super().forward(ids, sequence_parallel=sequence_parallel) \
            if sequence_parallel else super().forward(ids)
        return y * torch.tensor(self.embed_scale, dtype=y.dtype)


class Gemma3DecoderLayer(nn.Module):
    def __init__(self, config: InferenceConfig, layer_idx: int):
        super().__init__()
        ec = config.neuron_config
        layer_type = config.layer_types[layer_idx]
        is_sliding = layer_type == "sliding_attention"
        self.self_attn = AttentionBase(config, layer_idx)
        self.self_attn.sliding_window = config.sliding_window if is_sliding else 0
        self.self_attn.scale = float(config.query_pre_attn_scalar) ** -0.5
        hd = config.head_dim
        self.self_attn.q_norm = GemmaRMSNorm(hd, config.rms_norm_eps, dtype=ec.torch_dtype)
        self.self_attn.k_norm = GemmaRMSNorm(hd, config.rms_norm_eps, dtype=ec.torch_dtype)
        rp = config.rope_parameters[layer_type]
        self.rotary = RotaryEmbedding(
            hd,
            max_position_embeddings=getattr(config, "max_position_embeddings", 4096),
            base=rp.get("rope_theta", 10000.0),
            scaling=rp if rp.get("rope_type", "default") != "default" else None,
        )
        self.mlp = GatedMLP(
            config.hidden_size, config.intermediate_size, tp_degree=ec.tp_degree,
            dtype=ec.torch_dtype, hidden_act=config.hidden_act,
            fused_gate_up=ec.fused_gate_up,
        )
        mk = lambda: GemmaRMSNorm(config.hidden_size, config.rms_norm_eps, dtype=ec.torch_dtype)
        self.input_layernorm = mk()
        self.post_attention_layernorm = mk()
        self.pre_feedforward_layernorm = mk()
        self.post_feedforward_layernorm = mk()

    def forward(self, h, residual, cos, sin, kv_mgr, position_ids, phase,
                attention_mask, seq_ids, sequence_parallel=False):
        # sandwich-norm structure; the (h, residual) fusion contract is
        # collapsed (residual always returned as None)
        assert residual is None
        cos_l, sin_l = self.rotary.get_cos_sin(position_ids, device=h.device)
        r = h
        x = self.input_layernorm(h)
        x = self.self_attn(x, cos_l, sin_l, kv_mgr, position_ids, phase,
                           attention_mask, seq_ids)
        x = self.post_attention_layernorm(x)
        h = r + x
        r = h
        x = self.pre_feedforward_layernorm(h)
        x = self.mlp(x)
        x = self.post_feedforward_layernorm(x)
        return r + x, None


class Gemma3Model(DecoderModel):
    def __init__(self, config):
        super().__init__(config)
        ec = config.neuron_config
        self.embed_tokens = ScaledParallelEmbedding(
            config.vocab_size, config.hidden_size, dtype=ec.torch_dtype,
            tp_degree=ec.tp_degree, embed_scale=math.sqrt(config.hidden_size),
        )

    def build_layers(self, config):
        return nn.ModuleList(
            Gemma3DecoderLayer(config, i) for i in range(config.num_hidden_layers)
        )

    def build_norm(self, config):
        return GemmaRMSNorm(config.hidden_size, config.rms_norm_eps, dtype=self.dtype)


class Gemma3ForCausalLM(LlamaForCausalLM):
    _model_cls = Gemma3Model
    _config_cls = Gemma3InferenceConfig
