"""Qwen3: Llama architecture + per-head PRE_ROPE q/k RMSNorm and an explicit
head_dim (ref `models/qwen3/`, qk-norm noted in SURVEY §2.11)."""

from __future__ import annotations

import torch.nn as nn

from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
    LlamaDecoderLayer,
    LlamaForCausalLM,
    LlamaInferenceConfig,
    LlamaModel,
)
from neuronx_distributed_inference_amd.modules.norm import RMSNorm


class Qwen3InferenceConfig(LlamaInferenceConfig):
    def add_derived_config(self):
        super().add_derived_config()
        if not hasattr(self, "head_dim"):
            self.head_dim = self.hidden_size // self.num_attention_heads
        if not hasattr(self, "attention_bias"):
            self.attention_bias = False


class Qwen3DecoderLayer(LlamaDecoderLayer):
    def __init__(self, config: InferenceConfig, layer_idx: int):
        super().__init__(config, layer_idx)
        ec = config.neuron_config
        hd = config.head_dim
        # per-head qk-norm applied before RoPE (PRE_ROPE)
        self.self_attn.q_norm = RMSNorm(hd, eps=config.rms_norm_eps, dtype=ec.torch_dtype)
        self.self_attn.k_norm = RMSNorm(hd, eps=config.rms_norm_eps, dtype=ec.torch_dtype)


class Qwen3Model(LlamaModel):
    def build_layers(self, config):
        return nn.ModuleList(
            Qwen3DecoderLayer(config, i) for i in range(config.num_hidden_layers)
        )


class Qwen3ForCausalLM(LlamaForCausalLM):
    _model_cls = Qwen3Model
    _config_cls = Qwen3InferenceConfig

    @staticmethod
    def convert_hf_to_native_state_dict(sd, config):
        out = LlamaForCausalLM.convert_hf_to_native_state_dict(sd, config)
        # q_norm/k_norm live under self_attn.{q,k}_norm in HF too; the llama
        # key rewrite leaves them in place
        return out
