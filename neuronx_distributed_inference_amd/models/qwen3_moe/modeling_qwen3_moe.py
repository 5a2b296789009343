"""Qwen3-MoE: Qwen3 attention (qk-norm) + top-k MoE MLP
(ref `models/qwen3_moe/`, 543 LoC)."""

from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.models.llama.modeling_llama import LlamaForCausalLM
from neuronx_distributed_inference_amd.models.model_base import DecoderModel
from neuronx_distributed_inference_amd.models.qwen3.modeling_qwen3 import (
    Qwen3InferenceConfig,
)
from neuronx_distributed_inference_amd.modules.attention.attention_base import AttentionBase
from neuronx_distributed_inference_amd.modules.moe import MoELayer
from neuronx_distributed_inference_amd.modules.norm import RMSNorm


class Qwen3MoeInferenceConfig(Qwen3InferenceConfig):
    def get_required_attributes(self):
        return super().get_required_attributes() + ["num_experts", "num_experts_per_tok"]

    def add_derived_config(self):
        super().add_derived_config()
        if not hasattr(self, "norm_topk_prob"):
            self.norm_topk_prob = True
        # HF uses num_experts (<=4.x) / num_local_experts (>=5); accept both
        if not hasattr(self, "num_experts") and hasattr(self, "num_local_experts"):
            self.num_experts = self.num_local_experts


class Qwen3MoeDecoderLayer(nn.Module):
    def __init__(self, config: InferenceConfig, layer_idx: int):
        super().__init__()
        ec = config.neuron_config
        self.self_attn = AttentionBase(config, layer_idx)
        hd = config.head_dim
        self.self_attn.q_norm = RMSNorm(hd, config.rms_norm_eps, dtype=ec.torch_dtype)
        self.self_attn.k_norm = RMSNorm(hd, config.rms_norm_eps, dtype=ec.torch_dtype)
        self.mlp = MoELayer(config, normalize_top_k=config.norm_topk_prob)
        self.input_layernorm = RMSNorm(config.hidden_size, config.rms_norm_eps, dtype=ec.torch_dtype)
        self.post_attention_layernorm = RMSNorm(config.hidden_size, config.rms_norm_eps, dtype=ec.torch_dtype)

    def forward(self, h, residual, cos, sin, kv_mgr, position_ids, phase,
                attention_mask, seq_ids):
        if residual is None:
            residual = h
            h = self.input_layernorm(h)
        else:
            h, residual = self.input_layernorm(h, residual)
        h = self.self_attn(h, cos, sin, kv_mgr, position_ids, phase, attention_mask, seq_ids)
        h, residual = self.post_attention_layernorm(h, residual)
        h = self.mlp(h)
        return h, residual


class Qwen3MoeModel(DecoderModel):
    def build_layers(self, config):
        return nn.ModuleList(
            Qwen3MoeDecoderLayer(config, i) for i in range(config.num_hidden_layers)
        )


class Qwen3MoeForCausalLM(LlamaForCausalLM):
    _model_cls = Qwen3MoeModel
    _config_cls = Qwen3MoeInferenceConfig

    @staticmethod
    def convert_hf_to_native_state_dict(
        sd: Dict[str, torch.Tensor], config: InferenceConfig
    ) -> Dict[str, torch.Tensor]:
        E = config.num_experts
        out: Dict[str, torch.Tensor] = {}
        stacked: Dict[str, list] = {}
        for k, v in sd.items():
            nk = k[len("model."):] if k.startswith("model.") else k
            nk = nk.replace("self_attn.q_proj", "self_attn.qkv_proj.q_proj")
            nk = nk.replace("self_attn.k_proj", "self_attn.qkv_proj.k_proj")
            nk = nk.replace("self_attn.v_proj", "self_attn.qkv_proj.v_proj")
            nk = nk.replace("self_attn.o_proj", "self_attn.o_proj.o_proj")
            if ".mlp.gate.weight" in nk:
                out[nk.replace(".mlp.gate.weight", ".mlp.router.weight")] = v
                continue
            # transformers >= 5: stacked 3D expert tensors
            if nk.endswith(".mlp.experts.gate_up_proj"):
                inter = getattr(config, "moe_intermediate_size", config.intermediate_size)
                base = nk[: -len(".gate_up_proj")]
                g, u = v.split([inter, inter], dim=1)
                out[f"{base}.w_gate"] = g.contiguous()
                out[f"{base}.w_up"] = u.contiguous()
                continue
            if nk.endswith(".mlp.experts.down_proj"):
                out[nk.replace(".down_proj", ".w_down")] = v
                continue
            # legacy: per-expert modules
            if ".mlp.experts." in nk:
                parts = nk.split(".")
                e = int(parts[parts.index("experts") + 1])
                wname = parts[-2]  # gate_proj / up_proj / down_proj
                base = ".".join(parts[: parts.index("experts")])
                tgt = {"gate_proj": "w_gate", "up_proj": "w_up", "down_proj": "w_down"}[wname]
                stacked.setdefault(f"{base}.experts.{tgt}", [None] * E)[e] = v
                continue
            out[nk] = v
        for key, tensors in stacked.items():
            assert all(t is not None for t in tensors), f"missing experts for {key}"
            out[key] = torch.stack(tensors, dim=0)
        if "lm_head.weight" not in out and getattr(config, "tie_word_embeddings", False):
            out["lm_head.weight"] = out["embed_tokens.weight"]
        dtype = config.neuron_config.torch_dtype
        return {k: v.to(dtype) for k, v in out.items()}
