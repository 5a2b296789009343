"""Mixtral 8x7B / 8x22B: Llama attention + top-2 MoE MLP
(ref `models/mixtral/` via `modules/moe_v2.py` initialize_moe_module)."""

from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
    LlamaForCausalLM,
    LlamaInferenceConfig,
)
from neuronx_distributed_inference_amd.models.model_base import DecoderModel
from neuronx_distributed_inference_amd.modules.attention.attention_base import AttentionBase
from neuronx_distributed_inference_amd.modules.moe import MoELayer
from neuronx_distributed_inference_amd.modules.norm import RMSNorm


class MixtralInferenceConfig(LlamaInferenceConfig):
    def get_required_attributes(self):
        return super().get_required_attributes() + ["num_local_experts", "num_experts_per_tok"]


class MixtralDecoderLayer(nn.Module):
    def __init__(self, config: InferenceConfig, layer_idx: int):
        super().__init__()
        ec = config.neuron_config
        self.self_attn = AttentionBase(config, layer_idx)
        self.mlp = MoELayer(config, hidden_act=getattr(config, "hidden_act", "silu"))
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


class MixtralModel(DecoderModel):
    def build_layers(self, config):
        return nn.ModuleList(
            MixtralDecoderLayer(config, i) for i in range(config.num_hidden_layers)
        )


class MixtralForCausalLM(LlamaForCausalLM):
    _model_cls = MixtralModel
    _config_cls = MixtralInferenceConfig

    @staticmethod
    def convert_hf_to_native_state_dict(
        sd: Dict[str, torch.Tensor], config: InferenceConfig
    ) -> Dict[str, torch.Tensor]:
        """Stack per-expert w1/w3/w2 into the (E, ...) tensors
        (ref expert state-dict conversion, modeling_dbrx.py:51-114 pattern)."""
        E = config.num_local_experts
        inter = config.intermediate_size
        out: Dict[str, torch.Tensor] = {}
        stacked: Dict[str, list] = {}
        for k, v in sd.items():
            nk = k[len("model."):] if k.startswith("model.") else k
            nk = nk.replace("self_attn.q_proj", "self_attn.qkv_proj.q_proj")
            nk = nk.replace("self_attn.k_proj", "self_attn.qkv_proj.k_proj")
            nk = nk.replace("self_attn.v_proj", "self_attn.qkv_proj.v_proj")
            nk = nk.replace("self_attn.o_proj", "self_attn.o_proj.o_proj")
            nk = nk.replace(".block_sparse_moe.", ".mlp.")  # legacy naming
            if ".mlp.gate.weight" in nk:
                out[nk.replace(".gate.weight", ".router.weight")] = v
                continue
            # transformers >= 5 layout: stacked 3D expert tensors
            if nk.endswith(".mlp.experts.gate_up_proj"):
                base = nk[: -len(".gate_up_proj")]
                g, u = v.split([inter, inter], dim=1)
                out[f"{base}.w_gate"] = g.contiguous()
                out[f"{base}.w_up"] = u.contiguous()
                continue
            if nk.endswith(".mlp.experts.down_proj"):
                out[nk.replace(".down_proj", ".w_down")] = v
                continue
            # legacy layout: per-expert w1/w2/w3 modules
            if ".mlp.experts." in nk:
                parts = nk.split(".")
                e = int(parts[parts.index("experts") + 1])
                wname = parts[-2]  # w1 / w2 / w3
                base = ".".join(parts[: parts.index("experts")])
                tgt = {"w1": "w_gate", "w3": "w_up", "w2": "w_down"}[wname]
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
