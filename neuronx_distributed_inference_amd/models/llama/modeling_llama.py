"""Llama 2 / 3.x for MI355X.

The canonical dense model (ref `models/llama/modeling_llama.py`, 1378 LoC):
decoder layer = RMSNorm -> attention -> residual -> RMSNorm -> gated MLP ->
residual, with Llama-3 scaled RoPE handled in `modules.rope`
(ref :805-871) and HF->native state-dict key rewriting
(ref convert_hf_to_neuron_state_dict :1208-1276).

Also serves Mistral/Qwen2-style architectures through small subclasses.
"""

from __future__ import annotations

from typing import Dict, List

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.models.model_base import CausalLMBase, DecoderModel
from neuronx_distributed_inference_amd.modules.attention.attention_base import AttentionBase
from neuronx_distributed_inference_amd.modules.mlp import GatedMLP
from neuronx_distributed_inference_amd.modules.norm import RMSNorm


def fuse_qkv_keys(sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """cat q|k|v weights (and biases) into qkv_proj.qkv.* keys."""
    out = dict(sd)
    for suffix in ("weight", "bias"):
        qs = [k for k in sd if k.endswith(f"qkv_proj.q_proj.{suffix}")
              and "cte_qkv_proj" not in k and "tkg_qkv_proj" not in k]
        for qk in qs:
            base = qk[: -len(f"q_proj.{suffix}")]
            kk, vk = f"{base}k_proj.{suffix}", f"{base}v_proj.{suffix}"
            out[f"{base}qkv.{suffix}"] = torch.cat([sd[qk], sd[kk], sd[vk]], dim=0)
            for key in (qk, kk, vk):
                out.pop(key)
    return out


def fuse_gate_up_keys(sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    out = dict(sd)
    for suffix in ("weight", "bias"):
        gs = [k for k in sd if k.endswith(f"mlp.gate_proj.{suffix}")]
        for gk in gs:
            base = gk[: -len(f"gate_proj.{suffix}")]
            uk = f"{base}up_proj.{suffix}"
            out[f"{base}gate_up_proj.{suffix}"] = torch.cat([sd[gk], sd[uk]], dim=0)
            out.pop(gk)
            out.pop(uk)
    return out


class LlamaInferenceConfig(InferenceConfig):
    def get_required_attributes(self) -> List[str]:
        return [
            "hidden_size",
            "num_attention_heads",
            "num_hidden_layers",
            "num_key_value_heads",
            "vocab_size",
            "intermediate_size",
        ]

    def add_derived_config(self):
        if not hasattr(self, "rms_norm_eps"):
            self.rms_norm_eps = 1e-6
        if not hasattr(self, "rope_theta"):
            self.rope_theta = 10000.0
        if not hasattr(self, "num_key_value_heads"):
            self.num_key_value_heads = getattr(self, "num_attention_heads", None)
        if not hasattr(self, "tie_word_embeddings"):
            self.tie_word_embeddings = False


class LlamaDecoderLayer(nn.Module):
    def __init__(self, config: InferenceConfig, layer_idx: int):
        super().__init__()
        ec = config.neuron_config
        self.self_attn = AttentionBase(config, layer_idx)
        self.mlp = GatedMLP(
            config.hidden_size,
            config.intermediate_size,
            tp_degree=ec.tp_degree,
            dtype=ec.torch_dtype,
            hidden_act=getattr(config, "hidden_act", "silu"),
            bias=bool(getattr(config, "mlp_bias", False)),
            fused_gate_up=ec.fused_gate_up,
        )
        self.input_layernorm = RMSNorm(
            config.hidden_size, eps=config.rms_norm_eps, dtype=ec.torch_dtype
        )
        self.post_attention_layernorm = RMSNorm(
            config.hidden_size, eps=config.rms_norm_eps, dtype=ec.torch_dtype
        )

    def forward(self, h, residual, cos, sin, kv_mgr, position_ids, phase,
                attention_mask, seq_ids, sequence_parallel=False):
        # (h, residual) threading: the residual add is fused into the RMSNorm
        # kernel (one HBM pass instead of add + norm)
        if residual is None:
            residual = h
            h = self.input_layernorm(h)
        else:
            h, residual = self.input_layernorm(h, residual)
        h = self.self_attn(h, cos, sin, kv_mgr, position_ids, phase,
                           attention_mask, seq_ids, sequence_parallel=sequence_parallel)
        h, residual = self.post_attention_layernorm(h, residual)
        h = self.mlp(h, sequence_parallel=sequence_parallel)
        return h, residual


class LlamaModel(DecoderModel):
    def build_layers(self, config: InferenceConfig) -> nn.ModuleList:
        return nn.ModuleList(
            LlamaDecoderLayer(config, i) for i in range(config.num_hidden_layers)
        )


class LlamaForCausalLM(CausalLMBase):
    _model_cls = LlamaModel
    _config_cls = LlamaInferenceConfig

    @staticmethod
    def convert_hf_to_native_state_dict(
        sd: Dict[str, torch.Tensor], config: InferenceConfig
    ) -> Dict[str, torch.Tensor]:
        ec = config.neuron_config
        out: Dict[str, torch.Tensor] = {}
        for k, v in sd.items():
            nk = k
            if nk.startswith("model."):
                nk = nk[len("model."):]
            # attention/mlp projections keep their names but move under the
            # gqa/mlp wrappers
            nk = nk.replace("self_attn.q_proj", "self_attn.qkv_proj.q_proj")
            nk = nk.replace("self_attn.k_proj", "self_attn.qkv_proj.k_proj")
            nk = nk.replace("self_attn.v_proj", "self_attn.qkv_proj.v_proj")
            nk = nk.replace("self_attn.o_proj", "self_attn.o_proj.o_proj")
            out[nk] = v
        if ec.cp_degree > 1:
            # duplicate the attention projections for the CP prefill set
            # (same full weights, sharded over the prefill-TP subgroup)
            for k in list(out):
                if ".self_attn.qkv_proj." in k:
                    out[k.replace(".qkv_proj.", ".cte_qkv_proj.")] = out[k]
                elif ".self_attn.o_proj.o_proj." in k:
                    out[k.replace(".o_proj.o_proj.", ".cte_o_proj.o_proj.")] = out[k]
        if ec.attention_dp_degree > 1:
            for k in list(out):
                if ".self_attn.qkv_proj." in k and "cte_qkv_proj" not in k:
                    out[k.replace(".qkv_proj.", ".tkg_qkv_proj.")] = out[k]
                elif ".self_attn.o_proj.o_proj." in k:
                    out[k.replace(".o_proj.o_proj.", ".tkg_o_proj.o_proj.")] = out[k]
        if ec.fused_qkv:
            out = fuse_qkv_keys(out)
        if ec.fused_gate_up:
            out = fuse_gate_up_keys(out)
        if "lm_head.weight" not in out and getattr(config, "tie_word_embeddings", False):
            out["lm_head.weight"] = out["embed_tokens.weight"]
        dtype = config.neuron_config.torch_dtype
        return {k: v.to(dtype) for k, v in out.items()}
