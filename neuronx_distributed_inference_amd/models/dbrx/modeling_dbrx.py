"""DBRX 132B MoE (ref `models/dbrx/modeling_dbrx.py`, 300 LoC): fused-Wqkv
attention with clip_qkv, bias-free LayerNorm, 16-expert top-4 MoE with
stacked expert weights in the HF checkpoint."""

from __future__ import annotations

from typing import Dict, List

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.models.llama.modeling_llama import LlamaForCausalLM
from neuronx_distributed_inference_amd.models.model_base import DecoderModel
from neuronx_distributed_inference_amd.modules.attention.attention_base import AttentionBase
from neuronx_distributed_inference_amd.modules.moe import MoELayer
from neuronx_distributed_inference_amd.modules.norm import LayerNorm


class DbrxInferenceConfig(InferenceConfig):
    attribute_map = {
        "d_model": "hidden_size",
        "n_heads": "num_attention_heads",
        "n_layers": "num_hidden_layers",
        "max_seq_len": "max_position_embeddings",
    }

    def get_required_attributes(self) -> List[str]:
        return ["hidden_size", "num_attention_heads", "num_hidden_layers", "vocab_size"]

    def add_derived_config(self):
        attn = getattr(self, "attn_config", {}) or {}
        ffn = getattr(self, "ffn_config", {}) or {}
        if isinstance(attn, dict):
            self.num_key_value_heads = attn.get("kv_n_heads", self.num_attention_heads)
            self.clip_qkv = attn.get("clip_qkv", None)
            self.rope_theta = attn.get("rope_theta", 500000.0)
        if isinstance(ffn, dict):
            self.intermediate_size = ffn.get("ffn_hidden_size", 4 * self.hidden_size)
            self.num_experts = ffn.get("moe_num_experts", 16)
            self.num_experts_per_tok = ffn.get("moe_top_k", 4)
        if not hasattr(self, "hidden_act"):
            self.hidden_act = "silu"
        if not hasattr(self, "tie_word_embeddings"):
            self.tie_word_embeddings = False
        if not hasattr(self, "rms_norm_eps"):
            self.rms_norm_eps = 1e-5  # unused (LayerNorm) but base hooks read it


class DbrxDecoderLayer(nn.Module):
    def __init__(self, config: InferenceConfig, layer_idx: int):
        super().__init__()
        ec = config.neuron_config
        self.self_attn = AttentionBase(config, layer_idx)
        self.ffn = MoELayer(config, hidden_act="silu", normalize_top_k=True)
        self.norm_1 = LayerNorm(config.hidden_size, bias=False, dtype=ec.torch_dtype)
        self.norm_2 = LayerNorm(config.hidden_size, bias=False, dtype=ec.torch_dtype)

    def forward(self, h, residual, cos, sin, kv_mgr, position_ids, phase,
                attention_mask, seq_ids):
        if residual is None:
            residual = h
            h = self.norm_1(h)
        else:
            h, residual = self.norm_1(h, residual)
        h = self.self_attn(h, cos, sin, kv_mgr, position_ids, phase, attention_mask, seq_ids)
        h, residual = self.norm_2(h, residual)
        h = self.ffn(h)
        return h, residual


class DbrxModel(DecoderModel):
    def build_layers(self, config):
        return nn.ModuleList(
            DbrxDecoderLayer(config, i) for i in range(config.num_hidden_layers)
        )

    def build_norm(self, config):
        return LayerNorm(config.hidden_size, bias=False, dtype=self.dtype)


class DbrxForCausalLM(LlamaForCausalLM):
    _model_cls = DbrxModel
    _config_cls = DbrxInferenceConfig

    @staticmethod
    def convert_hf_to_native_state_dict(
        sd: Dict[str, torch.Tensor], config: InferenceConfig
    ) -> Dict[str, torch.Tensor]:
        """HF DBRX layout: transformer.blocks.N.{norm_attn_norm.{norm_1,norm_2,
        attn.{Wqkv,out_proj}}, ffn.{router.layer, experts.mlp.{w1,v1,w2}}};
        Wqkv is fused (split here), expert weights are stacked flat
        (E*I, H) / (E*I, H) / transposed for w2."""
        hs = config.hidden_size
        hd = hs // config.num_attention_heads
        q_rows = config.num_attention_heads * hd
        kv_rows = config.num_key_value_heads * hd
        E = config.num_experts
        inter = config.intermediate_size
        out: Dict[str, torch.Tensor] = {}
        for k, v in sd.items():
            nk = k
            if nk.startswith("transformer."):
                nk = nk[len("transformer."):]
            nk = nk.replace("blocks.", "layers.")
            nk = nk.replace("wte.weight", "embed_tokens.weight")
            nk = nk.replace("norm_f.", "norm.")
            if ".norm_attn_norm.attn.Wqkv.weight" in nk:
                base = nk.split(".norm_attn_norm")[0]
                q, kk, vv = torch.split(v, [q_rows, kv_rows, kv_rows], dim=0)
                out[f"{base}.self_attn.qkv_proj.q_proj.weight"] = q
                out[f"{base}.self_attn.qkv_proj.k_proj.weight"] = kk
                out[f"{base}.self_attn.qkv_proj.v_proj.weight"] = vv
                continue
            if ".norm_attn_norm.attn.out_proj.weight" in nk:
                out[nk.split(".norm_attn_norm")[0] + ".self_attn.o_proj.o_proj.weight"] = v
                continue
            nk = nk.replace("norm_attn_norm.norm_1", "norm_1")
            nk = nk.replace("norm_attn_norm.norm_2", "norm_2")
            if ".ffn.router.layer.weight" in nk:
                out[nk.replace(".ffn.router.layer.weight", ".ffn.router.weight")] = v
                continue
            if ".ffn.experts.mlp." in nk:
                base = nk.split(".ffn.experts.mlp.")[0]
                wname = nk.rsplit(".", 1)[-1]
                # w1 (gate) and v1 (up): (E*I, H) -> (E, I, H);
                # w2 (down): (E*I, H) stored transposed -> (E, H, I)
                if wname in ("w1", "v1"):
                    t = v.view(E, inter, hs)
                    tgt = "w_gate" if wname == "w1" else "w_up"
                    out[f"{base}.ffn.experts.{tgt}"] = t
                elif wname == "w2":
                    out[f"{base}.ffn.experts.w_down"] = (
                        v.view(E, inter, hs).transpose(1, 2).contiguous()
                    )
                continue
            out[nk] = v
        dtype = config.neuron_config.torch_dtype
        return {k: v.to(dtype) for k, v in out.items()}
