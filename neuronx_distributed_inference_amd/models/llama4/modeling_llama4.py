"""Llama4 (Scout/Maverick) text model.

Parity target: the reference's `models/llama4/` text stack (chunked local
attention + NoPE global layers + interleaved MoE). The
Llama4-specific behaviors, matched against HF `modeling_llama4.py`:

- complex-pair (interleaved) RoPE on "local" layers; every
  `no_rope_layers[i] == 0` layer is NoPE (no rotary at all);
- chunked attention on rope layers (`attention_chunk_size`): a query
  attends causally only within its own chunk — mapped onto the HIP
  attention kernels' negative-sliding-window mode;
- weightless L2 qk-norm AFTER rope on rope layers (`use_qk_norm`);
- log-temperature query scaling on NoPE layers
  (`attn_temperature_tuning`, floor_scale/attn_scale);
- MoE on `moe_layers` with top-1 sigmoid routing where the gate scales
  the expert INPUT (y = E(g*x)), plus an always-on shared expert; dense
  `intermediate_size_mlp` MLPs on the remaining layers.

The vision tower is not implemented (GAPS.md: multimodal).
"""

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
from neuronx_distributed_inference_amd.modules.mlp import GatedMLP
from neuronx_distributed_inference_amd.modules.moe import MoELayer
from neuronx_distributed_inference_amd.modules.norm import RMSNorm
from neuronx_distributed_inference_amd.parallel import mappings


class Llama4InferenceConfig(LlamaInferenceConfig):
    def get_required_attributes(self):
        return super().get_required_attributes() + ["num_local_experts"]

    def add_derived_config(self):
        super().add_derived_config()
        # HF Llama4 keeps rope params nested in rope_parameters/rope_scaling
        rp = getattr(self, "rope_parameters", None) or getattr(self, "rope_scaling", None) or {}
        if isinstance(rp, dict) and rp:
            self.rope_theta = rp.get("rope_theta", getattr(self, "rope_theta", 500000.0))
            self.rope_scaling = rp if rp.get("rope_type", "default") != "default" else None
        n = self.num_hidden_layers
        if not hasattr(self, "no_rope_layers") or self.no_rope_layers is None:
            # HF default: every 4th layer is NoPE (1 = rope)
            interval = getattr(self, "no_rope_layer_interval", 4)
            self.no_rope_layers = [int((i + 1) % interval != 0) for i in range(n)]
        if not hasattr(self, "moe_layers") or self.moe_layers is None:
            step = getattr(self, "interleave_moe_layer_step", 1)
            self.moe_layers = [i for i in range(n) if (i + 1) % step == 0]
        for name, default in (
            ("attention_chunk_size", 8192),
            ("use_qk_norm", True),
            ("attn_temperature_tuning", True),
            ("floor_scale", 8192.0),
            ("attn_scale", 0.1),
            ("num_experts_per_tok", 1),
            ("intermediate_size_mlp", self.intermediate_size),
            ("attention_bias", False),
        ):
            if not hasattr(self, name):
                setattr(self, name, default)


class L2Norm(nn.Module):
    """Weightless RMS normalization (HF Llama4TextL2Norm)."""

    def __init__(self, eps: float = 1e-6):
        super().__init__()
        self.eps = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        xf = x.float()
        return (xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + self.eps)).type_as(x)


class Llama4Attention(AttentionBase):
    def __init__(self, config: InferenceConfig, layer_idx: int):
        super().__init__(config, layer_idx)
        self.rope_interleaved = True
        self.use_rope = bool(config.no_rope_layers[layer_idx])
        if self.use_rope:
            if config.use_qk_norm:
                self.qk_norm_post_rope = L2Norm(config.rms_norm_eps)
            # chunked attention: negative window = chunk mode in the kernels
            self.sliding_window = -int(config.attention_chunk_size)
        elif config.attn_temperature_tuning:
            floor = float(config.floor_scale)
            scale = float(config.attn_scale)

            def _temp_scale(position_ids: torch.Tensor) -> torch.Tensor:
                p = position_ids.float()
                return torch.log1p(torch.floor((p + 1.0) / floor)) * scale + 1.0

            self.q_scale_fn = _temp_scale


class Llama4Moe(nn.Module):
    """Top-1 sigmoid router (input-scaled experts) + shared expert
    (HF Llama4TextMoe; ref llama4 MoE layers)."""

    def __init__(self, config: InferenceConfig):
        super().__init__()
        ec = config.neuron_config
        inner = MoELayer(config, hidden_act=getattr(config, "hidden_act", "silu"),
                         router_act="sigmoid", scale_input=True)
        self.router = inner.router
        self.experts = inner.experts
        self.shared_expert = GatedMLP(
            config.hidden_size, config.intermediate_size,
            tp_degree=ec.tp_degree, dtype=ec.torch_dtype,
            hidden_act=getattr(config, "hidden_act", "silu"),
        )

    def forward(self, h: torch.Tensor) -> torch.Tensor:
        B, S, H = h.shape
        flat = h.view(-1, H)
        weights, idx = self.router(flat)
        out = self.experts(flat, weights, idx)
        out = mappings.reduce_from_tensor_parallel_region(out).view(B, S, H)
        return out + self.shared_expert(h)


class Llama4DecoderLayer(nn.Module):
    def __init__(self, config: InferenceConfig, layer_idx: int):
        super().__init__()
        ec = config.neuron_config
        self.self_attn = Llama4Attention(config, layer_idx)
        if layer_idx in set(config.moe_layers):
            self.mlp = Llama4Moe(config)
        else:
            self.mlp = GatedMLP(
                config.hidden_size, config.intermediate_size_mlp,
                tp_degree=ec.tp_degree, dtype=ec.torch_dtype,
                hidden_act=getattr(config, "hidden_act", "silu"),
                fused_gate_up=ec.fused_gate_up,
            )
        self.input_layernorm = RMSNorm(config.hidden_size, config.rms_norm_eps,
                                       dtype=ec.torch_dtype)
        self.post_attention_layernorm = RMSNorm(config.hidden_size, config.rms_norm_eps,
                                                dtype=ec.torch_dtype)

    def forward(self, h, residual, cos, sin, kv_mgr, position_ids, phase,
                attention_mask, seq_ids):
        if residual is None:
            residual = h
            h = self.input_layernorm(h)
        else:
            h, residual = self.input_layernorm(h, residual)
        h = self.self_attn(h, cos, sin, kv_mgr, position_ids, phase,
                           attention_mask, seq_ids)
        h, residual = self.post_attention_layernorm(h, residual)
        h = self.mlp(h)
        return h, residual


class Llama4Model(DecoderModel):
    def build_layers(self, config):
        return nn.ModuleList(
            Llama4DecoderLayer(config, i) for i in range(config.num_hidden_layers)
        )


class Llama4ForCausalLM(LlamaForCausalLM):
    _model_cls = Llama4Model
    _config_cls = Llama4InferenceConfig

    @staticmethod
    def convert_hf_to_native_state_dict(
        sd: Dict[str, torch.Tensor], config: InferenceConfig
    ) -> Dict[str, torch.Tensor]:
        """HF Llama4 text checkpoint -> native names. Expert tensors come
        stacked in bmm layout: gate_up_proj (E, H, 2I) -> w_gate/w_up
        (E, I, H); down_proj (E, I, H) -> w_down (E, H, I)."""
        out: Dict[str, torch.Tensor] = {}
        for k, v in sd.items():
            nk = k
            for pre in ("language_model.model.", "model.", "language_model."):
                if nk.startswith(pre):
                    nk = nk[len(pre):]
                    break
            nk = nk.replace("self_attn.q_proj", "self_attn.qkv_proj.q_proj")
            nk = nk.replace("self_attn.k_proj", "self_attn.qkv_proj.k_proj")
            nk = nk.replace("self_attn.v_proj", "self_attn.qkv_proj.v_proj")
            nk = nk.replace("self_attn.o_proj", "self_attn.o_proj.o_proj")
            nk = nk.replace(".feed_forward.", ".mlp.")
            if nk.endswith(".mlp.experts.gate_up_proj"):
                base = nk[: -len(".gate_up_proj")]
                I = v.shape[-1] // 2
                out[f"{base}.w_gate"] = v[..., :I].transpose(1, 2).contiguous()
                out[f"{base}.w_up"] = v[..., I:].transpose(1, 2).contiguous()
                continue
            if nk.endswith(".mlp.experts.down_proj"):
                out[nk.replace(".down_proj", ".w_down")] = v.transpose(1, 2).contiguous()
                continue
            out[nk] = v
        if "lm_head.weight" not in out and getattr(config, "tie_word_embeddings", False):
            out["lm_head.weight"] = out["embed_tokens.weight"]
        dtype = config.neuron_config.torch_dtype
        return {k: v.to(dtype) for k, v in out.items()}
