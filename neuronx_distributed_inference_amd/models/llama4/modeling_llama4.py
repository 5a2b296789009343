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

The vision tower (unfold-conv ViT + pixel-shuffle adapter) and the
llava-style conditional-generation wrapper live at the bottom of this file.
"""

from __future__ import annotations

import math
from typing import Dict

import torch
import torch.nn as nn
import torch.nn.functional as F

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


# ----------------------------------------------------------------------
# Vision tower + conditional generation (HF Llama4VisionModel semantics):
# unfold-conv patch embedding, CLS token appended LAST, learned positional
# embedding, complex 2D rope with zeroed CLS angles, LayerNorm pre/post,
# pixel-shuffle adapter, linear projector, llava-style inline splice.
class _Llama4VisionAttention(nn.Module):
    def __init__(self, dim: int, heads: int):
        super().__init__()
        self.heads = heads
        self.head_dim = dim // heads
        self.q_proj = nn.Linear(dim, dim, bias=True)
        self.k_proj = nn.Linear(dim, dim, bias=True)
        self.v_proj = nn.Linear(dim, dim, bias=True)
        self.o_proj = nn.Linear(dim, dim, bias=True)

    def forward(self, x: torch.Tensor, freqs_ci: torch.Tensor) -> torch.Tensor:
        import math as _math

        B, P, _ = x.shape
        q = self.q_proj(x).view(B, P, self.heads, self.head_dim)
        k = self.k_proj(x).view(B, P, self.heads, self.head_dim)
        v = self.v_proj(x).view(B, P, self.heads, self.head_dim)
        qc = torch.view_as_complex(q.float().reshape(B, P, self.heads, -1, 2))
        kc = torch.view_as_complex(k.float().reshape(B, P, self.heads, -1, 2))
        fc = freqs_ci.view(1, P, 1, -1)
        q = torch.view_as_real(qc * fc).flatten(3).type_as(x)
        k = torch.view_as_real(kc * fc).flatten(3).type_as(x)
        from neuronx_distributed_inference_amd import ops as _ops

        out = _ops.attention_prefill(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            mask=torch.ones(1, 1, P, P, dtype=torch.bool, device=x.device),
            causal=False, scale=1.0 / _math.sqrt(self.head_dim),
        )
        return self.o_proj(out.transpose(1, 2).reshape(B, P, -1))


class _Llama4VisionLayer(nn.Module):
    def __init__(self, dim: int, heads: int, inter: int):
        super().__init__()
        self.self_attn = _Llama4VisionAttention(dim, heads)
        self.mlp_fc1 = nn.Linear(dim, inter, bias=True)
        self.mlp_fc2 = nn.Linear(inter, dim, bias=True)
        self.input_layernorm = nn.LayerNorm(dim)
        self.post_attention_layernorm = nn.LayerNorm(dim)

    def forward(self, h, freqs_ci):
        h = h + self.self_attn(self.input_layernorm(h), freqs_ci)
        return h + self.mlp_fc2(F.gelu(self.mlp_fc1(self.post_attention_layernorm(h))))


class Llama4VisionModel(nn.Module):
    def __init__(self, vc: dict):
        super().__init__()
        dim = vc["hidden_size"]
        self.dim = dim
        self.patch_size = vc["patch_size"]
        self.image_size = vc["image_size"]
        idx = self.image_size // self.patch_size
        self.num_patches = idx * idx + 1
        ch = vc.get("num_channels", 3)
        self.patch_embedding_linear = nn.Linear(
            ch * self.patch_size ** 2, dim, bias=False)
        self.class_embedding = nn.Parameter(torch.zeros(dim))
        self.positional_embedding_vlm = nn.Parameter(
            torch.zeros(self.num_patches, dim))
        self.layernorm_pre = nn.LayerNorm(dim)
        self.layernorm_post = nn.LayerNorm(dim)
        heads = vc["num_attention_heads"]
        self.layers = nn.ModuleList(
            _Llama4VisionLayer(dim, heads, vc["intermediate_size"])
            for _ in range(vc["num_hidden_layers"])
        )
        self.pixel_shuffle_ratio = vc.get("pixel_shuffle_ratio", 0.5)
        self.adapter_fc1 = nn.Linear(vc["intermediate_size"],
                                     vc["projector_input_dim"], bias=False)
        self.adapter_fc2 = nn.Linear(vc["projector_output_dim"],
                                     vc["projector_output_dim"], bias=False)
        # complex 2D rope table, CLS angles zeroed (HF _compute_freqs_ci)
        rp = vc.get("rope_parameters") or {}
        theta = rp.get("rope_theta", vc.get("rope_theta", 10000.0))
        head_dim = dim // heads
        img_idx = torch.arange(idx ** 2, dtype=torch.int32).reshape(idx ** 2, 1)
        img_idx = torch.cat([img_idx, img_idx[:1]], dim=0)
        img_idx[-1, -1] = -2
        fx = img_idx % idx
        fy = img_idx // idx
        freq_dim = head_dim // 2
        rope_freq = 1.0 / (theta ** (
            torch.arange(0, freq_dim, 2)[: (freq_dim // 2)].float() / freq_dim))
        freqs_x = ((fx + 1)[..., None] * rope_freq[None, None, :]) \
            .repeat_interleave(2, dim=-1)
        freqs_y = ((fy + 1)[..., None] * rope_freq[None, None, :]) \
            .repeat_interleave(2, dim=-1)
        freqs = torch.cat([freqs_x, freqs_y], dim=-1).float()[..., ::2]
        freqs = freqs.masked_fill(img_idx.reshape(-1, 1, 1) < 0, 0)
        self.register_buffer(
            "freqs_ci",
            torch.view_as_complex(torch.stack([freqs.cos(), freqs.sin()], dim=-1)),
            persistent=False)

    def forward(self, pixel_values: torch.Tensor) -> torch.Tensor:
        B = pixel_values.shape[0]
        x = F.unfold(pixel_values, kernel_size=self.patch_size,
                     stride=self.patch_size).permute(0, 2, 1)
        h = self.patch_embedding_linear(x.to(self.patch_embedding_linear.weight.dtype))
        h = torch.cat([h, self.class_embedding.expand(B, 1, self.dim)], dim=1)
        h = h + self.positional_embedding_vlm.to(h.dtype)
        h = self.layernorm_pre(h)
        for layer in self.layers:
            h = layer(h, self.freqs_ci)
        h = self.layernorm_post(h)[:, :-1, :]
        # pixel shuffle (HF pixel_shuffle) + MLP2 adapter
        r = self.pixel_shuffle_ratio
        P = h.shape[1]
        side = int(math.sqrt(P))
        t = h.view(B, side, side, -1)
        t = t.view(B, side, int(side * r), int(t.shape[-1] / r)).permute(0, 2, 1, 3).contiguous()
        t = t.view(B, int(side * r), int(side * r), int(t.shape[-1] / r)).permute(0, 2, 1, 3).contiguous()
        t = t.view(B, -1, t.shape[-1])
        return F.gelu(self.adapter_fc2(F.gelu(self.adapter_fc1(t))))


class Llama4VisionInferenceConfig(InferenceConfig):
    def get_required_attributes(self):
        return ["text_config", "vision_config", "image_token_id"]

    def add_derived_config(self):
        if isinstance(self.text_config, dict):
            tc = dict(self.text_config)
            tc.pop("neuron_config", None)
            self.text_config = Llama4InferenceConfig(
                neuron_config=self.neuron_config, **tc)
        for a in ("hidden_size", "num_attention_heads", "num_key_value_heads",
                  "num_hidden_layers", "vocab_size"):
            setattr(self, a, getattr(self.text_config, a))


class Llama4ForConditionalGeneration(nn.Module):
    """Vision tower + projector + Llama4 text LM (inline image splice)."""

    _config_cls = Llama4VisionInferenceConfig

    def __init__(self, config: InferenceConfig):
        super().__init__()
        self.config = config
        vc = config.vision_config if isinstance(config.vision_config, dict) \
            else config.vision_config.__dict__
        self.vision_model = Llama4VisionModel(vc)
        self.multi_modal_projector = nn.Linear(
            vc["vision_output_dim"], config.text_config.hidden_size, bias=False)
        self.language_model = Llama4ForCausalLM(config.text_config)

    @torch.inference_mode()
    def generate(self, input_ids, pixel_values=None, attention_mask=None,
                 max_new_tokens: int = 16, **kw):
        from neuronx_distributed_inference_amd.utils.hf_adapter import (
            GenerationConfig,
            HuggingFaceGenerationAdapter,
        )

        dm = self.language_model.model
        dm.multimodal_ctx = {}
        if pixel_values is not None:
            feats = self.vision_model(pixel_values)
            feats = self.multi_modal_projector(feats.reshape(-1, feats.shape[-1]))
            dm.multimodal_ctx = {
                "features": feats, "token_id": self.config.image_token_id,
            }
        adapter = HuggingFaceGenerationAdapter(self.language_model)
        return adapter.generate(
            input_ids, attention_mask=attention_mask,
            generation_config=GenerationConfig(max_new_tokens=max_new_tokens), **kw)

    @staticmethod
    def convert_hf_to_native_state_dict(sd, config):
        text_sd, out = {}, {}
        for k, v in sd.items():
            nk = k[len("model."):] if k.startswith("model.") else k
            if nk.startswith("language_model."):
                text_sd[nk[len("language_model."):]] = v
            elif nk == "lm_head.weight":
                text_sd["lm_head.weight"] = v
            elif nk == "multi_modal_projector.linear_1.weight":
                out["multi_modal_projector.weight"] = v
            elif nk.startswith("vision_model."):
                nk = nk.replace(".patch_embedding.linear.", ".patch_embedding_linear.")
                nk = nk.replace(".model.layers.", ".layers.")
                nk = nk.replace(".vision_adapter.mlp.fc1.", ".adapter_fc1.")
                nk = nk.replace(".vision_adapter.mlp.fc2.", ".adapter_fc2.")
                nk = nk.replace(".mlp.fc1.", ".mlp_fc1.")
                nk = nk.replace(".mlp.fc2.", ".mlp_fc2.")
                out[nk] = v
        conv = Llama4ForCausalLM.convert_hf_to_native_state_dict(
            text_sd, config.text_config)
        out.update({f"language_model.model.{k}": v for k, v in conv.items()})
        dtype = config.neuron_config.torch_dtype
        return {k: v.to(dtype) for k, v in out.items()}
