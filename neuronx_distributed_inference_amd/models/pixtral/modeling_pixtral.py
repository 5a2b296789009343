"""Pixtral (Mistral vision): llava-style image+text conditional generation.

Parity target: the reference's `models/pixtral/`.
Architecture matched against HF `modeling_pixtral.py` + `modeling_llava.py`:

- vision encoder: patch conv (no CLS token), RMSNorm pre-norm, 2D rope
  whose angle table is indexed by `h * max_patches_per_side + w` (h takes
  the even base frequencies, w the odd; stored [h|w]), block-diagonal
  attention (each image attends only itself), mistral-style gated MLPs.
  All images of a batch are flattened into ONE sequence of patches.
- projector: linear -> act -> linear on the final hidden state.
- text model: our Mistral stack (HIP kernels, KV cache, TP); the projected
  vision features replace the `image_token_id` placeholder embeddings via
  the engine's multimodal splice hook (`DecoderModel.multimodal_ctx`) —
  inline-splice, no cross-attention (contrast Mllama).
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.models.mistral.modeling_mistral import (
    MistralForCausalLM,
    MistralInferenceConfig,
)
from neuronx_distributed_inference_amd.modules.norm import RMSNorm


class PixtralInferenceConfig(InferenceConfig):
    def get_required_attributes(self) -> List[str]:
        return ["text_config", "vision_config", "image_token_id"]

    def add_derived_config(self):
        if isinstance(self.text_config, dict):
            tc = dict(self.text_config)
            tc.pop("neuron_config", None)
            self.text_config = MistralInferenceConfig(
                neuron_config=self.neuron_config, **tc
            )
        if not hasattr(self, "projector_hidden_act"):
            self.projector_hidden_act = "gelu"
        if not hasattr(self, "multimodal_projector_bias"):
            self.multimodal_projector_bias = True
        for a in ("hidden_size", "num_attention_heads", "num_key_value_heads",
                  "num_hidden_layers", "vocab_size"):
            setattr(self, a, getattr(self.text_config, a))


class _PixtralVisionAttention(nn.Module):
    def __init__(self, dim: int, heads: int):
        super().__init__()
        self.heads = heads
        self.head_dim = dim // heads
        self.scale = 1.0 / math.sqrt(self.head_dim)
        self.q_proj = nn.Linear(dim, dim, bias=False)
        self.k_proj = nn.Linear(dim, dim, bias=False)
        self.v_proj = nn.Linear(dim, dim, bias=False)
        self.o_proj = nn.Linear(dim, dim, bias=False)

    def forward(self, x, mask, cos, sin):
        B, S, _ = x.shape
        q = self.q_proj(x).view(B, S, self.heads, self.head_dim).transpose(1, 2)
        k = self.k_proj(x).view(B, S, self.heads, self.head_dim).transpose(1, 2)
        v = self.v_proj(x).view(B, S, self.heads, self.head_dim).transpose(1, 2)
        q, k = ops.rope_apply(q, k, cos, sin)
        out = ops.attention_prefill(q, k, v, mask=mask, causal=False, scale=self.scale)
        return self.o_proj(out.transpose(1, 2).reshape(B, S, -1))


class _PixtralVisionMLP(nn.Module):
    def __init__(self, dim: int, inter: int):
        super().__init__()
        self.gate_proj = nn.Linear(dim, inter, bias=False)
        self.up_proj = nn.Linear(dim, inter, bias=False)
        self.down_proj = nn.Linear(inter, dim, bias=False)

    def forward(self, x):
        return self.down_proj(F.silu(self.gate_proj(x)) * self.up_proj(x))


class _PixtralVisionLayer(nn.Module):
    def __init__(self, dim: int, heads: int, inter: int):
        super().__init__()
        self.attention = _PixtralVisionAttention(dim, heads)
        self.feed_forward = _PixtralVisionMLP(dim, inter)
        self.attention_norm = RMSNorm(dim, eps=1e-5)
        self.ffn_norm = RMSNorm(dim, eps=1e-5)

    def forward(self, h, mask, cos, sin):
        h = h + self.attention(self.attention_norm(h), mask, cos, sin)
        return h + self.feed_forward(self.ffn_norm(h))


class PixtralVisionModel(nn.Module):
    def __init__(self, vc: dict):
        super().__init__()
        dim, heads = vc["hidden_size"], vc["num_attention_heads"]
        inter = vc["intermediate_size"]
        self.patch_size = vc["patch_size"]
        self.max_side = vc["image_size"] // vc["patch_size"]
        self.patch_conv = nn.Conv2d(vc.get("num_channels", 3), dim,
                                    kernel_size=self.patch_size,
                                    stride=self.patch_size, bias=False)
        self.ln_pre = RMSNorm(dim, eps=1e-5)
        self.transformer = nn.ModuleList(
            _PixtralVisionLayer(dim, heads, inter)
            for _ in range(vc["num_hidden_layers"])
        )
        # 2D rope angle table, indexed by h * max_side + w: h-frequencies on
        # even slots, w-frequencies on odd (HF PixtralRotaryEmbedding)
        head_dim = dim // heads
        rp = vc.get("rope_parameters") or {}
        base = rp.get("rope_theta", vc.get("rope_theta", 10000.0))
        freqs = 1.0 / (base ** (torch.arange(0, head_dim, 2).float() / head_dim))
        h = torch.arange(self.max_side)
        freqs_h = torch.outer(h.float(), freqs[::2])
        freqs_w = torch.outer(h.float(), freqs[1::2])
        table = torch.cat(
            [
                freqs_h[:, None, :].repeat(1, self.max_side, 1),
                freqs_w[None, :, :].repeat(self.max_side, 1, 1),
            ],
            dim=-1,
        ).reshape(-1, head_dim // 2)
        # HF keeps [h-freqs | w-freqs] concatenated (no re-interleave)
        emb = torch.cat([table, table], dim=-1)
        self.register_buffer("rope_cos", emb.cos(), persistent=False)
        self.register_buffer("rope_sin", emb.sin(), persistent=False)

    def forward(self, pixel_values: torch.Tensor,
                image_sizes: Optional[torch.Tensor] = None) -> torch.Tensor:
        """pixel_values (N_img, C, H, W) -> (1, total_patches, dim)."""
        if image_sizes is None:
            image_sizes = [(pixel_values.shape[2], pixel_values.shape[3])] * \
                pixel_values.shape[0]
        embeds = self.patch_conv(pixel_values.to(self.patch_conv.weight.dtype))
        plist = [e[..., : s[0] // self.patch_size, : s[1] // self.patch_size]
                 for e, s in zip(embeds, image_sizes)]
        h = torch.cat([p.flatten(1).T for p in plist], dim=0).unsqueeze(0)
        h = self.ln_pre(h)
        pos = []
        for p in plist:
            hh, ww = p.shape[-2:]
            grid = torch.arange(hh).view(-1, 1) * self.max_side + torch.arange(ww)
            pos.append(grid.reshape(-1))
        pos = torch.cat(pos).to(h.device)
        cos = self.rope_cos[pos].unsqueeze(0)  # (1, S, D)
        sin = self.rope_sin[pos].unsqueeze(0)
        # block-diagonal: each image's patches attend only themselves
        sizes = [p.shape[-2] * p.shape[-1] for p in plist]
        S = h.shape[1]
        mask = torch.zeros(S, S, dtype=torch.bool, device=h.device)
        start = 0
        for n in sizes:
            mask[start:start + n, start:start + n] = True
            start += n
        mask = mask.view(1, 1, S, S)
        for layer in self.transformer:
            h = layer(h, mask, cos, sin)
        return h


class PixtralMultiModalProjector(nn.Module):
    def __init__(self, vision_dim: int, text_dim: int, act: str, bias: bool):
        super().__init__()
        self.linear_1 = nn.Linear(vision_dim, text_dim, bias=bias)
        self.linear_2 = nn.Linear(text_dim, text_dim, bias=bias)
        self.act = F.gelu if act == "gelu" else F.silu

    def forward(self, x):
        return self.linear_2(self.act(self.linear_1(x)))


class PixtralForConditionalGeneration(nn.Module):
    """Vision tower + projector + Mistral text LM with inline splice."""

    _config_cls = PixtralInferenceConfig

    def __init__(self, config: InferenceConfig):
        super().__init__()
        self.config = config
        vc = config.vision_config if isinstance(config.vision_config, dict) \
            else config.vision_config.__dict__
        self.vision_tower = PixtralVisionModel(vc)
        self.multi_modal_projector = PixtralMultiModalProjector(
            vc["hidden_size"], config.text_config.hidden_size,
            config.projector_hidden_act, bool(config.multimodal_projector_bias),
        )
        self.language_model = MistralForCausalLM(config.text_config)

    @torch.inference_mode()
    def generate(self, input_ids: torch.Tensor, pixel_values=None,
                 image_sizes=None, attention_mask=None,
                 max_new_tokens: int = 16, **kw):
        from neuronx_distributed_inference_amd.utils.hf_adapter import (
            GenerationConfig,
            HuggingFaceGenerationAdapter,
        )

        dm = self.language_model.model
        dm.multimodal_ctx = {}
        if pixel_values is not None:
            feats = self.vision_tower(pixel_values, image_sizes)
            feats = self.multi_modal_projector(feats.squeeze(0))
            dm.multimodal_ctx = {
                "features": feats, "token_id": self.config.image_token_id,
            }
        adapter = HuggingFaceGenerationAdapter(self.language_model)
        return adapter.generate(
            input_ids, attention_mask=attention_mask,
            generation_config=GenerationConfig(max_new_tokens=max_new_tokens), **kw
        )

    @staticmethod
    def convert_hf_to_native_state_dict(
        sd: Dict[str, torch.Tensor], config: InferenceConfig
    ) -> Dict[str, torch.Tensor]:
        text_sd, out = {}, {}
        for k, v in sd.items():
            nk = k[len("model."):] if k.startswith("model.") else k
            if nk.startswith("language_model."):
                text_sd[nk[len("language_model."):]] = v
            elif nk == "lm_head.weight":
                text_sd["lm_head.weight"] = v
            elif nk.startswith(("vision_tower.", "multi_modal_projector.")):
                out[nk.replace(".transformer.layers.", ".transformer.")] = v
        conv = MistralForCausalLM.convert_hf_to_native_state_dict(
            text_sd, config.text_config
        )
        out.update({f"language_model.model.{k}": v for k, v in conv.items()})
        dtype = config.neuron_config.torch_dtype
        return {k: v.to(dtype) for k, v in out.items()}
