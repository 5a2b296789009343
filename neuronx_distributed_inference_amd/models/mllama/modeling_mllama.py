"""Mllama (Llama-3.2 Vision): image+text conditional generation.

Parity target: the reference's `models/mllama/` (~2800 LoC multimodal
stack). Architecture matched against HF `modeling_mllama.py`:

- vision tower: tiled ViT — patch conv, class token, gated positional +
  pre/post tile aspect-ratio embeddings, a local transformer whose chosen
  intermediate layer outputs are concatenated onto the final hidden state,
  and a tanh-gated global transformer. Runs once per request (prefill
  shaped), replicated across ranks — library GEMMs.
- text model: standard llama decoder layers with the HIP decode/prefill
  kernels + KV cache, EXCEPT the `cross_attention_layers`, which are
  tanh-gated cross-attention blocks over the projected vision states with
  per-head q/k RMS norms. Cross K/V are projected ONCE at context encoding
  and decode steps hit the HIP decode kernel against them (constant
  full-length positions, like Whisper's cross attention).
- text-only requests skip the cross layers entirely (HF behavior).

`cross_attention_mask` (per-token tile masks) is supported with HF-exact
semantics: the (B, T, n_images, n_tiles) 0/1 mask expands to an additive
vision-token mask, fully-masked rows fall back to uniform attention, and
the full-row mask zeroes those rows' cross-MLP contribution; generated
tokens past the prompt reuse the last mask row (HF generation repeats
it per new token). Without a mask every text row attends every vision
token via the HIP decode kernel (the common single-image chat case).
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
    LlamaDecoderLayer,
    LlamaForCausalLM,
    LlamaInferenceConfig,
)
from neuronx_distributed_inference_amd.models.model_base import DecoderModel
from neuronx_distributed_inference_amd.modules.norm import RMSNorm
from neuronx_distributed_inference_amd.parallel.layers import (
    ColumnParallelLinear,
    ParallelEmbedding,
    RowParallelLinear,
)


# ----------------------------------------------------------------------
# configs
class MllamaTextInferenceConfig(LlamaInferenceConfig):
    def get_required_attributes(self):
        return super().get_required_attributes() + ["cross_attention_layers"]


class MllamaInferenceConfig(InferenceConfig):
    """Composite config: text_config + vision_config dicts (HF layout)."""

    def get_required_attributes(self) -> List[str]:
        return ["text_config", "vision_config"]

    def add_derived_config(self):
        if isinstance(self.text_config, dict):
            tc = dict(self.text_config)
            tc.pop("neuron_config", None)
            self.text_config = MllamaTextInferenceConfig(
                neuron_config=self.neuron_config, **tc
            )
        # expose decoder geometry at top level for generic plumbing
        for a in ("hidden_size", "num_attention_heads", "num_key_value_heads",
                  "num_hidden_layers", "vocab_size"):
            setattr(self, a, getattr(self.text_config, a))


# ----------------------------------------------------------------------
# vision tower (replicated, runs once per request)
class MllamaVisionAttention(nn.Module):
    def __init__(self, dim: int, heads: int):
        super().__init__()
        self.heads = heads
        self.head_dim = dim // heads
        self.scale = 1.0 / math.sqrt(self.head_dim)
        self.q_proj = nn.Linear(dim, dim, bias=False)
        self.k_proj = nn.Linear(dim, dim, bias=False)
        self.v_proj = nn.Linear(dim, dim, bias=False)
        self.o_proj = nn.Linear(dim, dim, bias=False)

    def forward(self, x: torch.Tensor, mask: Optional[torch.Tensor]) -> torch.Tensor:
        B, S, _ = x.shape
        q = self.q_proj(x).view(B, S, self.heads, self.head_dim).transpose(1, 2)
        k = self.k_proj(x).view(B, S, self.heads, self.head_dim).transpose(1, 2)
        v = self.v_proj(x).view(B, S, self.heads, self.head_dim).transpose(1, 2)
        out = ops.attention_prefill(q, k, v, mask=mask, causal=False, scale=self.scale)
        return self.o_proj(out.transpose(1, 2).reshape(B, S, -1))


class MllamaVisionMLP(nn.Module):
    def __init__(self, dim: int, inter: int):
        super().__init__()
        self.fc1 = nn.Linear(dim, inter)
        self.fc2 = nn.Linear(inter, dim)

    def forward(self, x):
        return self.fc2(F.gelu(self.fc1(x)))


class MllamaVisionEncoderLayer(nn.Module):
    def __init__(self, dim: int, heads: int, inter: int, norm_eps: float,
                 is_gated: bool):
        super().__init__()
        self.is_gated = is_gated
        self.self_attn = MllamaVisionAttention(dim, heads)
        self.mlp = MllamaVisionMLP(dim, inter)
        self.input_layernorm = nn.LayerNorm(dim, eps=norm_eps)
        self.post_attention_layernorm = nn.LayerNorm(dim, eps=norm_eps)
        if is_gated:
            self.gate_attn = nn.Parameter(torch.ones(1) * math.pi / 4)
            self.gate_ffn = nn.Parameter(torch.ones(1) * math.pi / 4)

    def forward(self, h: torch.Tensor, mask: Optional[torch.Tensor]) -> torch.Tensor:
        a = self.self_attn(self.input_layernorm(h), mask)
        if self.is_gated:
            a = self.gate_attn.tanh() * a
        h = h + a
        m = self.mlp(self.post_attention_layernorm(h))
        if self.is_gated:
            m = self.gate_ffn.tanh() * m
        return h + m


class MllamaPrecomputedAspectRatioEmbedding(nn.Module):
    def __init__(self, max_tiles: int, dim: int, max_ar_id: int, is_gated: bool):
        super().__init__()
        self.max_tiles, self.dim, self.is_gated = max_tiles, dim, is_gated
        self.embedding = nn.Embedding(max_ar_id + 1, max_tiles * dim)
        if is_gated:
            self.gate = nn.Parameter(torch.zeros(1))

    def forward(self, h: torch.Tensor, ar_ids: torch.Tensor) -> torch.Tensor:
        e = self.embedding(ar_ids).reshape(-1, self.max_tiles, 1, self.dim)
        if self.is_gated:
            e = e * self.gate.tanh()
        return h + e


class MllamaPrecomputedPositionEmbedding(nn.Module):
    def __init__(self, max_tiles: int, num_patches: int, dim: int, max_ar_id: int):
        super().__init__()
        self.max_tiles, self.num_patches, self.dim = max_tiles, num_patches, dim
        self.gate = nn.Parameter(torch.zeros(1))
        self.embedding = nn.Parameter(torch.zeros(num_patches, dim))
        self.tile_embedding = nn.Embedding(max_ar_id + 1, max_tiles * num_patches * dim)

    def forward(self, h: torch.Tensor, ar_ids: torch.Tensor) -> torch.Tensor:
        h = h + (1 - self.gate.tanh()) * self.embedding.view(1, 1, self.num_patches, self.dim)
        tile = self.tile_embedding(ar_ids).reshape(
            h.shape[0], self.max_tiles, self.num_patches, self.dim
        )
        return h + self.gate.tanh() * tile


class MllamaVisionModel(nn.Module):
    def __init__(self, vc: dict):
        super().__init__()
        dim = vc["hidden_size"]
        self.hidden_size = dim
        self.num_patches = (vc["image_size"] // vc["patch_size"]) ** 2 + 1
        self.max_num_tiles = vc["max_num_tiles"]
        self.intermediate_layers_indices = list(vc["intermediate_layers_indices"])
        max_ar = vc.get("max_aspect_ratio_id") or len(vc["supported_aspect_ratios"])
        heads, inter = vc["attention_heads"], vc["intermediate_size"]
        eps = vc.get("norm_eps", 1e-5)
        self.patch_embedding = nn.Conv2d(
            vc.get("num_channels", 3), dim, kernel_size=vc["patch_size"],
            stride=vc["patch_size"], bias=False,
        )
        self.class_embedding = nn.Parameter(torch.zeros(dim))
        self.gated_positional_embedding = MllamaPrecomputedPositionEmbedding(
            self.max_num_tiles, self.num_patches, dim, max_ar
        )
        self.pre_tile_positional_embedding = MllamaPrecomputedAspectRatioEmbedding(
            self.max_num_tiles, dim, max_ar, is_gated=True
        )
        self.post_tile_positional_embedding = MllamaPrecomputedAspectRatioEmbedding(
            self.max_num_tiles, dim, max_ar, is_gated=True
        )
        self.layernorm_pre = nn.LayerNorm(dim)
        self.layernorm_post = nn.LayerNorm(dim)
        self.transformer = nn.ModuleList(
            MllamaVisionEncoderLayer(dim, heads, inter, eps, is_gated=False)
            for _ in range(vc["num_hidden_layers"])
        )
        self.global_transformer = nn.ModuleList(
            MllamaVisionEncoderLayer(dim, heads, inter, eps, is_gated=True)
            for _ in range(vc["num_global_layers"])
        )

    def forward(self, pixel_values: torch.Tensor, aspect_ratio_ids: torch.Tensor,
                aspect_ratio_mask: torch.Tensor) -> torch.Tensor:
        """pixel_values (B, n_img, n_tiles, C, H, W) -> vision tokens
        (B, n_img, n_tiles, num_patches, dim * (1 + n_intermediate))."""
        B, n_img, n_tiles, C, H, W = pixel_values.shape
        px = pixel_values.reshape(B * n_img * n_tiles, C, H, W)
        ar_ids = aspect_ratio_ids.reshape(B * n_img, -1)
        h = self.patch_embedding(px.to(self.patch_embedding.weight.dtype))
        h = h.flatten(2).transpose(1, 2)  # (B*I*T, P, dim)
        _, P, dim = h.shape
        h = h.reshape(B * n_img, n_tiles, -1, dim)
        h = self.pre_tile_positional_embedding(h, ar_ids)
        h = h.reshape(B * n_img * n_tiles, P, dim)
        h = torch.cat([self.class_embedding.expand(h.shape[0], 1, dim), h], dim=1)
        P += 1
        h = h.reshape(B * n_img, n_tiles, P, dim)
        h = self.gated_positional_embedding(h, ar_ids)
        h = self.layernorm_pre(h)
        # pad patch dim to a multiple of 8 (HF does the same)
        pad = (8 - P % 8) % 8
        h = F.pad(h, (0, 0, 0, pad))
        Pp = P + pad
        # tile-validity mask, HF semantics: position i is blocked from j only
        # when BOTH are invalid (valid queries still see padded keys)
        v = aspect_ratio_mask.reshape(B * n_img, n_tiles, 1).expand(-1, -1, Pp).clone()
        v[:, :, P:] = 0
        v = v.reshape(B * n_img, n_tiles * Pp).bool()
        mask = (v.unsqueeze(-1) | v.unsqueeze(1)).unsqueeze(1)  # (B*I,1,S,S)
        h = h.view(B * n_img, n_tiles * Pp, dim)
        intermediates = []
        for i, layer in enumerate(self.transformer):
            if i in self.intermediate_layers_indices:
                intermediates.append(h)
            h = layer(h, mask)
        if len(self.transformer) in self.intermediate_layers_indices:
            intermediates.append(h)
        h = self.layernorm_post(h)
        h = h.reshape(B * n_img, n_tiles, Pp, dim)
        h = self.post_tile_positional_embedding(h, ar_ids)
        h = h.reshape(B * n_img, n_tiles * Pp, dim)
        for layer in self.global_transformer:
            h = layer(h, mask)
        h = h.reshape(B * n_img, n_tiles, Pp, dim)[:, :, :P]
        inter = torch.stack(intermediates, dim=-1)
        inter = inter.reshape(B * n_img, n_tiles, Pp, -1)[:, :, :P]
        out = torch.cat([h, inter], dim=-1)
        return out.reshape(B, n_img, n_tiles, P, -1)


# ----------------------------------------------------------------------
# text side
def _prepare_cross_attention_mask(cross_attention_mask: torch.Tensor,
                                  num_vision_tokens: int, dtype: torch.dtype):
    """(B, T, n_images, n_tiles) 0/1 tile mask -> additive (B,1,T,Sv) mask +
    (B,1,T,1) full-row mask, HF semantics (transformers mllama
    `_prepare_cross_attention_mask`): masked entries get finfo.min, and rows
    whose EVERY vision token is masked are flipped back to all-zero
    (uniform attention, no NaN) while the row mask later zeroes their MLP
    contribution."""
    B, T = cross_attention_mask.shape[:2]
    m = cross_attention_mask.repeat_interleave(num_vision_tokens, dim=3)
    m = m.view(B, T, -1).unsqueeze(1).to(dtype)
    inv = 1.0 - m
    neg = torch.finfo(dtype).min
    m = inv.masked_fill(inv.bool(), neg)
    row = (m != neg).any(dim=-1).type_as(m)[..., None]
    return m * row, row


class _MllamaTextCrossAttention(nn.Module):
    """The projection/norm block, named to mirror the HF checkpoint
    (layers.N.cross_attn.{q,k,v,o}_proj / q_norm / k_norm)."""

    def __init__(self, config: InferenceConfig):
        super().__init__()
        ec = config.neuron_config
        tp, dtype = ec.tp_degree, ec.torch_dtype
        d = config.hidden_size
        self.heads = config.num_attention_heads
        self.kv_heads = getattr(config, "num_key_value_heads", self.heads)
        self.head_dim = d // self.heads
        self.heads_per_rank = self.heads // tp
        self.kv_heads_per_rank = max(self.kv_heads // tp, 1)
        self.scale = 1.0 / math.sqrt(self.head_dim)
        self.q_proj = ColumnParallelLinear(d, self.heads * self.head_dim, bias=False,
                                           dtype=dtype, tp_degree=tp)
        self.k_proj = ColumnParallelLinear(d, self.kv_heads * self.head_dim, bias=False,
                                           dtype=dtype, tp_degree=tp)
        self.v_proj = ColumnParallelLinear(d, self.kv_heads * self.head_dim, bias=False,
                                           dtype=dtype, tp_degree=tp)
        self.o_proj = RowParallelLinear(self.heads * self.head_dim, d, bias=False,
                                        dtype=dtype, tp_degree=tp)
        self.q_norm = RMSNorm(self.head_dim, eps=config.rms_norm_eps, dtype=dtype)
        self.k_norm = RMSNorm(self.head_dim, eps=config.rms_norm_eps, dtype=dtype)

    def project_kv(self, states: torch.Tensor):
        B, Sv, _ = states.shape
        k = self.k_proj(states).view(B, Sv, self.kv_heads_per_rank, self.head_dim)
        k = self.k_norm(k).transpose(1, 2).contiguous()
        v = self.v_proj(states).view(B, Sv, self.kv_heads_per_rank,
                                     self.head_dim).transpose(1, 2).contiguous()
        return k, v

    def forward(self, x: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                mask: Optional[torch.Tensor] = None):
        B, S, _ = x.shape
        q = self.q_proj(x).view(B, S, self.heads_per_rank, self.head_dim)
        q = self.q_norm(q).transpose(1, 2)
        if mask is not None:
            # per-token tile mask (additive, (B,1,S,Sv)): eager masked path —
            # Sv is a few thousand vision tokens, so this is tiny at decode
            g = self.heads_per_rank // self.kv_heads_per_rank
            kk = k.repeat_interleave(g, dim=1) if g > 1 else k
            vv = v.repeat_interleave(g, dim=1) if g > 1 else v
            s = (q @ kk.transpose(-1, -2)) * self.scale + mask
            a = s.float().softmax(-1).to(q.dtype) @ vv
        else:
            # all-attend: constant full-length per-query positions through
            # the HIP decode kernel
            pos = torch.full((B, S), k.shape[2] - 1, device=x.device,
                             dtype=torch.int32)
            a = ops.attention_decode(q, k, v, pos, scale=self.scale)
        return self.o_proj(a.transpose(1, 2).reshape(B, S, -1))


class MllamaCrossAttentionLayer(nn.Module):
    """Tanh-gated cross-attention decoder layer (HF
    MllamaCrossAttentionDecoderLayer). Keeps the engine's (h, residual)
    threading convention; skipped when no vision context is set."""

    def __init__(self, config: InferenceConfig, layer_idx: int):
        super().__init__()
        ec = config.neuron_config
        tp, dtype = ec.tp_degree, ec.torch_dtype
        d = config.hidden_size
        self.cross_attn = _MllamaTextCrossAttention(config)
        self.input_layernorm = RMSNorm(d, eps=config.rms_norm_eps, dtype=dtype)
        self.post_attention_layernorm = RMSNorm(d, eps=config.rms_norm_eps, dtype=dtype)
        self.cross_attn_attn_gate = nn.Parameter(torch.zeros(1))
        self.cross_attn_mlp_gate = nn.Parameter(torch.zeros(1))
        from neuronx_distributed_inference_amd.modules.mlp import GatedMLP

        self.mlp = GatedMLP(d, config.intermediate_size, tp_degree=tp, dtype=dtype,
                            hidden_act=getattr(config, "hidden_act", "silu"))
        self.ctx: dict = {}  # shared per-model: {"states": (B, Sv, d)} or empty
        self._kv = None      # projected (k, v) for the current request

    def forward(self, h, residual, cos, sin, kv_mgr, position_ids, phase,
                attention_mask, seq_ids, sequence_parallel=False):
        states = self.ctx.get("states")
        if states is None:
            return h, residual  # text-only: HF skips cross layers
        if residual is None:
            residual = h
            x = self.input_layernorm(h)
        else:
            x, residual = self.input_layernorm(h, residual)
        if phase == "context_encoding" or self._kv is None:
            self._kv = self.cross_attn.project_kv(states)
        k, v = self._kv
        mask, rowq = self.ctx.get("mask"), None
        if mask is not None:
            # gather this step's rows by text position; positions past the
            # prompt reuse the LAST row (HF generation repeats it per new
            # token; ref mllama _update_model_kwargs cross_attention_mask cat)
            B, S = position_ids.shape
            Sv, T = mask.shape[-1], mask.shape[2]
            pos = position_ids.clamp(min=0, max=T - 1).long()
            mask = mask.gather(2, pos[:, None, :, None].expand(B, 1, S, Sv))
            rowq = self.ctx["row_mask"].gather(2, pos[:, None, :, None])
        a = self.cross_attn(x, k, v, mask)
        h = self.cross_attn_attn_gate.tanh() * a
        x, residual = self.post_attention_layernorm(h, residual)
        out = self.mlp(x)
        if rowq is not None:
            # rows whose every vision token is masked contribute no MLP
            # update either (HF full_text_row_masked_out_mask semantics)
            out = rowq[:, 0] * out
        h = self.cross_attn_mlp_gate.tanh() * out
        return h, residual


class MllamaTextModel(DecoderModel):
    def __init__(self, config: InferenceConfig):
        super().__init__(config)
        ec = config.neuron_config
        # HF mllama embeds vocab+8 special image tokens
        self.embed_tokens = ParallelEmbedding(
            config.vocab_size + 8, config.hidden_size, dtype=ec.torch_dtype,
            tp_degree=ec.tp_degree,
        )
        self.cross_ctx: dict = {}
        for layer in self.layers:
            if isinstance(layer, MllamaCrossAttentionLayer):
                layer.ctx = self.cross_ctx

    def build_layers(self, config: InferenceConfig) -> nn.ModuleList:
        cross = set(config.cross_attention_layers)
        return nn.ModuleList(
            MllamaCrossAttentionLayer(config, i) if i in cross
            else LlamaDecoderLayer(config, i)
            for i in range(config.num_hidden_layers)
        )


class MllamaTextForCausalLM(LlamaForCausalLM):
    _model_cls = MllamaTextModel
    _config_cls = MllamaTextInferenceConfig


class MllamaForConditionalGeneration(nn.Module):
    """Vision tower + projector + text causal LM (HF layout)."""

    _config_cls = MllamaInferenceConfig

    def __init__(self, config: InferenceConfig):
        super().__init__()
        self.config = config
        vc = config.vision_config if isinstance(config.vision_config, dict) \
            else config.vision_config.__dict__
        self.vision_model = MllamaVisionModel(vc)
        n_inter = len(vc["intermediate_layers_indices"])
        vision_out = vc.get("vision_output_dim") or vc["hidden_size"] * (1 + n_inter)
        self.multi_modal_projector = nn.Linear(
            vision_out, config.text_config.hidden_size, bias=True
        )
        self.language_model = MllamaTextForCausalLM(config.text_config)

    def encode_images(self, pixel_values, aspect_ratio_ids, aspect_ratio_mask):
        """-> (B, Sv, text_hidden) cross-attention states."""
        vis = self.vision_model(pixel_values, aspect_ratio_ids, aspect_ratio_mask)
        states = self.multi_modal_projector(vis)
        B = states.shape[0]
        return states.reshape(B, -1, states.shape[-1])

    @torch.inference_mode()
    def generate(self, input_ids: torch.Tensor, pixel_values=None,
                 aspect_ratio_ids=None, aspect_ratio_mask=None,
                 cross_attention_mask=None,
                 attention_mask=None, max_new_tokens: int = 16, **kw):
        from neuronx_distributed_inference_amd.utils.hf_adapter import (
            GenerationConfig,
            HuggingFaceGenerationAdapter,
        )

        self.language_model.model.cross_ctx.clear()
        for layer in self.language_model.model.layers:
            if isinstance(layer, MllamaCrossAttentionLayer):
                layer._kv = None
        if pixel_values is not None:
            states = self.encode_images(pixel_values, aspect_ratio_ids, aspect_ratio_mask)
            dtype = self.language_model.model.embed_tokens.weight.dtype
            self.language_model.model.cross_ctx["states"] = states.to(dtype)
            if cross_attention_mask is not None:
                m, row = _prepare_cross_attention_mask(
                    cross_attention_mask, self.vision_model.num_patches, dtype
                )
                self.language_model.model.cross_ctx["mask"] = m
                self.language_model.model.cross_ctx["row_mask"] = row
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
            elif nk.startswith(("vision_model.", "multi_modal_projector.")):
                out[nk] = v
            elif nk == "lm_head.weight":
                text_sd["lm_head.weight"] = v
        conv = LlamaForCausalLM.convert_hf_to_native_state_dict(
            text_sd, config.text_config
        )
        # HF vision layer container names -> ours (flat ModuleLists)
        for k, v in list(out.items()):
            nk = k.replace(".transformer.layers.", ".transformer.") \
                  .replace(".global_transformer.layers.", ".global_transformer.")
            if nk != k:
                out[nk] = out.pop(k)
        out.update({f"language_model.model.{k}": v for k, v in conv.items()})
        dtype = config.neuron_config.torch_dtype
        return {k: v.to(dtype) for k, v in out.items()}
