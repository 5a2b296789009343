"""Qwen2-VL: ViT vision tower + Qwen2 text model with multimodal (3D) rope.

Parity target: the reference's `models/qwen2_vl/`.
Architecture matched against HF `modeling_qwen2_vl.py`:

- vision tower: Conv3d patchify over (temporal_patch, patch, patch) chunks,
  2-axis (h, w) rotary embedding with block-major spatial-merge ordering,
  per-image block-diagonal full attention (fused qkv + proj, LayerNorm
  pre-norms, quick-GELU MLP), and a PatchMerger (LayerNorm + 2-layer MLP
  over spatial_merge_size^2 patch groups) into the text hidden size;
- text model: our Qwen2 stack (HIP kernels, KV cache, TP). M-ROPE is
  folded into the cos/sin tables BEFORE the rope kernel: the head-dim is
  split into `mrope_section` chunks cycling (temporal, height, width) axis
  positions, so the standard rotate-half HIP kernels (incl. the fused
  qkv_rope_update decode path) run unchanged. Decode steps use
  scalar positions + the request's rope delta (HF rope_deltas), which the
  engine hook applies per step;
- image features splice in via the engine's multimodal_ctx hook.

Video inputs (second grid + temporal fps scaling) are not implemented
(GAPS.md).
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.models.qwen2.modeling_qwen2 import (
    Qwen2ForCausalLM,
    Qwen2InferenceConfig,
)


class Qwen2VLInferenceConfig(InferenceConfig):
    def get_required_attributes(self) -> List[str]:
        return ["text_config", "vision_config", "image_token_id"]

    def add_derived_config(self):
        if isinstance(self.text_config, dict):
            tc = dict(self.text_config)
            tc.pop("neuron_config", None)
            self.text_config = Qwen2InferenceConfig(
                neuron_config=self.neuron_config, **tc
            )
        for a in ("hidden_size", "num_attention_heads", "num_key_value_heads",
                  "num_hidden_layers", "vocab_size"):
            setattr(self, a, getattr(self.text_config, a))


def _quick_gelu(x):
    return x * torch.sigmoid(1.702 * x)


class _VisionBlock(nn.Module):
    def __init__(self, dim: int, heads: int, mlp_hidden: int, act):
        super().__init__()
        self.heads = heads
        self.head_dim = dim // heads
        self.scale = 1.0 / math.sqrt(self.head_dim)
        self.norm1 = nn.LayerNorm(dim, eps=1e-6)
        self.norm2 = nn.LayerNorm(dim, eps=1e-6)
        self.attn_qkv = nn.Linear(dim, dim * 3, bias=True)
        self.attn_proj = nn.Linear(dim, dim)
        self.mlp_fc1 = nn.Linear(dim, mlp_hidden)
        self.mlp_fc2 = nn.Linear(mlp_hidden, dim)
        self.act = act

    def forward(self, h, mask, cos, sin):
        # h: (S, dim) packed sequence over all images
        S = h.shape[0]
        x = self.norm1(h)
        q, k, v = self.attn_qkv(x).reshape(S, 3, self.heads, self.head_dim) \
            .permute(1, 2, 0, 3).unbind(0)  # each (H, S, D)
        q, k = ops.rope_apply(q.unsqueeze(0), k.unsqueeze(0), cos, sin)
        out = ops.attention_prefill(q, k, v.unsqueeze(0), mask=mask,
                                    causal=False, scale=self.scale)
        h = h + self.attn_proj(out.squeeze(0).transpose(0, 1).reshape(S, -1))
        return h + self.mlp_fc2(self.act(self.mlp_fc1(self.norm2(h))))


class Qwen2VLVisionModel(nn.Module):
    def __init__(self, vc: dict, text_hidden: int):
        super().__init__()
        dim = vc["embed_dim"]
        self.dim = dim
        self.patch_size = vc["patch_size"]
        self.temporal_patch_size = vc.get("temporal_patch_size", 2)
        self.spatial_merge_size = vc.get("spatial_merge_size", 2)
        self.in_channels = vc.get("in_channels", 3)
        heads = vc["num_heads"]
        self.head_dim = dim // heads
        mlp_hidden = int(dim * vc.get("mlp_ratio", 4))
        act = _quick_gelu if vc.get("hidden_act", "quick_gelu") == "quick_gelu" \
            else F.gelu
        k = [self.temporal_patch_size, self.patch_size, self.patch_size]
        self.patch_embed_proj = nn.Conv3d(self.in_channels, dim, kernel_size=k,
                                          stride=k, bias=False)
        self.blocks = nn.ModuleList(
            _VisionBlock(dim, heads, mlp_hidden, act) for _ in range(vc["depth"])
        )
        m = self.spatial_merge_size
        self.merger_ln_q = nn.LayerNorm(dim, eps=1e-6)
        self.merger_mlp = nn.Sequential(
            nn.Linear(dim * m * m, dim * m * m), nn.GELU(),
            nn.Linear(dim * m * m, vc.get("hidden_size", text_hidden)),
        )
        # 2-axis rotary: h takes the first head_dim/4 freqs, w the same set
        inv = 1.0 / (10000.0 ** (
            torch.arange(0, self.head_dim // 2, 2, dtype=torch.float)
            / (self.head_dim // 2)))
        self.register_buffer("rope_inv_freq", inv, persistent=False)

    def _position_ids(self, grid_thw) -> torch.Tensor:
        """(total_patches, 2) h/w indices, block-major over merge blocks
        (HF get_vision_position_ids)."""
        m = self.spatial_merge_size
        out = []
        for t, h, w in grid_thw.tolist():
            hpos, wpos = torch.meshgrid(torch.arange(h), torch.arange(w),
                                        indexing="ij")
            shape = (h // m, m, w // m, m)
            hpos = hpos.reshape(shape).transpose(1, 2).flatten()
            wpos = wpos.reshape(shape).transpose(1, 2).flatten()
            out.append(torch.stack([hpos, wpos], dim=-1).repeat(t, 1))
        return torch.cat(out)

    def forward(self, pixel_values: torch.Tensor, grid_thw: torch.Tensor):
        """pixel_values (total_patches, C*tp*p*p) processor layout;
        grid_thw (n_images, 3). Returns merged features
        (total_patches / merge^2, text_hidden)."""
        x = pixel_values.view(-1, self.in_channels, self.temporal_patch_size,
                              self.patch_size, self.patch_size)
        h = self.patch_embed_proj(x.to(self.patch_embed_proj.weight.dtype))
        h = h.view(-1, self.dim)
        pos = self._position_ids(grid_thw).to(h.device)
        freqs = (pos.unsqueeze(-1).float() * self.rope_inv_freq).flatten(1)
        emb = torch.cat([freqs, freqs], dim=-1)  # (S, head_dim)
        cos = emb.cos().view(1, -1, emb.shape[-1])
        sin = emb.sin().view(1, -1, emb.shape[-1])
        # per-image block-diagonal attention (cu_seqlens blocks)
        sizes = [int(t * hh * ww) for t, hh, ww in grid_thw.tolist()]
        S = h.shape[0]
        mask = torch.zeros(S, S, dtype=torch.bool, device=h.device)
        start = 0
        for n in sizes:
            mask[start:start + n, start:start + n] = True
            start += n
        mask = mask.view(1, 1, S, S)
        for blk in self.blocks:
            h = blk(h, mask, cos, sin)
        m2 = self.spatial_merge_size ** 2
        return self.merger_mlp(self.merger_ln_q(h).view(-1, self.dim * m2))


class Qwen2VLForConditionalGeneration(nn.Module):
    """Vision tower + Qwen2 text LM with m-rope and inline image splice."""

    _config_cls = Qwen2VLInferenceConfig

    def __init__(self, config: InferenceConfig):
        super().__init__()
        self.config = config
        vc = config.vision_config if isinstance(config.vision_config, dict) \
            else config.vision_config.__dict__
        self.visual = Qwen2VLVisionModel(vc, config.text_config.hidden_size)
        self.language_model = Qwen2ForCausalLM(config.text_config)
        tc = config.text_config
        rs = getattr(tc, "rope_scaling", None)
        if not isinstance(rs, dict):
            rs = getattr(tc, "rope_parameters", None) or {}
        head_dim = getattr(tc, "head_dim", None) or \
            tc.hidden_size // tc.num_attention_heads
        # mrope_section counts rotate-half PAIRS; defaults to an even t/h/w
        # split of head_dim/2
        self.mrope_section = list(rs.get("mrope_section", None)
                                  or [head_dim // 6] * 3)
        theta = getattr(tc, "rope_theta", 1000000.0)
        inv = 1.0 / (theta ** (torch.arange(0, head_dim, 2).float() / head_dim))
        self.register_buffer("inv_freq", inv, persistent=False)
        self._mrope_ctx: dict = {}
        self.language_model.model._cos_sin_fn = self._mrope_cos_sin

    # ------------------------------------------------------------------
    def _compute_rope_index(self, input_ids: torch.Tensor,
                            grid_thw: Optional[torch.Tensor],
                            video_grid_thw: Optional[torch.Tensor] = None):
        """3D (t,h,w) positions for a mixed text+image/video prefill and
        the decode delta (HF get_rope_index behavior). Video segments
        advance the temporal axis per frame; the next text position jumps
        past max(t, lh, lw)."""
        B, S = input_ids.shape
        image_tok = self.config.image_token_id
        video_tok = getattr(self.config, "video_token_id", None)
        pos3 = torch.zeros(3, B, S, dtype=torch.long)
        deltas = torch.zeros(B, dtype=torch.long)
        for b in range(B):
            ids = input_ids[b].tolist()
            img_i = vid_i = 0
            cur = 0  # next text position
            i = 0
            m = self.visual.spatial_merge_size
            while i < S:
                is_img = ids[i] == image_tok
                is_vid = video_tok is not None and ids[i] == video_tok
                if is_img or is_vid:
                    if is_img:
                        t, h, w = grid_thw[img_i].tolist()
                        img_i += 1
                    else:
                        t, h, w = video_grid_thw[vid_i].tolist()
                        vid_i += 1
                    lh, lw = h // m, w // m
                    n = t * lh * lw
                    ti = torch.arange(t).view(-1, 1).expand(t, lh * lw).reshape(-1)
                    r = torch.arange(lh).view(-1, 1).expand(lh, lw).reshape(-1)
                    c = torch.arange(lw).view(1, -1).expand(lh, lw).reshape(-1)
                    pos3[0, b, i:i + n] = cur + ti
                    pos3[1, b, i:i + n] = cur + r.repeat(t)
                    pos3[2, b, i:i + n] = cur + c.repeat(t)
                    cur = cur + max(t, lh, lw)
                    i += n
                else:
                    pos3[:, b, i] = cur
                    cur += 1
                    i += 1
            deltas[b] = cur - S
        return pos3, deltas

    def _mrope_cos_sin(self, position_ids: torch.Tensor, device=None):
        """Build COMBINED (B,S,head_dim) cos/sin: head-dim sections cycle
        the (t,h,w) axes, so the standard rotate-half kernels apply."""
        ctx = self._mrope_ctx
        B, S = position_ids.shape
        dev = device or position_ids.device
        if S > 1 and ctx.get("pos3d") is not None:
            pos3 = ctx["pos3d"].to(dev)
            if pos3.shape[2] < S:  # bucket padding: pad rows are unused
                pad = torch.zeros(3, B, S - pos3.shape[2], dtype=pos3.dtype,
                                  device=dev)
                pos3 = torch.cat([pos3, pad], dim=2)
            pos3 = pos3[:, :, :S]
        else:
            delta = ctx.get("delta")
            p = position_ids.clamp(min=0)
            if delta is not None:
                p = p + delta.to(p.device).view(-1, 1)
            pos3 = p.unsqueeze(0).expand(3, B, S)
        freqs = pos3.unsqueeze(-1).float().to(dev) * self.inv_freq.to(dev)
        emb = torch.cat([freqs, freqs], dim=-1)     # (3, B, S, head_dim)
        cos, sin = emb.cos(), emb.sin()
        sec = self.mrope_section * 2
        cos = torch.cat([c[i % 3] for i, c in enumerate(cos.split(sec, dim=-1))],
                        dim=-1)
        sin = torch.cat([c[i % 3] for i, c in enumerate(sin.split(sec, dim=-1))],
                        dim=-1)
        return cos.float(), sin.float()

    # ------------------------------------------------------------------
    @torch.inference_mode()
    def generate(self, input_ids: torch.Tensor, pixel_values=None,
                 image_grid_thw=None, pixel_values_videos=None,
                 video_grid_thw=None, attention_mask=None,
                 max_new_tokens: int = 16, **kw):
        from neuronx_distributed_inference_amd.utils.hf_adapter import (
            GenerationConfig,
            HuggingFaceGenerationAdapter,
        )

        dm = self.language_model.model
        dm.multimodal_ctx = {}
        self._mrope_ctx = {}
        splices = []
        if pixel_values is not None:
            splices.append((self.config.image_token_id,
                            self.visual(pixel_values, image_grid_thw)))
        if pixel_values_videos is not None:
            splices.append((self.config.video_token_id,
                            self.visual(pixel_values_videos, video_grid_thw)))
        if splices:
            dm.multimodal_ctx = {"splices": splices}
            pos3, deltas = self._compute_rope_index(
                input_ids, image_grid_thw, video_grid_thw)
            self._mrope_ctx = {"pos3d": pos3, "delta": deltas}
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
            elif nk.startswith("visual."):
                nk = nk.replace(".patch_embed.proj.", ".patch_embed_proj.")
                nk = nk.replace(".attn.qkv.", ".attn_qkv.")
                nk = nk.replace(".attn.proj.", ".attn_proj.")
                nk = nk.replace(".mlp.fc1.", ".mlp_fc1.")
                nk = nk.replace(".mlp.fc2.", ".mlp_fc2.")
                nk = nk.replace(".merger.ln_q.", ".merger_ln_q.")
                nk = nk.replace(".merger.mlp.", ".merger_mlp.")
                out[nk] = v
        conv = Qwen2ForCausalLM.convert_hf_to_native_state_dict(
            text_sd, config.text_config
        )
        out.update({f"language_model.model.{k}": v for k, v in conv.items()})
        dtype = config.neuron_config.torch_dtype
        return {k: v.to(dtype) for k, v in out.items()}
