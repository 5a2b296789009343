"""Qwen3-VL: ViT with DeepStack multi-level features + Qwen3 text model
with interleaved m-rope.

Parity target: the reference's `models/qwen3_vl/`.
Architecture matched against HF `modeling_qwen3_vl.py`:

- vision tower: Conv3d patchify (biased), LEARNED position grid
  bilinearly resampled to each image's merged grid
  (transformers.vision_utils interpolation indices/weights), 2-axis
  rotary, per-image block-diagonal attention; DeepStack: hidden states
  after chosen blocks go through their own post-shuffle PatchMergers and
  are ADDED to the text hidden states after decoder layers 0..n-1 at the
  image positions (engine `_layer_post_fns` hook);
- text model: our Qwen3 stack (per-head q/k RMS norms). M-rope uses
  Qwen3-VL's INTERLEAVED section layout — [t h w t h w ... t t] over
  frequency slots instead of contiguous chunks — folded into the cos/sin
  tables so the standard rotate-half HIP kernels run unchanged;
- rope index: text positions advance by 1; each vision block advances by
  max(lh, lw) with (t=cur, h=cur+row, w=cur+col) merged-grid positions;
  decode uses scalar positions + per-request delta.

Video inputs are not implemented (GAPS.md).
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.models.qwen3.modeling_qwen3 import (
    Qwen3ForCausalLM,
    Qwen3InferenceConfig,
)


class Qwen3VLInferenceConfig(InferenceConfig):
    def get_required_attributes(self) -> List[str]:
        return ["text_config", "vision_config", "image_token_id"]

    def add_derived_config(self):
        if isinstance(self.text_config, dict):
            tc = dict(self.text_config)
            tc.pop("neuron_config", None)
            self.text_config = Qwen3InferenceConfig(
                neuron_config=self.neuron_config, **tc
            )
        for a in ("hidden_size", "num_attention_heads", "num_key_value_heads",
                  "num_hidden_layers", "vocab_size"):
            setattr(self, a, getattr(self.text_config, a))


class _PatchMerger(nn.Module):
    def __init__(self, dim: int, merge2: int, out_dim: int, postshuffle: bool):
        super().__init__()
        hidden = dim * merge2
        self.postshuffle = postshuffle
        self.norm = nn.LayerNorm(hidden if postshuffle else dim, eps=1e-6)
        self.linear_fc1 = nn.Linear(hidden, hidden)
        self.linear_fc2 = nn.Linear(hidden, out_dim)
        self.hidden = hidden

    def forward(self, x):
        if self.postshuffle:
            x = self.norm(x.view(-1, self.hidden))
        else:
            x = self.norm(x).view(-1, self.hidden)
        return self.linear_fc2(F.gelu(self.linear_fc1(x)))


class _VisionBlock(nn.Module):
    def __init__(self, dim: int, heads: int, inter: int, act):
        super().__init__()
        self.heads = heads
        self.head_dim = dim // heads
        self.scale = 1.0 / math.sqrt(self.head_dim)
        self.norm1 = nn.LayerNorm(dim, eps=1e-6)
        self.norm2 = nn.LayerNorm(dim, eps=1e-6)
        self.attn_qkv = nn.Linear(dim, dim * 3, bias=True)
        self.attn_proj = nn.Linear(dim, dim)
        self.mlp_fc1 = nn.Linear(dim, inter, bias=True)
        self.mlp_fc2 = nn.Linear(inter, dim, bias=True)
        self.act = act

    def forward(self, h, mask, cos, sin):
        S = h.shape[0]
        x = self.norm1(h)
        q, k, v = self.attn_qkv(x).reshape(S, 3, self.heads, self.head_dim) \
            .permute(1, 2, 0, 3).unbind(0)
        q, k = ops.rope_apply(q.unsqueeze(0), k.unsqueeze(0), cos, sin)
        out = ops.attention_prefill(q, k, v.unsqueeze(0), mask=mask,
                                    causal=False, scale=self.scale)
        h = h + self.attn_proj(out.squeeze(0).transpose(0, 1).reshape(S, -1))
        return h + self.mlp_fc2(self.act(self.mlp_fc1(self.norm2(h))))


class Qwen3VLVisionModel(nn.Module):
    def __init__(self, vc: dict):
        super().__init__()
        dim = vc["hidden_size"]
        self.dim = dim
        self.patch_size = vc["patch_size"]
        self.temporal_patch_size = vc.get("temporal_patch_size", 2)
        self.spatial_merge_size = vc.get("spatial_merge_size", 2)
        self.in_channels = vc.get("in_channels", 3)
        heads = vc["num_heads"]
        self.head_dim = dim // heads
        act = F.gelu if vc.get("hidden_act") in ("gelu", "gelu_pytorch_tanh") \
            else (lambda x: x * torch.sigmoid(1.702 * x))
        if vc.get("hidden_act") == "gelu_pytorch_tanh":
            act = lambda x: F.gelu(x, approximate="tanh")  # noqa: E731
        k = [self.temporal_patch_size, self.patch_size, self.patch_size]
        self.patch_embed_proj = nn.Conv3d(self.in_channels, dim, kernel_size=k,
                                          stride=k, bias=True)
        self.pos_embed = nn.Embedding(vc["num_position_embeddings"], dim)
        self.num_grid_per_side = int(vc["num_position_embeddings"] ** 0.5)
        self.blocks = nn.ModuleList(
            _VisionBlock(dim, heads, vc["intermediate_size"], act)
            for _ in range(vc["depth"])
        )
        m2 = self.spatial_merge_size ** 2
        out_dim = vc["out_hidden_size"]
        self.merger = _PatchMerger(dim, m2, out_dim, postshuffle=False)
        self.deepstack_visual_indexes = list(vc["deepstack_visual_indexes"])
        self.deepstack_merger_list = nn.ModuleList(
            _PatchMerger(dim, m2, out_dim, postshuffle=True)
            for _ in self.deepstack_visual_indexes
        )
        inv = 1.0 / (10000.0 ** (
            torch.arange(0, self.head_dim // 2, 2, dtype=torch.float)
            / (self.head_dim // 2)))
        self.register_buffer("rope_inv_freq", inv, persistent=False)

    def _position_ids(self, grid_thw) -> torch.Tensor:
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
        """Returns (merged_features, [deepstack_features...])."""
        from transformers.vision_utils import (
            get_vision_interpolation_indices_and_weights,
        )

        x = pixel_values.view(-1, self.in_channels, self.temporal_patch_size,
                              self.patch_size, self.patch_size)
        h = self.patch_embed_proj(x.to(self.patch_embed_proj.weight.dtype))
        h = h.view(-1, self.dim)
        idxs, weights = get_vision_interpolation_indices_and_weights(
            grid_thw, num_grid_per_side=self.num_grid_per_side,
            mode="bilinear", align_corners=True,
            spatial_merge_size=self.spatial_merge_size,
        )
        h = h + (self.pos_embed(idxs) * weights[:, :, None]).sum(1).to(h.dtype)
        pos = self._position_ids(grid_thw).to(h.device)
        freqs = (pos.unsqueeze(-1).float() * self.rope_inv_freq).flatten(1)
        emb = torch.cat([freqs, freqs], dim=-1)
        cos = emb.cos().view(1, -1, emb.shape[-1])
        sin = emb.sin().view(1, -1, emb.shape[-1])
        sizes = [int(t * hh * ww) for t, hh, ww in grid_thw.tolist()]
        S = h.shape[0]
        mask = torch.zeros(S, S, dtype=torch.bool, device=h.device)
        start = 0
        for n in sizes:
            mask[start:start + n, start:start + n] = True
            start += n
        mask = mask.view(1, 1, S, S)
        deepstack = []
        for i, blk in enumerate(self.blocks):
            h = blk(h, mask, cos, sin)
            if i in self.deepstack_visual_indexes:
                deepstack.append(
                    self.deepstack_merger_list[
                        self.deepstack_visual_indexes.index(i)](h))
        return self.merger(h), deepstack


class Qwen3VLForConditionalGeneration(nn.Module):
    """DeepStack ViT + Qwen3 text LM with interleaved m-rope."""

    _config_cls = Qwen3VLInferenceConfig

    def __init__(self, config: InferenceConfig):
        super().__init__()
        self.config = config
        vc = config.vision_config if isinstance(config.vision_config, dict) \
            else config.vision_config.__dict__
        self.visual = Qwen3VLVisionModel(vc)
        self.language_model = Qwen3ForCausalLM(config.text_config)
        tc = config.text_config
        rs = getattr(tc, "rope_scaling", None)
        if not isinstance(rs, dict):
            rs = getattr(tc, "rope_parameters", None) or {}
        head_dim = getattr(tc, "head_dim", None) or \
            tc.hidden_size // tc.num_attention_heads
        self.mrope_section = list(rs.get("mrope_section", None)
                                  or [head_dim // 6] * 3)
        theta = getattr(tc, "rope_theta", 1000000.0)
        inv = 1.0 / (theta ** (torch.arange(0, head_dim, 2).float() / head_dim))
        self.register_buffer("inv_freq", inv, persistent=False)
        self._mrope_ctx: dict = {}
        dm = self.language_model.model
        dm._cos_sin_fn = self._mrope_cos_sin
        # deepstack injection after decoder layers 0..n-1
        n_ds = len(self.visual.deepstack_visual_indexes)
        dm._layer_post_fns = {
            li: self._make_deepstack_fn(li) for li in range(n_ds)
        }

    def _make_deepstack_fn(self, li: int):
        def fn(h, input_ids, phase):
            # [(placeholder_token_id, per-level features), ...] — one entry
            # per modality (image / video), injected at its own positions
            ds = self._mrope_ctx.get("deepstack")
            if ds is None or phase != "context_encoding":
                return h
            for tok, levels in ds:
                mask = (input_ids == tok).unsqueeze(-1)
                add = torch.zeros_like(h).masked_scatter(
                    mask, levels[li].to(h.dtype).reshape(-1, h.shape[-1]))
                h = h + add
            return h
        return fn

    # -- rope index: text +1; vision block advances by max(lh, lw) with
    #    (t=cur, h=cur+row, w=cur+col). Videos are split into per-frame
    #    (1, h, w) grids (HF qwen3-vl: timestamp text tokens separate the
    #    frames, so each frame is its own vision segment).
    def _compute_rope_index(self, input_ids, grid_thw,
                            video_grid_thw=None):
        B, S = input_ids.shape
        image_tok = self.config.image_token_id
        video_tok = getattr(self.config, "video_token_id", None)
        frames = []
        if video_grid_thw is not None:
            for t, h, w in video_grid_thw.tolist():
                frames.extend([[1, h, w]] * t)
        pos3 = torch.zeros(3, B, S, dtype=torch.long)
        deltas = torch.zeros(B, dtype=torch.long)
        m = self.visual.spatial_merge_size
        for b in range(B):
            ids = input_ids[b].tolist()
            img_i, cur, i = 0, 0, 0
            vid_i = 0
            while i < S:
                is_vid = video_tok is not None and ids[i] == video_tok
                if ids[i] == image_tok or is_vid:
                    if is_vid:
                        t, h, w = frames[vid_i]
                        vid_i += 1
                    else:
                        t, h, w = grid_thw[img_i].tolist()
                        img_i += 1
                    lh, lw = h // m, w // m
                    n = t * lh * lw
                    r = torch.arange(lh).view(-1, 1).expand(lh, lw).reshape(-1)
                    c = torch.arange(lw).view(1, -1).expand(lh, lw).reshape(-1)
                    pos3[0, b, i:i + n] = cur
                    pos3[1, b, i:i + n] = cur + r.repeat(t)
                    pos3[2, b, i:i + n] = cur + c.repeat(t)
                    cur += max(lh, lw)
                    i += n
                else:
                    pos3[:, b, i] = cur
                    cur += 1
                    i += 1
            deltas[b] = cur - S
        return pos3, deltas

    def _mrope_cos_sin(self, position_ids: torch.Tensor, device=None):
        """Interleaved m-rope folded into (B,S,head_dim) cos/sin: start from
        the T angles and overwrite slots 1,4,7,... with H and 2,5,8,...
        with W (HF apply_interleaved_mrope)."""
        ctx = self._mrope_ctx
        B, S = position_ids.shape
        dev = device or position_ids.device
        if S > 1 and ctx.get("pos3d") is not None:
            pos3 = ctx["pos3d"].to(dev)
            if pos3.shape[2] < S:
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
        f = freqs[0].clone()  # (B, S, head_dim/2), T everywhere
        for axis, off in ((1, 1), (2, 2)):
            sl = slice(off, self.mrope_section[axis] * 3, 3)
            f[..., sl] = freqs[axis][..., sl]
        emb = torch.cat([f, f], dim=-1)
        return emb.cos().float(), emb.sin().float()

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
        splices, ds = [], []
        if pixel_values is not None:
            feats, deepstack = self.visual(pixel_values, image_grid_thw)
            splices.append((self.config.image_token_id, feats))
            ds.append((self.config.image_token_id, deepstack))
        if pixel_values_videos is not None:
            vfeats, vdeepstack = self.visual(pixel_values_videos,
                                             video_grid_thw)
            splices.append((self.config.video_token_id, vfeats))
            ds.append((self.config.video_token_id, vdeepstack))
        if splices:
            dm.multimodal_ctx = {"splices": splices}
            pos3, deltas = self._compute_rope_index(
                input_ids, image_grid_thw, video_grid_thw)
            self._mrope_ctx = {"pos3d": pos3, "delta": deltas,
                               "deepstack": ds}
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
                nk = nk.replace(".mlp.linear_fc1.", ".mlp_fc1.")
                nk = nk.replace(".mlp.linear_fc2.", ".mlp_fc2.")
                out[nk] = v
        conv = Qwen3ForCausalLM.convert_hf_to_native_state_dict(
            text_sd, config.text_config
        )
        out.update({f"language_model.model.{k}": v for k, v in conv.items()})
        dtype = config.neuron_config.torch_dtype
        return {k: v.to(dtype) for k, v in out.items()}
