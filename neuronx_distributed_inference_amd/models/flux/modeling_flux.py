"""Flux (rectified-flow MMDiT) text-to-image transformer + pipeline.

Parity target: the reference's `models/diffusers/flux/` (modeling_flux.py
1511 LoC + pipeline.py). Behavior matched against the
reference's blocks (modeling_flux.py:535-760, embeddings.py:141-330,
normalization.py:35-187) — written fresh for MI355X on this framework's
TP layers; module names mirror the diffusers checkpoint layout so real
Flux weights load by direct key mapping.

Architecture:
- x_embedder (packed 2x2 latent patches -> inner), context_embedder
  (T5 states -> inner), combined timestep+guidance+pooled-CLIP embedding
  (sinusoidal 256 -> silu MLPs, summed);
- 3-axis rope over cat(txt_ids, img_ids), per-axis cos/sin
  repeat-interleaved pairs;
- N double-stream MMDiT blocks: AdaLN-Zero per stream, joint attention
  over [text; image] with per-head RMS q/k norms, gated residuals, gated
  gelu-tanh FFNs;
- M single-stream blocks over the concatenated sequence: AdaLN-Zero
  single, fused attention + gelu-tanh MLP through one output projection;
- AdaLN-continuous out norm + proj to the patch channels.
- FlowMatchEuler scheduler with Flux's dynamic time shift;
  FluxPipeline.denoise() runs the loop from prompt embeddings (T5/CLIP
  encoders and the VAE are external: `transformers`' T5EncoderModel /
  CLIPTextModel produce the embeddings; diffusers' VAE decodes — neither
  ships in this image, so the pipeline returns LATENTS and the tests
  verify structure/determinism, not a diffusers golden. GAPS.md).
"""

from __future__ import annotations

import math
from typing import List, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.parallel.layers import (
    ColumnParallelLinear,
    RowParallelLinear,
)


class FluxInferenceConfig(InferenceConfig):
    def get_required_attributes(self) -> List[str]:
        return ["num_layers", "num_single_layers", "attention_head_dim",
                "num_attention_heads", "in_channels", "joint_attention_dim",
                "pooled_projection_dim"]

    def add_derived_config(self):
        if not hasattr(self, "axes_dims_rope"):
            self.axes_dims_rope = [16, 56, 56]
        if not hasattr(self, "guidance_embeds"):
            self.guidance_embeds = True
        if not hasattr(self, "rope_theta"):
            self.rope_theta = 10000


def timestep_sinusoidal(t: torch.Tensor, dim: int = 256) -> torch.Tensor:
    """diffusers get_timestep_embedding, flip_sin_to_cos=True, shift=0."""
    half = dim // 2
    exponent = -math.log(10000) * torch.arange(half, dtype=torch.float32,
                                               device=t.device) / half
    emb = t.float()[:, None] * exponent.exp()[None, :]
    return torch.cat([emb.cos(), emb.sin()], dim=-1)


class TimestepEmbedding(nn.Module):
    def __init__(self, dim: int, tp: int, dtype):
        super().__init__()
        self.linear_1 = ColumnParallelLinear(256, dim, bias=True, dtype=dtype,
                                             tp_degree=tp)
        self.linear_2 = RowParallelLinear(dim, dim, bias=True, dtype=dtype,
                                          tp_degree=tp)

    def forward(self, x):
        return self.linear_2(F.silu(self.linear_1(x)))


class TextProjection(nn.Module):
    def __init__(self, in_dim: int, dim: int, tp: int, dtype):
        super().__init__()
        self.linear_1 = ColumnParallelLinear(in_dim, dim, bias=True, dtype=dtype,
                                             tp_degree=tp)
        self.linear_2 = RowParallelLinear(dim, dim, bias=True, dtype=dtype,
                                          tp_degree=tp)

    def forward(self, x):
        return self.linear_2(F.silu(self.linear_1(x)))


class CombinedTimestepEmbed(nn.Module):
    """timestep + (optional) guidance + pooled text -> temb
    (ref embeddings.py:240-300)."""

    def __init__(self, dim: int, pooled_dim: int, guidance: bool, tp: int, dtype):
        super().__init__()
        self.timestep_embedder = TimestepEmbedding(dim, tp, dtype)
        self.guidance_embedder = TimestepEmbedding(dim, tp, dtype) if guidance else None
        self.text_embedder = TextProjection(pooled_dim, dim, tp, dtype)

    def forward(self, timestep, guidance, pooled):
        emb = self.timestep_embedder(timestep_sinusoidal(timestep).to(pooled.dtype))
        if self.guidance_embedder is not None and guidance is not None:
            emb = emb + self.guidance_embedder(
                timestep_sinusoidal(guidance).to(pooled.dtype))
        return emb + self.text_embedder(pooled)


def rope_3axis(ids: torch.Tensor, axes_dim: List[int], theta: float):
    """(L, n_axes) ids -> cos/sin (L, head_dim), per-axis pairs
    repeat-interleaved (ref embeddings.py FluxPosEmbed/get_1d_rotary)."""
    cos_l, sin_l = [], []
    for i, d in enumerate(axes_dim):
        freqs = 1.0 / (theta ** (torch.arange(0, d, 2, dtype=torch.float64,
                                              device=ids.device) / d))
        ang = torch.outer(ids[:, i].double(), freqs)
        cos_l.append(ang.cos().repeat_interleave(2, dim=1).float())
        sin_l.append(ang.sin().repeat_interleave(2, dim=1).float())
    return torch.cat(cos_l, dim=-1), torch.cat(sin_l, dim=-1)


def apply_rope_pairs(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor):
    """Interleaved-pair rotation with (L, D) tables (ref apply_rotary_emb
    use_real_unbind_dim=-1)."""
    xr, xi = x.reshape(*x.shape[:-1], -1, 2).unbind(-1)
    rot = torch.stack([-xi, xr], dim=-1).flatten(-2)
    return (x.float() * cos[None, None] + rot.float() * sin[None, None]).to(x.dtype)


class HeadRMSNorm(nn.Module):
    """Per-head RMS norm with weight (diffusers RMSNorm, eps 1e-6)."""

    def __init__(self, dim: int, dtype):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim, dtype=dtype))

    def forward(self, x):
        xf = x.float()
        out = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-6)
        return (out * self.weight.float()).to(x.dtype)


class AdaLayerNormZero(nn.Module):
    def __init__(self, dim: int, tp: int, dtype, n_chunks: int = 6):
        super().__init__()
        self.linear = ColumnParallelLinear(dim, n_chunks * dim, bias=True,
                                           dtype=dtype, tp_degree=tp,
                                           gather_output=True)
        self.norm = nn.LayerNorm(dim, elementwise_affine=False, eps=1e-6)
        self.n = n_chunks

    def forward(self, x, emb):
        parts = self.linear(F.silu(emb)).chunk(self.n, dim=1)
        shift, scale = parts[0], parts[1]
        x = self.norm(x) * (1 + scale[:, None]) + shift[:, None]
        return (x,) + parts[2:]


class FeedForward(nn.Module):
    """gelu-approximate MLP; diffusers names net.0.proj / net.2."""

    def __init__(self, dim: int, tp: int, dtype, ratio: float = 4.0):
        super().__init__()
        inner = int(dim * ratio)
        self.net = nn.ModuleDict({
            "0": nn.ModuleDict({"proj": ColumnParallelLinear(
                dim, inner, bias=True, dtype=dtype, tp_degree=tp)}),
            "2": RowParallelLinear(inner, dim, bias=True, dtype=dtype,
                                   tp_degree=tp),
        })

    def forward(self, x):
        return self.net["2"](F.gelu(self.net["0"]["proj"](x), approximate="tanh"))


class FluxJointAttention(nn.Module):
    """Double-stream joint attention (ref modeling_flux.py:831-1270)."""

    def __init__(self, dim: int, heads: int, head_dim: int, tp: int, dtype):
        super().__init__()
        self.heads_per_rank = heads // tp
        self.head_dim = head_dim
        inner = heads * head_dim
        mk = lambda i, o: ColumnParallelLinear(i, o, bias=True, dtype=dtype,  # noqa: E731
                                               tp_degree=tp)
        self.to_q, self.to_k, self.to_v = mk(dim, inner), mk(dim, inner), mk(dim, inner)
        self.add_q_proj, self.add_k_proj = mk(dim, inner), mk(dim, inner)
        self.add_v_proj = mk(dim, inner)
        self.norm_q = HeadRMSNorm(head_dim, dtype)
        self.norm_k = HeadRMSNorm(head_dim, dtype)
        self.norm_added_q = HeadRMSNorm(head_dim, dtype)
        self.norm_added_k = HeadRMSNorm(head_dim, dtype)
        self.to_out = nn.ModuleDict({"0": RowParallelLinear(
            inner, dim, bias=True, dtype=dtype, tp_degree=tp)})
        self.to_add_out = RowParallelLinear(inner, dim, bias=True, dtype=dtype,
                                            tp_degree=tp)

    def _shape(self, t, B):
        return t.view(B, -1, self.heads_per_rank, self.head_dim).transpose(1, 2)

    def forward(self, img, txt, cos, sin):
        B = img.shape[0]
        q = self.norm_q(self._shape(self.to_q(img), B))
        k = self.norm_k(self._shape(self.to_k(img), B))
        v = self._shape(self.to_v(img), B)
        qc = self.norm_added_q(self._shape(self.add_q_proj(txt), B))
        kc = self.norm_added_k(self._shape(self.add_k_proj(txt), B))
        vc = self._shape(self.add_v_proj(txt), B)
        q = torch.cat([qc, q], dim=2)
        k = torch.cat([kc, k], dim=2)
        v = torch.cat([vc, v], dim=2)
        q = apply_rope_pairs(q, cos, sin)
        k = apply_rope_pairs(k, cos, sin)
        out = F.scaled_dot_product_attention(q, k, v)
        out = out.transpose(1, 2).reshape(B, -1, self.heads_per_rank * self.head_dim)
        Lt = txt.shape[1]
        return self.to_out["0"](out[:, Lt:]), self.to_add_out(out[:, :Lt])


class FluxTransformerBlock(nn.Module):
    """Double-stream MMDiT block (ref modeling_flux.py:640-760)."""

    def __init__(self, dim: int, heads: int, head_dim: int, tp: int, dtype):
        super().__init__()
        self.norm1 = AdaLayerNormZero(dim, tp, dtype)
        self.norm1_context = AdaLayerNormZero(dim, tp, dtype)
        self.attn = FluxJointAttention(dim, heads, head_dim, tp, dtype)
        self.norm2 = nn.LayerNorm(dim, elementwise_affine=False, eps=1e-6)
        self.norm2_context = nn.LayerNorm(dim, elementwise_affine=False, eps=1e-6)
        self.ff = FeedForward(dim, tp, dtype)
        self.ff_context = FeedForward(dim, tp, dtype)

    def forward(self, img, txt, temb, cos, sin):
        n_img, g_msa, s_mlp, sc_mlp, g_mlp = self.norm1(img, temb)
        n_txt, cg_msa, cs_mlp, csc_mlp, cg_mlp = self.norm1_context(txt, temb)
        a_img, a_txt = self.attn(n_img, n_txt, cos, sin)
        img = img + g_msa[:, None] * a_img
        h = self.norm2(img) * (1 + sc_mlp[:, None]) + s_mlp[:, None]
        img = img + g_mlp[:, None] * self.ff(h)
        txt = txt + cg_msa[:, None] * a_txt
        h = self.norm2_context(txt) * (1 + csc_mlp[:, None]) + cs_mlp[:, None]
        txt = txt + cg_mlp[:, None] * self.ff_context(h)
        return img, txt


class FluxSingleTransformerBlock(nn.Module):
    """Single-stream block: fused attn + mlp (ref modeling_flux.py:535-640)."""

    def __init__(self, dim: int, heads: int, head_dim: int, tp: int, dtype,
                 mlp_ratio: float = 4.0):
        super().__init__()
        self.heads_per_rank = heads // tp
        self.head_dim = head_dim
        inner = heads * head_dim
        self.mlp_hidden = int(dim * mlp_ratio)
        self.norm_linear = ColumnParallelLinear(dim, 3 * dim, bias=True,
                                                dtype=dtype, tp_degree=tp,
                                                gather_output=True)
        self.norm_ln = nn.LayerNorm(dim, elementwise_affine=False, eps=1e-6)
        self.proj_mlp = ColumnParallelLinear(dim, self.mlp_hidden, bias=True,
                                             dtype=dtype, tp_degree=tp)
        mk = lambda i, o: ColumnParallelLinear(i, o, bias=True, dtype=dtype,  # noqa: E731
                                               tp_degree=tp)
        self.attn_to_q, self.attn_to_k = mk(dim, inner), mk(dim, inner)
        self.attn_to_v = mk(dim, inner)
        self.attn_norm_q = HeadRMSNorm(head_dim, dtype)
        self.attn_norm_k = HeadRMSNorm(head_dim, dtype)
        self.proj_out = RowParallelLinear(inner + self.mlp_hidden, dim,
                                          bias=True, dtype=dtype, tp_degree=tp)

    def forward(self, h, temb, cos, sin):
        B = h.shape[0]
        shift, scale, gate = self.norm_linear(F.silu(temb)).chunk(3, dim=1)
        x = self.norm_ln(h) * (1 + scale[:, None]) + shift[:, None]
        mlp = F.gelu(self.proj_mlp(x), approximate="tanh")
        sh = lambda t: t.view(B, -1, self.heads_per_rank, self.head_dim).transpose(1, 2)  # noqa: E731
        q = apply_rope_pairs(self.attn_norm_q(sh(self.attn_to_q(x))), cos, sin)
        k = apply_rope_pairs(self.attn_norm_k(sh(self.attn_to_k(x))), cos, sin)
        v = sh(self.attn_to_v(x))
        a = F.scaled_dot_product_attention(q, k, v).transpose(1, 2).reshape(
            B, -1, self.heads_per_rank * self.head_dim)
        return h + gate[:, None] * self.proj_out(torch.cat([a, mlp], dim=-1))


class AdaLayerNormContinuous(nn.Module):
    """(ref normalization.py:147-187) — NOTE: chunk order is (scale, shift)."""

    def __init__(self, dim: int, tp: int, dtype):
        super().__init__()
        self.linear = ColumnParallelLinear(dim, 2 * dim, bias=True, dtype=dtype,
                                           tp_degree=tp, gather_output=True)
        self.norm = nn.LayerNorm(dim, elementwise_affine=False, eps=1e-6)

    def forward(self, x, emb):
        scale, shift = self.linear(F.silu(emb)).chunk(2, dim=1)
        return self.norm(x) * (1 + scale[:, None]) + shift[:, None]


class FluxTransformer2DModel(nn.Module):
    _config_cls = FluxInferenceConfig

    def __init__(self, config: InferenceConfig):
        super().__init__()
        ec = config.neuron_config
        tp, dtype = ec.tp_degree, ec.torch_dtype
        self.config = config
        heads, hd = config.num_attention_heads, config.attention_head_dim
        dim = heads * hd
        self.inner_dim = dim
        self.x_embedder = nn.Linear(config.in_channels, dim)
        self.context_embedder = nn.Linear(config.joint_attention_dim, dim)
        self.time_text_embed = CombinedTimestepEmbed(
            dim, config.pooled_projection_dim, config.guidance_embeds, tp, dtype)
        self.transformer_blocks = nn.ModuleList(
            FluxTransformerBlock(dim, heads, hd, tp, dtype)
            for _ in range(config.num_layers))
        self.single_transformer_blocks = nn.ModuleList(
            FluxSingleTransformerBlock(dim, heads, hd, tp, dtype)
            for _ in range(config.num_single_layers))
        self.norm_out = AdaLayerNormContinuous(dim, tp, dtype)
        self.proj_out = nn.Linear(dim, config.in_channels)
        self.axes_dim = list(config.axes_dims_rope)
        self.theta = float(config.rope_theta)

    def forward(self, hidden_states, encoder_hidden_states, pooled_projections,
                timestep, img_ids, txt_ids, guidance=None):
        # diffusers convention: the pipeline passes sigma; the transformer
        # scales by 1000 before the sinusoidal projection
        timestep = timestep.float() * 1000.0
        if guidance is not None:
            guidance = guidance.float() * 1000.0
        img = self.x_embedder(hidden_states)
        txt = self.context_embedder(encoder_hidden_states)
        temb = self.time_text_embed(timestep, guidance, pooled_projections)
        ids = torch.cat([txt_ids, img_ids], dim=0)
        cos, sin = rope_3axis(ids, self.axes_dim, self.theta)
        cos, sin = cos.to(img.device), sin.to(img.device)
        for blk in self.transformer_blocks:
            img, txt = blk(img, txt, temb, cos, sin)
        h = torch.cat([txt, img], dim=1)
        for blk in self.single_transformer_blocks:
            h = blk(h, temb, cos, sin)
        h = h[:, txt.shape[1]:]
        return self.proj_out(self.norm_out(h, temb))

    @staticmethod
    def convert_hf_to_native_state_dict(sd, config):
        """Diffusers FluxTransformer2DModel checkpoint -> native names."""
        out = {}
        for k, v in sd.items():
            nk = k
            nk = nk.replace(".norm.linear.", ".norm_linear.")   # single blocks
            nk = nk.replace(".attn.to_q.", ".attn_to_q.")
            nk = nk.replace(".attn.to_k.", ".attn_to_k.")
            nk = nk.replace(".attn.to_v.", ".attn_to_v.")
            nk = nk.replace(".attn.norm_q.", ".attn_norm_q.")
            nk = nk.replace(".attn.norm_k.", ".attn_norm_k.")
            out[nk] = v.to(config.neuron_config.torch_dtype)
        return out


# ----------------------------------------------------------------------
class FlowMatchEulerScheduler:
    """Rectified-flow Euler with Flux's dynamic time shift
    (diffusers FlowMatchEulerDiscreteScheduler behavior; ref pipeline
    calculate_shift/retrieve_timesteps)."""

    def __init__(self, base_shift=0.5, max_shift=1.15,
                 base_seq_len=256, max_seq_len=4096):
        self.base_shift, self.max_shift = base_shift, max_shift
        self.base_seq_len, self.max_seq_len = base_seq_len, max_seq_len

    def sigmas_for(self, num_steps: int, seq_len: int) -> torch.Tensor:
        sigmas = torch.linspace(1.0, 1.0 / num_steps, num_steps)
        m = (self.max_shift - self.base_shift) / (self.max_seq_len - self.base_seq_len)
        mu = m * seq_len + (self.base_shift - m * self.base_seq_len)
        shifted = math.exp(mu) / (math.exp(mu) + (1.0 / sigmas - 1.0))
        return torch.cat([shifted, torch.zeros(1)])

    @staticmethod
    def step(latents, velocity, sigma, sigma_next):
        return latents + (sigma_next - sigma) * velocity.float()


class FluxPipeline:
    """Denoising loop from prompt embeddings to PACKED latents.

    T5/CLIP prompt encoding and VAE decode are external (transformers /
    diffusers); this pipeline owns the MI355X-hot part: the transformer
    denoise loop."""

    def __init__(self, transformer: FluxTransformer2DModel,
                 scheduler: Optional[FlowMatchEulerScheduler] = None):
        self.transformer = transformer
        self.scheduler = scheduler or FlowMatchEulerScheduler()

    @staticmethod
    def prepare_ids(height_lat: int, width_lat: int, txt_len: int, device):
        h, w = height_lat // 2, width_lat // 2
        img_ids = torch.zeros(h, w, 3, device=device)
        img_ids[..., 1] = torch.arange(h, device=device).view(-1, 1)
        img_ids[..., 2] = torch.arange(w, device=device).view(1, -1)
        return img_ids.view(-1, 3), torch.zeros(txt_len, 3, device=device)

    @torch.inference_mode()
    def denoise(self, prompt_embeds: torch.Tensor, pooled_embeds: torch.Tensor,
                height_lat: int = 64, width_lat: int = 64,
                num_steps: int = 4, guidance_scale: float = 3.5,
                generator: Optional[torch.Generator] = None) -> torch.Tensor:
        dev = prompt_embeds.device
        B = prompt_embeds.shape[0]
        L = (height_lat // 2) * (width_lat // 2)
        C = self.transformer.config.in_channels
        dtype = next(self.transformer.parameters()).dtype
        latents = torch.randn(B, L, C, generator=generator,
                              dtype=torch.float32, device=dev)
        img_ids, txt_ids = self.prepare_ids(height_lat, width_lat,
                                            prompt_embeds.shape[1], dev)
        sigmas = self.scheduler.sigmas_for(num_steps, L)
        guidance = torch.full((B,), guidance_scale, device=dev) \
            if self.transformer.config.guidance_embeds else None
        for i in range(num_steps):
            t = torch.full((B,), sigmas[i].item(), device=dev)
            v = self.transformer(latents.to(dtype), prompt_embeds,
                                 pooled_embeds, t, img_ids, txt_ids,
                                 guidance=guidance)
            latents = self.scheduler.step(latents, v, sigmas[i].item(),
                                          sigmas[i + 1].item())
        return latents
