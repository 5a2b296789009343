"""DeepSeek-V3 for MI355X (ref `models/deepseek/modeling_deepseek.py`
:79-266 — Multi-head Latent Attention that does NOT subclass the attention
base; yarn rope utils `rope_util.py`).

MLA layout: q through a low-rank a/b projection pair with rmsnorm between;
KV compressed to a (kv_lora_rank + qk_rope_head_dim) latent that IS the KV
cache (tiny — the point of MLA), expanded per step through kv_b_proj.
TP shards the head-dim projections (q_b, kv_b column-parallel; o_proj
row-parallel); the latent projections and the latent cache are replicated.
The absorbed-wkv_b decode trick (ref :142-266) is a round-2 optimization;
this implementation expands the latent (torch GEMMs).

MoE: sigmoid scores + e_score_correction_bias, node-group top-k routing,
routed_scaling_factor, shared experts; first_k_dense_replace dense layers.
"""

from __future__ import annotations

import math
from typing import Dict, List

import torch
import torch.nn as nn
import torch.nn.functional as F

from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.models.llama.modeling_llama import LlamaForCausalLM
from neuronx_distributed_inference_amd.models.model_base import DecoderModel
from neuronx_distributed_inference_amd.modules.attention.attention_base import PHASE_CTE
from neuronx_distributed_inference_amd.modules.kvcache.kv_cache_manager import KVCacheManager
from neuronx_distributed_inference_amd.modules.mlp import GatedMLP
from neuronx_distributed_inference_amd.modules.norm import RMSNorm
from neuronx_distributed_inference_amd.modules.rope import RotaryEmbedding
from neuronx_distributed_inference_amd.ops import ref as ref_ops
from neuronx_distributed_inference_amd.parallel import mappings
from neuronx_distributed_inference_amd.parallel.layers import (
    ColumnParallelLinear,
    RowParallelLinear,
)


class DeepseekV3InferenceConfig(InferenceConfig):
    def get_required_attributes(self) -> List[str]:
        return ["hidden_size", "num_attention_heads", "num_hidden_layers",
                "vocab_size", "intermediate_size", "kv_lora_rank",
                "qk_rope_head_dim", "qk_nope_head_dim", "v_head_dim"]

    def add_derived_config(self):
        if not hasattr(self, "rms_norm_eps"):
            self.rms_norm_eps = 1e-6
        if not hasattr(self, "tie_word_embeddings"):
            self.tie_word_embeddings = False
        if not hasattr(self, "first_k_dense_replace"):
            self.first_k_dense_replace = 0
        if not hasattr(self, "rope_interleave"):
            self.rope_interleave = True
        if not hasattr(self, "attention_bias"):
            self.attention_bias = False
        rp = getattr(self, "rope_parameters", None) or {}
        self.rope_theta = rp.get("rope_theta", getattr(self, "rope_theta", 10000.0))
        self.rope_scaling = rp if rp.get("rope_type", "default") != "default" else None
        self.qk_head_dim = self.qk_nope_head_dim + self.qk_rope_head_dim
        # our base builds a kv manager from num_key_value_heads/head_dim; the
        # MLA model replaces it, but keep the attrs coherent
        self.num_key_value_heads = getattr(self, "num_key_value_heads", self.num_attention_heads)


def _rope_interleave(q, k, cos, sin):
    # cos/sin (B, S, D) = cat(freqs, freqs); first half holds per-pair angle
    d = cos.shape[-1] // 2
    c = cos[..., :d].unsqueeze(1)
    s = sin[..., :d].unsqueeze(1)
    q1, q2 = q[..., 0::2], q[..., 1::2]
    k1, k2 = k[..., 0::2], k[..., 1::2]
    q_out = torch.cat([q1 * c - q2 * s, q2 * c + q1 * s], dim=-1)
    k_out = torch.cat([k1 * c - k2 * s, k2 * c + k1 * s], dim=-1)
    return q_out.to(q.dtype), k_out.to(k.dtype)


class DeepseekV3Attention(nn.Module):
    def __init__(self, config: InferenceConfig, layer_idx: int):
        super().__init__()
        ec = config.neuron_config
        dtype = ec.torch_dtype
        self.layer_idx = layer_idx
        self.config = config
        self.num_heads = config.num_attention_heads
        self.tp = ec.tp_degree
        self.heads_per_rank = self.num_heads // self.tp
        self.q_lora_rank = getattr(config, "q_lora_rank", None)
        self.kv_lora_rank = config.kv_lora_rank
        self.rope_dim = config.qk_rope_head_dim
        self.nope_dim = config.qk_nope_head_dim
        self.v_dim = config.v_head_dim
        self.qk_head_dim = self.nope_dim + self.rope_dim
        bias = bool(config.attention_bias)

        H = config.hidden_size
        if self.q_lora_rank is None:
            self.q_proj = ColumnParallelLinear(
                H, self.num_heads * self.qk_head_dim, bias=False, dtype=dtype,
                tp_degree=self.tp)
        else:
            self.q_a_proj = nn.Linear(H, self.q_lora_rank, bias=bias, dtype=dtype)
            self.q_a_layernorm = RMSNorm(self.q_lora_rank, config.rms_norm_eps, dtype=dtype)
            self.q_b_proj = ColumnParallelLinear(
                self.q_lora_rank, self.num_heads * self.qk_head_dim, bias=False,
                dtype=dtype, tp_degree=self.tp)
        self.kv_a_proj_with_mqa = nn.Linear(
            H, self.kv_lora_rank + self.rope_dim, bias=bias, dtype=dtype)
        self.kv_a_layernorm = RMSNorm(self.kv_lora_rank, config.rms_norm_eps, dtype=dtype)
        self.kv_b_proj = ColumnParallelLinear(
            self.kv_lora_rank, self.num_heads * (self.nope_dim + self.v_dim),
            bias=False, dtype=dtype, tp_degree=self.tp)
        self.o_proj = RowParallelLinear(
            self.num_heads * self.v_dim, H, bias=bias, dtype=dtype, tp_degree=self.tp)

        scaling = self.qk_head_dim ** -0.5
        rp = getattr(config, "rope_parameters", None) or {}
        if rp.get("rope_type", "default") != "default" and rp.get("mscale_all_dim"):
            f = rp["factor"]
            ms = 0.1 * rp["mscale_all_dim"] * math.log(f) + 1.0 if f > 1 else 1.0
            scaling = scaling * ms * ms
        self.scaling = scaling
        self.interleave = bool(getattr(config, "rope_interleave", True))

    def forward(self, h, cos, sin, kv_mgr, position_ids, phase, attention_mask,
                seq_ids):
        B, S, _ = h.shape
        if self.q_lora_rank is None:
            q = self.q_proj(h)
        else:
            q = self.q_b_proj(self.q_a_layernorm(self.q_a_proj(h)))
        q = q.view(B, S, self.heads_per_rank, self.qk_head_dim).transpose(1, 2)
        q_pass, q_rot = torch.split(q, [self.nope_dim, self.rope_dim], dim=-1)

        ckv = self.kv_a_proj_with_mqa(h)
        kv_nope, k_rot = torch.split(ckv, [self.kv_lora_rank, self.rope_dim], dim=-1)
        kv_nope = self.kv_a_layernorm(kv_nope).view(B, 1, S, self.kv_lora_rank)
        k_rot = k_rot.view(B, 1, S, self.rope_dim)

        if self.interleave:
            q_rot, k_rot = _rope_interleave(q_rot, k_rot, cos, sin)
        else:
            q_rot, k_rot = ref_ops.rope_apply(q_rot, k_rot, cos, sin)

        # the compressed latent IS the cache entry
        latent = torch.cat([kv_nope, k_rot], dim=-1)        # (B,1,S,rank+rope)
        kv_mgr.update_cache(self.layer_idx, latent, latent, position_ids, seq_ids)

        q_states = torch.cat([q_pass, q_rot], dim=-1)       # (B,hpr,S,qk_dim)

        if phase == PHASE_CTE:
            k, v = self._expand(kv_nope, k_rot)
            mask = None
            Sq = S
            if attention_mask is not None:
                from neuronx_distributed_inference_amd.modules.attention.utils import (
                    padded_causal_mask,
                )
                mask = padded_causal_mask(attention_mask, Sq)
            else:
                from neuronx_distributed_inference_amd.modules.attention.utils import (
                    causal_mask,
                )
                mask = causal_mask(Sq, Sq, device=h.device)
            out = ref_ops.sdpa(q_states, k, v, mask=mask, scale=self.scaling)
        else:
            # ABSORBED decode (ref modeling_deepseek.py:142-266 wkv_b trick):
            # fold W_uk into q so attention runs directly over the latent
            # cache — score = (W_uk^T q_nope)·c + q_rope·k_rope, out =
            # W_uv (p·c). No per-step K/V expansion, no host sync; pure
            # batched library GEMMs over (rank+rope)-dim rows, graph-safe.
            kc, _ = kv_mgr.get_cache(self.layer_idx)
            if seq_ids is not None:
                lat = kc[seq_ids.long()].to(h.dtype)
            else:
                lat = kc[:B].to(h.dtype)                    # (B,1,Smax,D)
            lat = lat.squeeze(1)                            # (B,Smax,D)
            w = self.kv_b_proj.weight.view(
                self.heads_per_rank, self.nope_dim + self.v_dim, self.kv_lora_rank)
            w_uk = w[:, : self.nope_dim]                    # (hpr, nope, rank)
            w_uv = w[:, self.nope_dim:]                     # (hpr, v, rank)
            q_lat = torch.einsum("bhsd,hdr->bhsr", q_pass, w_uk)
            q_abs = torch.cat([q_lat, q_rot], dim=-1)       # (B,hpr,S,rank+rope)
            scores = torch.einsum("bhsr,btr->bhst", q_abs.float(), lat.float())
            Smax = lat.shape[1]
            ar = torch.arange(Smax, device=h.device).view(1, 1, 1, Smax)
            qpos = position_ids.view(B, 1, S, 1)            # per-query positions
            scores = (scores * self.scaling).masked_fill(ar > qpos, float("-inf"))
            p = torch.softmax(scores, dim=-1)
            p = torch.nan_to_num(p)                          # pad queries
            o_lat = torch.einsum("bhst,btr->bhsr", p,
                                 lat[..., : self.kv_lora_rank].float())
            out = torch.einsum("bhsr,hvr->bhsv", o_lat,
                               w_uv.float()).to(h.dtype)

        out = out.transpose(1, 2).reshape(B, S, -1)
        return self.o_proj(out)

    def _expand(self, kv_nope, k_rot):
        """(B,1,S,rank)+(B,1,S,rope) -> k (B,hpr,S,qk_dim), v (B,hpr,S,v_dim)."""
        B, _, S, _ = kv_nope.shape
        kv = self.kv_b_proj(kv_nope).view(B, S, self.heads_per_rank,
                                          self.nope_dim + self.v_dim).transpose(1, 2)
        k_nope, v = torch.split(kv, [self.nope_dim, self.v_dim], dim=-1)
        k_rot = k_rot.expand(-1, self.heads_per_rank, -1, -1)
        k = torch.cat([k_nope, k_rot], dim=-1)
        return k, v


class DeepseekV3Gate(nn.Module):
    """Sigmoid-scored, node-group-limited top-k router
    (ref DeepseekV3TopkRouter)."""

    def __init__(self, config):
        super().__init__()
        E = config.n_routed_experts
        self.weight = nn.Parameter(torch.empty(E, config.hidden_size))
        self.e_score_correction_bias = nn.Parameter(torch.zeros(E))
        self.top_k = config.num_experts_per_tok
        self.n_group = getattr(config, "n_group", 1)
        self.topk_group = getattr(config, "topk_group", 1)
        self.num_experts = E
        self.norm_topk_prob = getattr(config, "norm_topk_prob", True)
        self.routed_scaling_factor = getattr(config, "routed_scaling_factor", 1.0)

    def forward(self, flat):
        logits = F.linear(flat.float(), self.weight.float())
        scores = logits.sigmoid()
        choice = scores + self.e_score_correction_bias
        g = self.n_group
        group_scores = choice.view(-1, g, self.num_experts // g).topk(
            min(2, self.num_experts // g), dim=-1)[0].sum(dim=-1)
        group_idx = torch.topk(group_scores, k=self.topk_group, dim=-1, sorted=False)[1]
        group_mask = torch.zeros_like(group_scores)
        group_mask.scatter_(1, group_idx, 1)
        score_mask = group_mask.unsqueeze(-1).expand(
            -1, g, self.num_experts // g).reshape(-1, self.num_experts)
        choice = choice.masked_fill(~score_mask.bool(), float("-inf"))
        idx = torch.topk(choice, k=self.top_k, dim=-1, sorted=False)[1]
        weights = scores.gather(1, idx)
        if self.norm_topk_prob:
            weights = weights / (weights.sum(dim=-1, keepdim=True) + 1e-20)
        weights = weights * self.routed_scaling_factor
        return weights, idx


class DeepseekV3MoE(nn.Module):
    def __init__(self, config):
        super().__init__()
        ec = config.neuron_config
        from neuronx_distributed_inference_amd.modules.moe import ExpertMLPs

        self.gate = DeepseekV3Gate(config)
        self.experts = ExpertMLPs(
            config.n_routed_experts, config.hidden_size,
            config.moe_intermediate_size, ec.tp_degree, ec.torch_dtype,
            hidden_act="silu", ep_degree=ec.ep_degree,
        )
        n_shared = getattr(config, "n_shared_experts", 0) or 0
        self.shared_experts = None
        if n_shared:
            self.shared_experts = GatedMLP(
                config.hidden_size, config.moe_intermediate_size * n_shared,
                tp_degree=ec.tp_degree, dtype=ec.torch_dtype, hidden_act="silu",
            )

    def forward(self, h):
        B, S, H = h.shape
        flat = h.view(-1, H)
        weights, idx = self.gate(flat)
        out = self.experts(flat, weights, idx)
        out = mappings.reduce_from_tensor_parallel_region(out).view(B, S, H)
        if self.shared_experts is not None:
            out = out + self.shared_experts(h)
        return out


class DeepseekV3DecoderLayer(nn.Module):
    def __init__(self, config, layer_idx):
        super().__init__()
        ec = config.neuron_config
        self.self_attn = DeepseekV3Attention(config, layer_idx)
        if layer_idx < config.first_k_dense_replace or not hasattr(config, "n_routed_experts"):
            self.mlp = GatedMLP(
                config.hidden_size, config.intermediate_size,
                tp_degree=ec.tp_degree, dtype=ec.torch_dtype, hidden_act="silu",
                fused_gate_up=ec.fused_gate_up,
            )
            self._moe = False
        else:
            self.mlp = DeepseekV3MoE(config)
            self._moe = True
        self.input_layernorm = RMSNorm(config.hidden_size, config.rms_norm_eps,
                                       dtype=ec.torch_dtype)
        self.post_attention_layernorm = RMSNorm(config.hidden_size, config.rms_norm_eps,
                                                dtype=ec.torch_dtype)

    def forward(self, h, residual, cos, sin, kv_mgr, position_ids, phase,
                attention_mask, seq_ids, sequence_parallel=False):
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


class DeepseekV3Model(DecoderModel):
    def build_layers(self, config):
        return nn.ModuleList(
            DeepseekV3DecoderLayer(config, i) for i in range(config.num_hidden_layers)
        )

    def build_rotary(self, config):
        return RotaryEmbedding(
            config.qk_rope_head_dim,
            max_position_embeddings=getattr(config, "max_position_embeddings", 4096),
            base=config.rope_theta,
            scaling=getattr(config, "rope_scaling", None),
        )

    def __init__(self, config):
        super().__init__(config)
        ec = config.neuron_config
        # replace the dense KV manager with the latent cache (Hkv=1,
        # D = kv_lora_rank + qk_rope_head_dim, replicated on every rank)
        self.kv_mgr = KVCacheManager(
            num_layers=config.num_hidden_layers,
            num_kv_heads=1,
            head_dim=config.kv_lora_rank + config.qk_rope_head_dim,
            max_len=ec.max_length,
            batch_size=ec.kv_cache_batch_size,
            dtype=torch.float8_e4m3fn if ec.kv_cache_quant else ec.torch_dtype,
            tp_degree=1,
        )


class DeepseekV3ForCausalLM(LlamaForCausalLM):
    _model_cls = DeepseekV3Model
    _config_cls = DeepseekV3InferenceConfig

    @staticmethod
    def convert_hf_to_native_state_dict(
        sd: Dict[str, torch.Tensor], config: InferenceConfig
    ) -> Dict[str, torch.Tensor]:
        out: Dict[str, torch.Tensor] = {}
        for k, v in sd.items():
            nk = k[len("model."):] if k.startswith("model.") else k
            # MLA projections keep HF names; experts stack like qwen3-moe
            if ".mlp.experts." in nk and nk.split(".")[-2].isdigit() is False:
                pass
            out[nk] = v
        # stack per-expert tensors
        E = getattr(config, "n_routed_experts", 0)
        if E:
            stacked: Dict[str, list] = {}
            drop = []
            for k, v in list(out.items()):
                if ".mlp.experts." in k:
                    parts = k.split(".")
                    if parts[parts.index("experts") + 1].isdigit():
                        e = int(parts[parts.index("experts") + 1])
                        wname = parts[-2]
                        base = ".".join(parts[: parts.index("experts")])
                        tgt = {"gate_proj": "w_gate", "up_proj": "w_up",
                               "down_proj": "w_down"}[wname]
                        stacked.setdefault(f"{base}.experts.{tgt}", [None] * E)[e] = v
                        drop.append(k)
                    elif k.endswith(".gate_up_proj"):
                        inter = config.moe_intermediate_size
                        g, u = v.split([inter, inter], dim=1)
                        out[k.replace(".gate_up_proj", ".w_gate")] = g.contiguous()
                        out[k.replace(".gate_up_proj", ".w_up")] = u.contiguous()
                        drop.append(k)
                    elif k.endswith(".down_proj"):
                        out[k.replace(".down_proj", ".w_down")] = v
                        drop.append(k)
            for k in drop:
                out.pop(k, None)
            for key, tensors in stacked.items():
                assert all(t is not None for t in tensors), f"missing experts {key}"
                out[key] = torch.stack(tensors, dim=0)
        if "lm_head.weight" not in out and getattr(config, "tie_word_embeddings", False):
            out["lm_head.weight"] = out["embed_tokens.weight"]
        dtype = config.neuron_config.torch_dtype
        conv = {}
        for k, v in out.items():
            if k.endswith("e_score_correction_bias") or k.endswith(".gate.weight"):
                conv[k] = v.float()
            else:
                conv[k] = v.to(dtype)
        return conv
