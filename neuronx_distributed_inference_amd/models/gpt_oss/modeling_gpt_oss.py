"""GPT-OSS for MI355X (ref `models/gpt_oss/`, 1984 LoC: attention sinks,
mixed sliding/full attention layers, biased projections, clamped-GLU MoE).

The MoE uses GPT-OSS's layout: `gate_up_proj` (E, H, 2I) with gate/up
INTERLEAVED along the last dim plus per-expert biases, activation
glu = gate*sigmoid(1.702*gate) with clamping, out = (up+1)*glu. TP shards
the pair dim (contiguous 2*I/tp column slices keep gate/up pairs together);
EP shards the expert dim; the down bias is pre-divided by moe_tp at
preshard time so the TP all-reduce restores it exactly once.

Real MXFP4 checkpoints are expected pre-dequantized (the HF loader's
default); the in-kernel MXFP4 path is future work.
"""

from __future__ import annotations

from typing import Dict, List

import torch
import torch.nn as nn
import torch.nn.functional as F

from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.models.llama.modeling_llama import LlamaForCausalLM
from neuronx_distributed_inference_amd.models.model_base import DecoderModel
from neuronx_distributed_inference_amd.modules.attention.attention_base import AttentionBase
from neuronx_distributed_inference_amd.modules.moe import MoERouter
from neuronx_distributed_inference_amd.modules.norm import RMSNorm
from neuronx_distributed_inference_amd.parallel import mappings
from neuronx_distributed_inference_amd.parallel.layers import divide, mark_parallel


class GptOssInferenceConfig(InferenceConfig):
    def get_required_attributes(self) -> List[str]:
        return ["hidden_size", "num_attention_heads", "num_hidden_layers",
                "num_key_value_heads", "vocab_size", "intermediate_size",
                "num_local_experts", "num_experts_per_tok"]

    def add_derived_config(self):
        if not hasattr(self, "rms_norm_eps"):
            self.rms_norm_eps = 1e-5
        if not hasattr(self, "head_dim"):
            self.head_dim = self.hidden_size // self.num_attention_heads
        if not hasattr(self, "attention_bias"):
            self.attention_bias = True
        self.attention_o_bias = True
        if not hasattr(self, "tie_word_embeddings"):
            self.tie_word_embeddings = False
        if not hasattr(self, "layer_types"):
            self.layer_types = [
                "sliding_attention" if i % 2 == 0 else "full_attention"
                for i in range(self.num_hidden_layers)
            ]
        if not hasattr(self, "swiglu_limit"):
            self.swiglu_limit = 7.0
        rp = getattr(self, "rope_parameters", None)
        if rp:
            self.rope_theta = rp.get("rope_theta", getattr(self, "rope_theta", 150000.0))
            self.rope_scaling = rp if rp.get("rope_type", "default") != "default" else None


class GptOssExperts(nn.Module):
    """Per-expert biased gate_up (interleaved) + clamped GLU + biased down.

    Weights are stored [N, K] per expert (gate_up (El, 2Ipp, H), down
    (El, H, Ipp)) so the grouped expert GEMM serves the GPU
    path directly. With ``neuron_config.moe_mxfp4`` the weights stay in
    the checkpoint's OCP MXFP4 packing (e2m1 nibbles + e8m0 per-32
    scales, 4.25 bits/weight — ref models/gpt_oss/mx_layout_transform.py)
    and are dequantized in-kernel during LDS staging: ~3.8x less weight
    traffic on the memory-bound decode path."""

    def __init__(self, config: InferenceConfig):
        super().__init__()
        ec = config.neuron_config
        E = config.num_local_experts
        H = config.hidden_size
        I = config.intermediate_size
        tp, ep = ec.tp_degree, ec.ep_degree
        moe_tp = divide(tp, ep)
        El = divide(E, ep)
        Ipp = divide(I, moe_tp)
        dtype = ec.torch_dtype
        self.num_experts = E
        self.num_local_experts = El
        self.ep_degree = ep
        self.alpha = 1.702
        self.limit = float(getattr(config, "swiglu_limit", 7.0))
        self.mxfp4 = bool(getattr(ec, "moe_mxfp4", False))
        self.hidden = H
        self.ipp = Ipp

        if self.mxfp4:
            self.gate_up_q = nn.Parameter(
                torch.zeros(El, 2 * Ipp, H // 2, dtype=torch.uint8),
                requires_grad=False)
            self.gate_up_s = nn.Parameter(
                torch.full((El, 2 * Ipp, H // 32), 127, dtype=torch.uint8),
                requires_grad=False)
            self.down_q = nn.Parameter(
                torch.zeros(El, H, Ipp // 2, dtype=torch.uint8),
                requires_grad=False)
            self.down_s = nn.Parameter(
                torch.full((El, H, Ipp // 32), 127, dtype=torch.uint8),
                requires_grad=False)
        else:
            self.gate_up_proj = nn.Parameter(torch.empty(El, 2 * Ipp, H, dtype=dtype))
            self.down_proj = nn.Parameter(torch.empty(El, H, Ipp, dtype=dtype))
        self.gate_up_proj_bias = nn.Parameter(torch.empty(El, 2 * Ipp, dtype=dtype))
        self.down_proj_bias = nn.Parameter(torch.empty(El, H, dtype=dtype))
        for p in self.parameters():
            mark_parallel(p, partition_dim=0)

        def _preshard(part_dim, divide_by_tp=False, pack=1):
            def fn(w, tp, El=El, moe_tp=moe_tp):
                blocks = []
                per = (w.shape[part_dim] // moe_tp) if part_dim is not None else None
                for r in range(tp):
                    ep_idx, tp_idx = r // moe_tp, r % moe_tp
                    blk = w[ep_idx * El:(ep_idx + 1) * El]
                    if part_dim is not None:
                        blk = blk.narrow(part_dim, tp_idx * per, per)
                    if divide_by_tp:
                        blk = blk / moe_tp
                    blocks.append(blk)
                return torch.cat(blocks, dim=0)
            return fn

        if self.mxfp4:
            self.gate_up_q.preshard_fn = _preshard(1)   # pairs stay adjacent
            self.gate_up_s.preshard_fn = _preshard(1)
            self.down_q.preshard_fn = _preshard(2)      # packed I dim
            self.down_s.preshard_fn = _preshard(2)
        else:
            self.gate_up_proj.preshard_fn = _preshard(1)
            self.down_proj.preshard_fn = _preshard(2)
        self.gate_up_proj_bias.preshard_fn = _preshard(1)
        self.down_proj_bias.preshard_fn = _preshard(None, divide_by_tp=True)

    def _local_offset(self) -> int:
        if self.ep_degree <= 1:
            return 0
        import torch.distributed as dist

        from neuronx_distributed_inference_amd.parallel import state as ps
        if not dist.is_initialized():
            return 0
        tp_rank = ps.get_tensor_model_parallel_rank()
        moe_tp = ps.get_tensor_model_parallel_size() // self.ep_degree
        return (tp_rank // max(moe_tp, 1)) * self.num_local_experts

    def _glu(self, gu: torch.Tensor) -> torch.Tensor:
        gate, up = gu[..., 0::2], gu[..., 1::2]
        gate = gate.clamp(max=self.limit)
        up = up.clamp(min=-self.limit, max=self.limit)
        glu = gate * torch.sigmoid(gate * self.alpha)
        return (up + 1) * glu

    def _gemm(self, x_in, which, expert_ids, x_gather=None, combine=None):
        from neuronx_distributed_inference_amd import ops as _ops

        kw = {} if combine is None else dict(
            combine_out=combine[0], combine_scale=combine[1],
            combine_gather=combine[2])
        if which == "gate_up":
            if self.mxfp4:
                return _ops.moe_grouped_gemm_mx4(
                    x_in, self.gate_up_q, self.gate_up_s, expert_ids,
                    bias=self.gate_up_proj_bias, x_gather=x_gather)
            return _ops.moe_grouped_gemm(
                x_in, self.gate_up_proj, expert_ids,
                bias=self.gate_up_proj_bias, x_gather=x_gather)
        if self.mxfp4:
            return _ops.moe_grouped_gemm_mx4(
                x_in, self.down_q, self.down_s, expert_ids,
                bias=self.down_proj_bias, **kw)
        return _ops.moe_grouped_gemm(
            x_in, self.down_proj, expert_ids, bias=self.down_proj_bias, **kw)

    def _forward_grouped(self, x, weights, idx):
        from neuronx_distributed_inference_amd import ops as _ops

        T, H = x.shape
        El = self.num_local_experts
        top_k = idx.shape[1]
        flat = (idx - self._local_offset()).view(-1)
        local = (flat >= 0) & (flat < El)
        flat_routed = torch.where(local, flat, torch.full_like(flat, El))
        order, dest, expert_ids, gtab, Mpad = _ops.moe_grouped_tables(
            flat_routed, El + 1, top_k)
        expert_ids = torch.where(expert_ids >= El,
                                 torch.full_like(expert_ids, -1), expert_ids)
        gatew = weights.view(-1)[order]
        lmask = local[order]
        gu = self._gemm(x, "gate_up", expert_ids, x_gather=gtab)
        h = self._glu(gu)
        gw_tab = torch.zeros(Mpad, dtype=torch.float32, device=x.device)
        gw_tab[dest] = torch.where(lmask, gatew.float(),
                                   torch.zeros_like(gatew, dtype=torch.float32))
        out32 = torch.zeros(T, H, dtype=torch.float32, device=x.device)
        self._gemm(h.contiguous(), "down", expert_ids,
                   combine=(out32, gw_tab, gtab))
        return out32.to(x.dtype)

    def _dense_w(self, e_local: int, which: str) -> torch.Tensor:
        """[N, K] weight of one local expert (dequantized if mxfp4)."""
        from neuronx_distributed_inference_amd import ops as _ops

        if which == "gate_up":
            if self.mxfp4:
                return _ops.mx4_dequant(self.gate_up_q[e_local],
                                        self.gate_up_s[e_local])
            return self.gate_up_proj[e_local]
        if self.mxfp4:
            return _ops.mx4_dequant(self.down_q[e_local], self.down_s[e_local])
        return self.down_proj[e_local]

    def forward(self, x: torch.Tensor, weights: torch.Tensor, idx: torch.Tensor):
        T, H = x.shape
        if x.is_cuda:
            from neuronx_distributed_inference_amd import ops as _ops

            # the grouped kernel is the GPU default for BOTH modes here:
            # GPT-OSS has no dense-bmm formulation (interleaved gate/up +
            # biases + clamped GLU), and the per-expert loop costs one
            # host sync per expert (measured 11.3 ms/layer at E=32 vs
            # 0.57 ms grouped). mxfp4 additionally keeps weights packed.
            if (_ops.extension_available() and H % 64 == 0
                    and self.ipp % 64 == 0):
                return self._forward_grouped(x, weights, idx)
        out = torch.zeros(T, H, dtype=x.dtype, device=x.device)
        offset = self._local_offset()
        for e_local in range(self.num_local_experts):
            e = offset + e_local
            sel = (idx == e)
            rows = sel.any(dim=-1)
            if not bool(rows.any()):
                continue
            xe = x[rows]
            wgu = self._dense_w(e_local, "gate_up").to(x.dtype)
            gu = xe @ wgu.t() + self.gate_up_proj_bias[e_local]
            he = self._glu(gu)
            wd = self._dense_w(e_local, "down").to(x.dtype)
            he = he @ wd.t() + self.down_proj_bias[e_local]
            w = (weights * sel).sum(dim=-1)[rows].to(x.dtype).unsqueeze(-1)
            out[rows] += he * w
        return out


class GptOssMoE(nn.Module):
    def __init__(self, config: InferenceConfig):
        super().__init__()
        E = config.num_local_experts
        self.router = MoERouter(
            config.hidden_size, E, config.num_experts_per_tok,
            normalize=True, pre_softmax=False,
        )
        self.router_bias = nn.Parameter(torch.zeros(E))
        self.experts = GptOssExperts(config)

    def forward(self, h: torch.Tensor) -> torch.Tensor:
        B, S, H = h.shape
        flat = h.view(-1, H)
        logits = F.linear(flat.float(), self.router.weight.float(), self.router_bias.float())
        vals, idx = logits.topk(self.router.top_k, dim=-1)
        weights = torch.softmax(vals, dim=-1)
        out = self.experts(flat, weights, idx)
        out = mappings.reduce_from_tensor_parallel_region(out)
        return out.view(B, S, H)


class GptOssDecoderLayer(nn.Module):
    def __init__(self, config: InferenceConfig, layer_idx: int):
        super().__init__()
        ec = config.neuron_config
        is_sliding = config.layer_types[layer_idx] == "sliding_attention"
        self.self_attn = AttentionBase(config, layer_idx)
        self.self_attn.sliding_window = (config.sliding_window or 0) if is_sliding else 0
        # learned attention sinks, one per q head, sharded with the heads
        # (ref modules/attention/sink.py)
        hpr = self.self_attn.num_heads_per_rank
        self.self_attn.sinks = nn.Parameter(
            torch.zeros(hpr, dtype=ec.torch_dtype)
        )
        mark_parallel(self.self_attn.sinks, partition_dim=0)
        self.mlp = GptOssMoE(config)
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


class GptOssModel(DecoderModel):
    def build_layers(self, config):
        return nn.ModuleList(
            GptOssDecoderLayer(config, i) for i in range(config.num_hidden_layers)
        )


class GptOssForCausalLM(LlamaForCausalLM):
    _model_cls = GptOssModel
    _config_cls = GptOssInferenceConfig

    @staticmethod
    def convert_hf_to_native_state_dict(
        sd: Dict[str, torch.Tensor], config: InferenceConfig
    ) -> Dict[str, torch.Tensor]:
        mxfp4 = bool(getattr(config.neuron_config, "moe_mxfp4", False))
        from neuronx_distributed_inference_amd import ops as _ops

        out: Dict[str, torch.Tensor] = {}
        pending_scales: Dict[str, torch.Tensor] = {}
        for k, v in sd.items():
            nk = k[len("model."):] if k.startswith("model.") else k
            nk = nk.replace("self_attn.q_proj", "self_attn.qkv_proj.q_proj")
            nk = nk.replace("self_attn.k_proj", "self_attn.qkv_proj.k_proj")
            nk = nk.replace("self_attn.v_proj", "self_attn.qkv_proj.v_proj")
            nk = nk.replace("self_attn.o_proj", "self_attn.o_proj.o_proj")
            nk = nk.replace("self_attn.sinks", "self_attn.sinks")
            nk = nk.replace(".mlp.router.bias", ".mlp.router_bias")
            out[nk] = v
        # expert weights: native checkpoints are MXFP4 (`*_blocks` e2m1
        # nibbles + `*_scales` e8m0; ref mx_layout_transform.py). With
        # moe_mxfp4 keep the packing (the grouped kernel dequantizes
        # in-LDS); otherwise dequantize to the [N, K] dense layout.
        # Pre-dequantized HF dumps carry dense (E, H, 2I)/(E, I, H) —
        # transpose those into [N, K].
        conv: Dict[str, torch.Tensor] = {}
        for k, v in list(out.items()):
            if k.endswith("experts.gate_up_proj_blocks") or k.endswith(
                    "experts.down_proj_blocks"):
                which = "gate_up" if "gate_up" in k else "down"
                base = k[: -len("_blocks")]
                sk = base + "_scales"
                blocks = v.reshape(v.shape[0], v.shape[1], -1)  # (E, N, K/2)
                scales = out[sk]
                if mxfp4:
                    conv[base.replace(f"{which}_proj", f"{which}_q")
                         if False else base.replace("gate_up_proj", "gate_up_q")
                         .replace("down_proj", "down_q")] = blocks.contiguous()
                    conv[base.replace("gate_up_proj", "gate_up_s")
                         .replace("down_proj", "down_s")] = scales.contiguous()
                else:
                    conv[base] = _ops.mx4_dequant(blocks, scales)
            elif k.endswith("_scales") and ("gate_up_proj" in k or "down_proj" in k):
                continue  # consumed with its blocks
            elif k.endswith("experts.gate_up_proj") or k.endswith("experts.down_proj"):
                conv[k] = v.transpose(1, 2).contiguous()  # dense HF -> [N, K]
            else:
                conv[k] = v
        out = conv
        if "lm_head.weight" not in out and getattr(config, "tie_word_embeddings", False):
            out["lm_head.weight"] = out["embed_tokens.weight"]
        dtype = config.neuron_config.torch_dtype
        converted = {}
        for k, v in out.items():
            if k == "mlp.router_bias" or k.endswith(".router_bias"):
                converted[k] = v.float()
            elif v.dtype == torch.uint8:
                converted[k] = v
            else:
                converted[k] = v.to(dtype)
        return converted
