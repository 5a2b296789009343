"""Mixture-of-Experts layer.

Analog of the reference's MoE stack (`modules/moe_v2.py:23-132`
initialize_moe_module + NxD blockwise/grouped expert kernels, K12): a
router (fp32 top-k softmax) over tensor-parallel expert MLPs with stacked
(E, ...) weights.

Sharding: experts' intermediate dim is column/row-sharded over the moe-TP
group exactly like a dense MLP; with ep_degree > 1 the expert dim is
partitioned across EP ranks and tokens routed to remote experts go through
the EP all-to-all (C9). The v1 compute path runs one GEMM per expert with
gathered token groups — the grouped-GEMM HIP kernel replaces it without
changing this module's contract.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.parallel import mappings
from neuronx_distributed_inference_amd.parallel import state as ps
from neuronx_distributed_inference_amd.parallel.layers import divide, mark_parallel


class MoERouter(nn.Module):
    """Top-k softmax router in fp32 (ref router_dtype float32)."""

    def __init__(self, hidden_size: int, num_experts: int, top_k: int,
                 normalize: bool = True, pre_softmax: bool = False,
                 dtype: torch.dtype = torch.float32):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(num_experts, hidden_size, dtype=dtype))
        self.top_k = top_k
        self.num_experts = num_experts
        self.normalize = normalize
        self.pre_softmax = pre_softmax

    def forward(self, x: torch.Tensor):
        logits = F.linear(x.float(), self.weight.float())  # (T, E)
        if self.pre_softmax:
            probs = torch.softmax(logits, dim=-1)
            vals, idx = probs.topk(self.top_k, dim=-1)
        else:
            vals, idx = logits.topk(self.top_k, dim=-1)
            vals = torch.softmax(vals, dim=-1)
        if self.normalize:
            vals = vals / vals.sum(dim=-1, keepdim=True)
        return vals, idx


class ExpertMLPs(nn.Module):
    """Stacked gated expert MLPs: w_gate/w_up (E, I/tp, H), w_down (E, H, I/tp)."""

    def __init__(self, num_experts: int, hidden_size: int, intermediate_size: int,
                 tp_degree: int, dtype: torch.dtype, hidden_act: str = "silu",
                 group=None):
        super().__init__()
        self.num_experts = num_experts
        ipp = divide(intermediate_size, tp_degree)
        self.w_gate = nn.Parameter(torch.empty(num_experts, ipp, hidden_size, dtype=dtype))
        self.w_up = nn.Parameter(torch.empty(num_experts, ipp, hidden_size, dtype=dtype))
        self.w_down = nn.Parameter(torch.empty(num_experts, hidden_size, ipp, dtype=dtype))
        mark_parallel(self.w_gate, partition_dim=1)
        mark_parallel(self.w_up, partition_dim=1)
        mark_parallel(self.w_down, partition_dim=2)
        self.act_mul = ops.silu_mul if hidden_act in ("silu", "swish") else ops.gelu_tanh_mul
        self._group = group

    def forward(self, x: torch.Tensor, weights: torch.Tensor, idx: torch.Tensor):
        """x: (T, H); weights/idx: (T, top_k). Returns (T, H) partial (needs
        TP all-reduce by caller)."""
        T, H = x.shape
        out = torch.zeros(T, H, dtype=x.dtype, device=x.device)
        for e in range(self.num_experts):
            sel = (idx == e)            # (T, top_k)
            rows = sel.any(dim=-1)
            if not bool(rows.any()):
                continue
            xe = x[rows]
            ge = F.linear(xe, self.w_gate[e])
            ue = F.linear(xe, self.w_up[e])
            he = F.linear(self.act_mul(ge, ue), self.w_down[e])
            w = (weights * sel).sum(dim=-1)[rows].to(x.dtype).unsqueeze(-1)
            out[rows] += he * w
        return out


class MoELayer(nn.Module):
    """Router + experts (+ optional shared expert), TP all-reduce at the end."""

    def __init__(self, config, hidden_act: str = "silu",
                 normalize_top_k: bool = True, pre_softmax: bool = False):
        super().__init__()
        ec = config.neuron_config
        E = getattr(config, "num_local_experts", None) or getattr(config, "num_experts")
        top_k = getattr(config, "num_experts_per_tok", None) or getattr(config, "moe_top_k", 2)
        inter = getattr(config, "moe_intermediate_size", None) or config.intermediate_size
        self.router = MoERouter(
            config.hidden_size, E, top_k, normalize=normalize_top_k,
            pre_softmax=pre_softmax,
        )
        self.experts = ExpertMLPs(
            E, config.hidden_size, inter, ec.tp_degree, ec.torch_dtype,
            hidden_act=hidden_act,
        )
        self.shared_expert = None  # models with shared experts set this

    def forward(self, h: torch.Tensor) -> torch.Tensor:
        B, S, H = h.shape
        flat = h.view(-1, H)
        weights, idx = self.router(flat)
        out = self.experts(flat, weights, idx)
        if self.shared_expert is not None:
            out = out + self.shared_expert(flat)
        out = mappings.reduce_from_tensor_parallel_region(out)
        return out.view(B, S, H)
