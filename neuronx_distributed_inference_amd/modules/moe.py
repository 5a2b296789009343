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
                 act: str = "softmax",
                 dtype: torch.dtype = torch.float32):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(num_experts, hidden_size, dtype=dtype))
        self.top_k = top_k
        self.num_experts = num_experts
        self.normalize = normalize
        self.pre_softmax = pre_softmax
        self.act = act  # "softmax" | "sigmoid" (Llama4: sigmoid of top-k logits)

    def forward(self, x: torch.Tensor):
        logits = F.linear(x.float(), self.weight.float())  # (T, E)
        if (logits.is_cuda and ops.extension_available()
                and self.top_k <= 8 and self.num_experts >= self.top_k):
            # fused top-k + activation + normalize: one launch vs the
            # ~6-kernel torch chain (launch-bound at decode sizes)
            sig = self.act == "sigmoid"
            return ops.router_topk(
                logits, self.top_k, sig,
                (not sig) and self.pre_softmax,
                (not sig) and self.normalize)
        if self.act == "sigmoid":
            vals, idx = logits.topk(self.top_k, dim=-1)
            return torch.sigmoid(vals), idx
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
    """Stacked gated expert MLPs.

    TP x EP sharding (ref hybrid moe_tp/moe_ep,
    `modules/moe_v2.py:135-161`): the intermediate dim is sharded over
    moe_tp = tp/ep ranks and the EXPERT dim over ep ranks, so rank r of the
    TP group holds experts [ (r//moe_tp)*E/ep , ...+E/ep ) with I/moe_tp
    columns each. Because TP replicates activations, a single all-reduce
    over the full TP group (done by MoELayer) sums both the I-shards and
    the expert subsets — the AR dispatch option of the reference's
    EPDispatchOption enum (attention_base.py:103-113).

    Local shapes: w_gate/w_up (E/ep, I/moe_tp, H), w_down (E/ep, H, I/moe_tp).
    The preshard hook stacks each global rank's 3D block on dim 0 so the
    standard dim-0 slicing yields it.
    """

    def __init__(self, num_experts: int, hidden_size: int, intermediate_size: int,
                 tp_degree: int, dtype: torch.dtype, hidden_act: str = "silu",
                 group=None, ep_degree: int = 1, scale_input: bool = False):
        super().__init__()
        self.num_experts = num_experts
        self.ep_degree = ep_degree
        # Llama4 quirk: the router gate scales the expert INPUT (y = E(g*x)),
        # not the output (HF Llama4TextMoe routed_in scaling)
        self.scale_input = scale_input
        moe_tp = divide(tp_degree, ep_degree)
        self.num_local_experts = divide(num_experts, ep_degree)
        ipp = divide(intermediate_size, moe_tp)
        El = self.num_local_experts
        self.w_gate = nn.Parameter(torch.empty(El, ipp, hidden_size, dtype=dtype))
        self.w_up = nn.Parameter(torch.empty(El, ipp, hidden_size, dtype=dtype))
        self.w_down = nn.Parameter(torch.empty(El, hidden_size, ipp, dtype=dtype))

        def _preshard(part_dim):
            def fn(w, tp, El=El, moe_tp=moe_tp, ipp=ipp):
                # full w: (E, I, H) or (E, H, I); build per-global-rank 3D
                # blocks stacked on dim 0 (rank r: ep_idx = r // moe_tp,
                # tp_idx = r % moe_tp)
                blocks = []
                for r in range(tp):
                    ep_idx, tp_idx = r // moe_tp, r % moe_tp
                    blk = w[ep_idx * El:(ep_idx + 1) * El]
                    blk = blk.narrow(part_dim, tp_idx * ipp, ipp)
                    blocks.append(blk)
                return torch.cat(blocks, dim=0)
            return fn

        mark_parallel(self.w_gate, partition_dim=0)
        mark_parallel(self.w_up, partition_dim=0)
        mark_parallel(self.w_down, partition_dim=0)
        self.w_gate.preshard_fn = _preshard(1)
        self.w_up.preshard_fn = _preshard(1)
        self.w_down.preshard_fn = _preshard(2)
        self.act_mul = ops.silu_mul if hidden_act in ("silu", "swish") else ops.gelu_tanh_mul
        self._group = group
        self.expert_offset = 0  # resolved from the live EP rank at forward

    def _local_offset(self) -> int:
        if self.ep_degree <= 1:
            return 0
        from neuronx_distributed_inference_amd.parallel import state as ps

        import torch.distributed as dist
        if not dist.is_initialized():
            return 0
        tp_rank = ps.get_tensor_model_parallel_rank()
        moe_tp = ps.get_tensor_model_parallel_size() // self.ep_degree
        return (tp_rank // max(moe_tp, 1)) * self.num_local_experts

    def quantize_fp8(self) -> bool:
        """Convert the stacked expert weights to fp8-e4m3 (1 byte/elem,
        per-output-channel fp32 scale): HALVES the decode weight stream.
        Returns False (and leaves bf16) when the shard shapes don't meet
        the fp8 kernel's column-tile constraint (N % 128 == 0)."""
        I, H = self.w_gate.shape[1], self.w_down.shape[1]
        if I % 128 or H % 128:
            return False
        for name in ("w_gate", "w_up", "w_down"):
            w = getattr(self, name)
            q, sc = ops.fp8_moe_quantize(w.data)
            del self._parameters[name]
            self.register_buffer(name + "_q", q)
            self.register_buffer(name + "_s", sc)
        self.fp8 = True
        return True

    def quantize_mx4(self) -> bool:
        """Convert the stacked expert weights to OCP MXFP4 (e2m1 nibbles
        + e8m0 per-32 block scales, 4.25 bits/weight — a 3.8x memory cut
        AND faster than bf16 on K%256 shapes via the all-glds v3 kernel:
        160 vs 217 us at the Mixtral gate GEMM)."""
        I, H = self.w_gate.shape[1], self.w_down.shape[1]
        K1, K2 = self.w_gate.shape[2], self.w_down.shape[2]
        if I % 64 or H % 64 or K1 % 256 or K2 % 256:
            return False
        for name in ("w_gate", "w_up", "w_down"):
            w = getattr(self, name)
            q, sc = ops.mx4_quantize(w.data.float())
            del self._parameters[name]
            self.register_buffer(name + "_q", q.to(w.device))
            self.register_buffer(name + "_s", sc.to(w.device))
        self.mx4 = True
        return True

    def _forward_dense(self, x: torch.Tensor, weights: torch.Tensor,
                       idx: torch.Tensor) -> torch.Tensor:
        T, H = x.shape
        El = self.num_local_experts
        gates = torch.zeros(T, El, dtype=x.dtype, device=x.device)
        gates = gates.scatter(1, idx, weights.to(x.dtype))
        gte = gates.t().unsqueeze(-1)                      # (E, T, 1)
        if self.scale_input:
            xe = x.unsqueeze(0) * gte                      # Llama4: gate input
        else:
            xe = x.unsqueeze(0).expand(El, T, H)
        g = torch.bmm(xe, self.w_gate.transpose(1, 2))
        u = torch.bmm(xe, self.w_up.transpose(1, 2))
        y = torch.bmm(self.act_mul(g, u), self.w_down.transpose(1, 2))
        if not self.scale_input:
            y = y * gte
        return y.sum(dim=0)

    def _forward_grouped(self, x: torch.Tensor, weights: torch.Tensor,
                         idx: torch.Tensor) -> torch.Tensor:
        """Grouped expert GEMM: sort assignments by expert,
        pad each segment to 64-row tiles, one kernel launch per
        projection for ALL experts. Static shapes + tensor-only table
        construction -> hipGraph-capturable; no top_k/E redundant FLOPs
        (the dense-bmm path is compute-bound at decode batch >=128).
        Non-local assignments (EP) route to a sentinel segment whose
        tiles the kernel skips; their contribution is masked out."""
        T, H = x.shape
        El = self.num_local_experts
        offset = self._local_offset()
        top_k = idx.shape[1]
        flat = (idx - offset).view(-1)
        local = (flat >= 0) & (flat < El)
        flat_routed = torch.where(local, flat, torch.full_like(flat, El))
        fp8 = getattr(self, "fp8", False)
        mx4 = getattr(self, "mx4", False)
        # fine-grained MoE (few tokens per expert): a 32-row tile halves
        # the padded-row waste. bf16 kernel only (quantized kernels are
        # 64-row); opt-in via NXDI_MOE_TILE32=1 while it soaks.
        import os as _os

        if fp8 or mx4:
            I0 = self.w_gate_q.shape[1]
        else:
            I0 = self.w_gate.shape[1]
        tile = 32 if (
            not fp8 and not mx4
            and _os.environ.get("NXDI_MOE_TILE32") != "0"
            and T * top_k < 48 * El
            and I0 % 128 == 0 and H % 128 == 0
        ) else 64
        order, dest, expert_ids, gtab, Mpad = ops.moe_grouped_tables(
            flat_routed, El + 1, top_k, tile=tile)
        # sentinel segment (id El) never computes
        expert_ids = torch.where(expert_ids >= El,
                                 torch.full_like(expert_ids, -1), expert_ids)
        gatew = weights.view(-1)[order]
        lmask = local[order]

        # widest column tile that still yields >= ~2 blocks/CU — wide
        # tiles cut x re-reads 4x but run at occupancy 1, so a small
        # grid (TP-sharded N or small batch) must fall back to narrower
        # tiles to fill the chip
        def _bn(N, ntiles=Mpad // tile):
            if tile == 32:
                return 256 if (N % 256 == 0
                               and (N // 256) * ntiles >= 256) else 128
            # occupancy is 1 block/CU at bn>=192 (120 KB LDS), 2 below
            for cand in (256, 192, 128):
                occ = 1 if cand >= 192 else 2
                if N % cand == 0 and (N // cand) * ntiles >= 256 * occ:
                    return cand
            return 64

        if fp8 or mx4:
            I, Hn = self.w_gate_q.shape[1], self.w_down_q.shape[1]
        else:
            I, Hn = self.w_gate.shape[1], self.w_down.shape[1]
        if self.scale_input:
            # input scaling is nonlinear through the GLU: keep the
            # materialized gather for this (rare) variant
            tok = order // top_k
            x_rows = x.index_select(0, tok) * gatew.to(x.dtype).unsqueeze(-1)
            x_in = torch.zeros(Mpad, H, dtype=x.dtype, device=x.device)
            x_in.index_copy_(0, dest, x_rows)
            xg = None
        else:
            x_in = x                      # token gather fuses into staging
            xg = gtab
        if mx4:
            g = ops.moe_grouped_gemm_mx4(x_in, self.w_gate_q, self.w_gate_s,
                                         expert_ids, x_gather=xg)
            u = ops.moe_grouped_gemm_mx4(x_in, self.w_up_q, self.w_up_s,
                                         expert_ids, x_gather=xg)
        elif fp8:
            # fp8 column tiles are 128/256 only
            def _bn8(N, ntiles=Mpad // 64):
                if N % 256 == 0 and (N // 256) * ntiles >= 256:
                    return 256
                return 128

            g = ops.moe_grouped_gemm_fp8(x_in, self.w_gate_q, self.w_gate_s,
                                         expert_ids, bn=_bn8(I), x_gather=xg)
            u = ops.moe_grouped_gemm_fp8(x_in, self.w_up_q, self.w_up_s,
                                         expert_ids, bn=_bn8(I), x_gather=xg)
        else:
            g = ops.moe_grouped_gemm(x_in, self.w_gate, expert_ids, bn=_bn(I),
                                     x_gather=xg, mt=tile)
            u = ops.moe_grouped_gemm(x_in, self.w_up, expert_ids, bn=_bn(I),
                                     x_gather=xg, mt=tile)
        h = self.act_mul(g, u)
        # routing-weighted combine fuses into the down epilogue (fp32
        # atomics, one rmw per element); non-local assignments are
        # zero-scaled
        gw_tab = torch.zeros(Mpad, dtype=torch.float32, device=x.device)
        scale = gatew.float() if not self.scale_input \
            else torch.ones_like(gatew, dtype=torch.float32)
        gw_tab[dest] = torch.where(lmask, scale, torch.zeros_like(scale))
        out32 = torch.zeros(T, H, dtype=torch.float32, device=x.device)
        if mx4:
            ops.moe_grouped_gemm_mx4(h, self.w_down_q, self.w_down_s,
                                     expert_ids, combine_out=out32,
                                     combine_scale=gw_tab,
                                     combine_gather=gtab)
        elif fp8:
            ops.moe_grouped_gemm_fp8(h, self.w_down_q, self.w_down_s,
                                     expert_ids, bn=_bn8(Hn),
                                     combine_out=out32, combine_scale=gw_tab,
                                     combine_gather=gtab)
        else:
            ops.moe_grouped_gemm(h, self.w_down, expert_ids, bn=_bn(Hn),
                                 combine_out=out32, combine_scale=gw_tab,
                                 combine_gather=gtab, mt=tile)
        return out32.to(x.dtype)

    def forward(self, x: torch.Tensor, weights: torch.Tensor, idx: torch.Tensor):
        """x: (T, H); weights/idx: (T, top_k) over GLOBAL expert ids.
        Returns (T, H) partial (summed by the caller's TP all-reduce)."""
        T, H = x.shape
        # Grouped expert GEMM is the GPU DEFAULT (opt out with
        # NXDI_MOE_GROUPED=0). With wide column tiles (bn=256: 4x fewer
        # x re-reads) and the fused one-launch segment-table kernel
        # (142 -> 13 us of launch latency), grouped beats the dense bmm
        # path on Mixtral decode T=256 top-2 (826 vs 877 us/layer;
        # profiles/r02_bench_summary.md) on top of removing the top_k/E
        # redundant FLOPs.
        import os as _os

        if getattr(self, "fp8", False) or getattr(self, "mx4", False):
            return self._forward_grouped(x, weights, idx)
        grouped_env = _os.environ.get("NXDI_MOE_GROUPED")
        # small batches leave too few 64-row tiles to fill 256 CUs at the
        # wide-tile occupancy (measured: Mixtral T=64 grouped 27.7 vs
        # dense 23.7 ms/step; T=256 grouped 31.2 vs dense 32.0) — dense
        # bmm keeps those
        big_enough = x.shape[0] * idx.shape[1] >= 384
        use_grouped = (
            x.is_cuda and ops.extension_available()
            and x.dtype in (torch.bfloat16, torch.float16)
            and self.w_gate.shape[2] % 64 == 0 and self.w_gate.shape[1] % 64 == 0
            and H % 64 == 0
            and (grouped_env == "1" or (grouped_env != "0" and big_enough))
        )
        if use_grouped:
            return self._forward_grouped(x, weights, idx)
        # decode-sized batches: DENSE batched-GEMM path. Every expert's
        # weights must stream from HBM for any realistic decode batch
        # anyway, so the dense bmm has the SAME memory traffic as the
        # gather loop while being free of host syncs and dynamic shapes —
        # i.e. hipGraph-capturable (the reference gets this from the
        # traced NEFF; SURVEY K12 grouped-GEMM analog).
        if self.ep_degree == 1 and T <= 512:
            return self._forward_dense(x, weights, idx)
        out = torch.zeros(T, H, dtype=x.dtype, device=x.device)
        offset = self._local_offset()
        for e_local in range(self.num_local_experts):
            e = offset + e_local
            sel = (idx == e)            # (T, top_k)
            rows = sel.any(dim=-1)
            if not bool(rows.any()):
                continue
            xe = x[rows]
            w = (weights * sel).sum(dim=-1)[rows].to(x.dtype).unsqueeze(-1)
            if self.scale_input:
                xe = xe * w
            ge = F.linear(xe, self.w_gate[e_local])
            ue = F.linear(xe, self.w_up[e_local])
            he = F.linear(self.act_mul(ge, ue), self.w_down[e_local])
            out[rows] += he if self.scale_input else he * w
        return out


class MoELayer(nn.Module):
    """Router + experts (+ optional shared expert), TP all-reduce at the end."""

    def __init__(self, config, hidden_act: str = "silu",
                 normalize_top_k: bool = True, pre_softmax: bool = False,
                 router_act: str = "softmax", scale_input: bool = False):
        super().__init__()
        ec = config.neuron_config
        E = getattr(config, "num_local_experts", None) or getattr(config, "num_experts")
        top_k = getattr(config, "num_experts_per_tok", None) or getattr(config, "moe_top_k", 2)
        inter = getattr(config, "moe_intermediate_size", None) or config.intermediate_size
        self.router = MoERouter(
            config.hidden_size, E, top_k, normalize=normalize_top_k,
            pre_softmax=pre_softmax, act=router_act,
        )
        self.experts = ExpertMLPs(
            E, config.hidden_size, inter, ec.tp_degree, ec.torch_dtype,
            hidden_act=hidden_act, ep_degree=ec.ep_degree, scale_input=scale_input,
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
