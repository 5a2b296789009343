"""Gated MLP (SwiGLU / GeGLU) with tensor-parallel projections.

Analog of the reference's kernel-fused MLP path (K5,
`models/llama/modeling_llama.py:454-737`): gate/up are column-parallel,
down is row-parallel (all-reduce C1); the activation x multiply is the
fused HIP `silu_mul` kernel so the intermediate never round-trips HBM twice.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.parallel.layers import (
    ColumnParallelLinear,
    RowParallelLinear,
)


class GatedMLP(nn.Module):
    def __init__(
        self,
        hidden_size: int,
        intermediate_size: int,
        tp_degree: int,
        dtype: torch.dtype,
        hidden_act: str = "silu",
        bias: bool = False,
        group=None,
        fused_gate_up: bool = False,
    ):
        super().__init__()
        self.fused = fused_gate_up
        if fused_gate_up:
            # one GEMM for gate|up: checkpoint carries cat([gate, up]); the
            # preshard hook reorders rows rank-major so a dim-0 slice yields
            # [gate_r | up_r]
            self.gate_up_proj = ColumnParallelLinear(
                hidden_size, 2 * intermediate_size, bias=bias, dtype=dtype,
                tp_degree=tp_degree, group=group,
            )

            def _fuse_preshard(w, tp, I=intermediate_size):
                g, u = torch.split(w, [I, I], dim=0)
                per = I // tp
                parts = []
                for r in range(tp):
                    parts.append(g[r * per:(r + 1) * per])
                    parts.append(u[r * per:(r + 1) * per])
                return torch.cat(parts, dim=0)

            self.gate_up_proj.weight.preshard_fn = _fuse_preshard
            if bias:
                self.gate_up_proj.bias.preshard_fn = _fuse_preshard
        else:
            self.gate_proj = ColumnParallelLinear(
                hidden_size, intermediate_size, bias=bias, dtype=dtype,
                tp_degree=tp_degree, group=group,
            )
            self.up_proj = ColumnParallelLinear(
                hidden_size, intermediate_size, bias=bias, dtype=dtype,
                tp_degree=tp_degree, group=group,
            )
        self.down_proj = RowParallelLinear(
            intermediate_size, hidden_size, bias=bias, dtype=dtype,
            tp_degree=tp_degree, group=group,
        )
        if hidden_act in ("silu", "swish"):
            self.act_mul = ops.silu_mul
            self.act_mul_fused = ops.silu_mul_fused
        elif hidden_act in ("gelu", "gelu_pytorch_tanh", "gelu_new"):
            self.act_mul = ops.gelu_tanh_mul
            self.act_mul_fused = ops.gelu_tanh_mul_fused
        else:
            raise ValueError(f"unsupported hidden_act {hidden_act}")

    def forward(self, x: torch.Tensor, sequence_parallel: bool = False) -> torch.Tensor:
        sp = sequence_parallel
        if self.fused:
            y = self.gate_up_proj(x, sequence_parallel=sp)
            return self.down_proj(self.act_mul_fused(y), sequence_parallel=sp)
        g = self.gate_proj(x, sequence_parallel=sp)
        u = self.up_proj(x, sequence_parallel=sp)
        return self.down_proj(self.act_mul(g, u), sequence_parallel=sp)
