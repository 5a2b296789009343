"""Normalization modules backed by the HIP RMSNorm kernel (K7-analog,
ref `modules/custom_calls.py:30-34`)."""

from __future__ import annotations

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd import ops


class RMSNorm(nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-6, dtype: torch.dtype = torch.float32):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size, dtype=dtype))
        self.variance_epsilon = eps

    def forward(self, x: torch.Tensor, residual: torch.Tensor = None):
        out, res = ops.rms_norm(x, self.weight, self.variance_epsilon, residual)
        if residual is None:
            return out
        return out, res


class GemmaRMSNorm(nn.Module):
    """Gemma convention: scale by (1 + weight)."""

    def __init__(self, hidden_size: int, eps: float = 1e-6, dtype: torch.dtype = torch.float32):
        super().__init__()
        self.weight = nn.Parameter(torch.zeros(hidden_size, dtype=dtype))
        self.variance_epsilon = eps

    def forward(self, x: torch.Tensor):
        out, _ = ops.rms_norm(x, 1.0 + self.weight.float(), self.variance_epsilon)
        return out


class LayerNorm(nn.Module):
    """LayerNorm (optionally bias-free, DBRX-style) with the same
    (x, residual) -> (out, new_residual) contract as RMSNorm."""

    def __init__(self, hidden_size: int, eps: float = 1e-5, bias: bool = False,
                 dtype: torch.dtype = torch.float32):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size, dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(hidden_size, dtype=dtype)) if bias else None
        self.eps = eps
        self.hidden_size = hidden_size

    def forward(self, x: torch.Tensor, residual: torch.Tensor = None):
        orig = x.dtype
        if residual is not None:
            x = x + residual
            new_res = x
        xf = torch.nn.functional.layer_norm(
            x.float(), (self.hidden_size,), self.weight.float(),
            self.bias.float() if self.bias is not None else None, self.eps,
        ).to(orig)
        if residual is None:
            return xf
        return xf, new_res
