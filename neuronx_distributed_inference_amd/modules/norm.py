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
