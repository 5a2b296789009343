"""FP8 weight quantization: CPU dequant path close to full precision;
quantize_model_fp8 converts the right modules."""

import pytest
import torch

from neuronx_distributed_inference_amd.models.llama.modeling_llama import LlamaForCausalLM
from neuronx_distributed_inference_amd.parallel.layers import (
    ColumnParallelLinear,
    quantize_model_fp8,
)


@torch.inference_mode()
def test_fp8_linear_close():
    torch.manual_seed(0)
    lin = ColumnParallelLinear(64, 32, dtype=torch.float32, tp_degree=1)
    with torch.no_grad():
        lin.weight.normal_(0, 0.3)
    x = torch.randn(4, 64)
    want = torch.nn.functional.linear(x, lin.weight)
    lin.quantize_fp8()
    assert lin.weight.dtype == torch.float8_e4m3fn
    got = lin(x)
    rel = (got - want).abs().max() / want.abs().max()
    assert rel < 0.05, rel


@torch.inference_mode()
def test_quantize_model_keeps_generation_reasonable(tiny_llama_config):
    torch.manual_seed(0)
    m = LlamaForCausalLM(tiny_llama_config).eval()
    with torch.no_grad():
        for p in m.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    ids = torch.randint(0, 128, (2, 8))
    m.ec.output_logits = True
    m.model.sampler = None
    base = m(input_ids=ids).logits.clone()
    m.reset()
    n = quantize_model_fp8(m.model)
    assert n > 0
    assert m.model.lm_head.weight.dtype != torch.float8_e4m3fn  # skipped
    out = m(input_ids=ids).logits
    rel = (out - base).abs().max() / base.abs().std()
    assert rel < 2.0  # fp8 noise, not garbage
