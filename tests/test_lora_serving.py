"""Multi-LoRA serving: wrapped model output == base + manual low-rank delta;
per-row adapter selection; slot 0 (zero adapter) is a no-op."""

import pytest
import torch

from neuronx_distributed_inference_amd.models.llama.modeling_llama import LlamaForCausalLM
from neuronx_distributed_inference_amd.modules.lora_serving import (
    LoraServingConfig,
    lora_layers,
    wrap_model_with_lora,
)


@pytest.fixture
def model(tiny_llama_config):
    torch.manual_seed(0)
    m = LlamaForCausalLM(tiny_llama_config).eval()
    with torch.no_grad():
        for p in m.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    return m


@torch.inference_mode()
def test_zero_adapter_is_identity(model):
    ids = torch.randint(0, 128, (2, 6))
    base_out = model(input_ids=ids).logits.clone()
    model.reset()
    cfg = LoraServingConfig(max_loras=2, max_lora_rank=4)
    state = wrap_model_with_lora(model.model, cfg, max_batch=2)
    assert len(lora_layers(model.model)) > 0
    state.set(torch.zeros(2, dtype=torch.long))
    out = model(input_ids=ids).logits
    torch.testing.assert_close(out, base_out, atol=1e-5, rtol=1e-5)


@torch.inference_mode()
def test_adapter_changes_output_per_row(model):
    torch.manual_seed(3)
    cfg = LoraServingConfig(max_loras=3, max_lora_rank=4, lora_alpha=8.0)
    state = wrap_model_with_lora(model.model, cfg, max_batch=2)
    # install a random adapter into slot 1 of every wrapped layer
    for name, layer in lora_layers(model.model).items():
        r = 4
        a = torch.randn(r, layer.lora_A.shape[2]) * 0.1
        b = torch.randn(layer.lora_B.shape[1], r) * 0.1
        layer.load_adapter(1, a, b)

    ids = torch.randint(0, 128, (2, 6))
    state.set(torch.tensor([0, 0]))
    model.reset()
    out_base = model(input_ids=ids).logits.clone()
    state.set(torch.tensor([0, 1]))
    model.reset()
    out_mix = model(input_ids=ids).logits
    # row 0 unchanged (slot 0 = zero adapter), row 1 changed
    torch.testing.assert_close(out_mix[0], out_base[0], atol=1e-5, rtol=1e-5)
    assert (out_mix[1] - out_base[1]).abs().max() > 1e-3


@torch.inference_mode()
def test_single_layer_delta_math():
    """Wrapped column-parallel linear equals base + scaling * x A^T B^T."""
    from neuronx_distributed_inference_amd.modules.lora_serving import (
        LoraState,
        MultiLoraLinear,
    )
    from neuronx_distributed_inference_amd.parallel.layers import ColumnParallelLinear

    torch.manual_seed(0)
    base = ColumnParallelLinear(16, 32, dtype=torch.float32, tp_degree=1)
    with torch.no_grad():
        base.weight.normal_(0, 0.1)
    cfg = LoraServingConfig(max_loras=2, max_lora_rank=4, lora_alpha=4.0)
    state = LoraState(2)
    wrapped = MultiLoraLinear(base, cfg, state, "column")
    a = torch.randn(4, 16) * 0.2
    b = torch.randn(32, 4) * 0.2
    wrapped.load_adapter(1, a, b)
    state.set(torch.tensor([1, 1]))
    x = torch.randn(2, 3, 16)
    got = wrapped(x)
    want = torch.nn.functional.linear(x, base.weight) + (x @ a.T @ b.T) * (4.0 / 4)
    torch.testing.assert_close(got, want, atol=1e-5, rtol=1e-5)


@torch.inference_mode()
def test_adapter_cache_lru(model):
    from neuronx_distributed_inference_amd.modules.lora_serving import AdapterCache

    cfg = LoraServingConfig(max_loras=3, max_lora_rank=4)  # 2 dynamic slots
    state = wrap_model_with_lora(model.model, cfg, max_batch=2)
    cache = AdapterCache(model.model, cfg, state)
    layers = lora_layers(model.model)
    any_layer = next(iter(layers))
    torch.manual_seed(0)

    def mk():
        layer = layers[any_layer]
        return {any_layer: (torch.randn(4, layer.lora_A.shape[2]),
                            torch.randn(layer.lora_B.shape[1], 4))}

    for nm in ("a", "b", "c"):
        cache.register_adapter(nm, mk())
    m1 = cache.ensure_loaded(["a", "b"])
    assert set(m1.values()) == {1, 2}
    m2 = cache.ensure_loaded(["c"])       # evicts "a" (LRU)
    assert m2["c"] == m1["a"]
    assert "a" not in cache.slot_of
    ids = cache.set_request_adapters(["b", ""])
    assert ids[1] == 0  # '' -> zero adapter slot


def test_multi_lora_embedding_and_conv2d():
    """Embedding/Conv2d LoRA variants (ref lora_layer.py:10-239): delta
    math vs explicit dense computation, per-request adapter selection."""
    import torch
    import torch.nn.functional as F

    from neuronx_distributed_inference_amd.modules.lora_serving import (
        LoraServingConfig,
        LoraState,
        MultiLoraConv2d,
        MultiLoraEmbedding,
    )
    from neuronx_distributed_inference_amd.parallel.layers import ParallelEmbedding

    torch.manual_seed(0)
    cfg = LoraServingConfig(max_loras=3, max_lora_rank=4, lora_alpha=8)
    state = LoraState(max_batch=2)

    emb = ParallelEmbedding(32, 16, dtype=torch.float32, tp_degree=1)
    with torch.no_grad():
        emb.weight.normal_(0, 0.1)
    ml = MultiLoraEmbedding(emb, cfg, state)
    a = torch.randn(4, 32) * 0.1   # (r, vocab)
    b = torch.randn(16, 4) * 0.1   # (hidden, r)
    ml.load_adapter(1, a, b)
    ids = torch.randint(0, 32, (2, 5))
    state.set(torch.tensor([1, 0]))
    y = ml(ids)
    base = emb(ids)
    onehot = F.one_hot(ids[0], 32).float()
    want0 = base[0] + (onehot @ a.t() @ b.t()) * (8 / 4)
    torch.testing.assert_close(y[0], want0, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(y[1], base[1])  # slot 0 = empty adapter

    conv = torch.nn.Conv2d(3, 8, 4, stride=4)
    with torch.no_grad():
        conv.weight.normal_(0, 0.1)
        conv.bias.zero_()
    mc = MultiLoraConv2d(conv, cfg, state)
    ca = torch.randn(4, 3, 4, 4) * 0.1       # (r, Cin, kh, kw)
    cb = torch.randn(8, 4) * 0.1             # (Cout, r)
    mc.load_adapter(1, ca, cb)
    x = torch.randn(2, 3, 8, 8)
    state.set(torch.tensor([1, 0]))
    y = mc(x)
    base = conv(x)
    h = F.conv2d(x[:1], ca, stride=4)
    want0 = base[0] + (F.conv2d(h, cb.view(8, 4, 1, 1)) * (8 / 4))[0]
    torch.testing.assert_close(y[0], want0, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(y[1], base[1])
