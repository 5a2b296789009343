"""Tensor capture/replacement + snapshot hooks."""

import torch

from neuronx_distributed_inference_amd.models.llama.modeling_llama import LlamaForCausalLM
from neuronx_distributed_inference_amd.utils.tensor_capture import (
    TensorCapture,
    TensorReplacement,
)


@torch.inference_mode()
def test_tensor_capture_and_replacement(tiny_llama_config):
    torch.manual_seed(0)
    m = LlamaForCausalLM(tiny_llama_config).eval()
    with torch.no_grad():
        for p in m.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    ids = torch.randint(0, 128, (1, 6))
    with TensorCapture(m.model, ["layers.0.mlp", "norm"]) as cap:
        m(input_ids=ids)
    assert len(cap.captured["layers.0.mlp"]) == 1
    assert cap.captured["layers.0.mlp"][0].shape[-1] == 64

    # replacement: zero the input of layer 1's mlp and verify output changes
    m.reset()
    base = m(input_ids=ids).logits.clone()
    m.reset()
    golden = torch.zeros(1, 6, 64)
    with TensorReplacement(m.model, {"layers.1.mlp": golden}):
        repl = m(input_ids=ids).logits
    assert not torch.equal(base, repl)


def test_snapshot_hook(tiny_llama_config, tmp_path, monkeypatch):
    import os

    from neuronx_distributed_inference_amd.utils.snapshot import attach_snapshot_hook

    torch.manual_seed(0)
    m = LlamaForCausalLM(tiny_llama_config).eval()
    monkeypatch.setenv("NXDI_SNAPSHOT_OUTPUT_PATH", str(tmp_path))
    monkeypatch.setenv("NXDI_SNAPSHOT_CAPTURE_AT_REQUESTS", "0")
    hook = attach_snapshot_hook(m)
    assert hook is not None
    ids = torch.randint(0, 128, (1, 5))
    m(input_ids=ids)
    files = os.listdir(tmp_path)
    assert any(f.startswith("context_encoding_model_req0") for f in files)
    payload = torch.load(os.path.join(tmp_path, files[0]), weights_only=False)
    assert torch.equal(payload["input_ids"], ids)


def test_async_executor_overlap_and_backpressure():
    """AsyncExecutor: results arrive in order, back-pressure caps inflight,
    drain() completes everything (ref modules/async_execution.py analog)."""
    from neuronx_distributed_inference_amd.modules.async_execution import (
        AsyncExecutor,
    )

    calls = []

    def work(i):
        calls.append(i)
        return torch.full((2,), float(i))

    with AsyncExecutor(max_inflight=2) as ex:
        handles = [ex.submit(work, i) for i in range(5)]
        outs = [h.result() for h in handles]
    assert calls == list(range(5))
    for i, o in enumerate(outs):
        assert torch.equal(o, torch.full((2,), float(i)))
    assert all(h.done() for h in handles)


def test_reconstruct_kv_cache():
    from neuronx_distributed_inference_amd.utils.snapshot import (
        reconstruct_kv_cache,
    )

    full = torch.randn(2, 4, 16, 8)    # (B, Hkv=4, S, D)
    shards = [full[:, :2], full[:, 2:]]
    out = reconstruct_kv_cache(shards)
    assert torch.equal(out, full)
    rep = [full[:, :1].clone(), full[:, :1].clone()]
    out = reconstruct_kv_cache(rep)
    assert torch.equal(out, full[:, :1])
