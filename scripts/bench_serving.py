"""Serving-engine benchmark: TPOT at partial occupancy (the decode
batch-bucket ladder win) + async depth A/B.

An engine captured at batch 64 serving only 8 live requests should pay
~batch-8-graph cost per token, not full-batch cost.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from neuronx_distributed_inference_amd.config import (  # noqa: E402
    EngineConfig,
    OnDeviceSamplingConfig,
)
from neuronx_distributed_inference_amd.models.llama.modeling_llama import (  # noqa: E402
    LlamaForCausalLM,
    LlamaInferenceConfig,
)
from neuronx_distributed_inference_amd.serving import ServingEngine  # noqa: E402


def build(batch, ladder=True):
    ec = EngineConfig(
        batch_size=batch, max_batch_size=batch, seq_len=512, max_length=512,
        max_context_length=128, torch_dtype="bfloat16", tp_degree=1,
        on_device_sampling_config=OnDeviceSamplingConfig(top_k=1),
        enable_batch_bucketing=ladder, fused_qkv=True, fused_gate_up=True,
    )
    cfg = LlamaInferenceConfig(
        neuron_config=ec, vocab_size=32000, hidden_size=4096,
        intermediate_size=14336, num_hidden_layers=32,
        num_attention_heads=32, num_key_value_heads=8,
        max_position_embeddings=1024, rms_norm_eps=1e-5, rope_theta=500000.0,
        hidden_act="silu", tie_word_embeddings=False)
    with torch.device("cuda:0"):
        m = LlamaForCausalLM(cfg).eval()
    with torch.no_grad():
        for p in m.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.02)
    m.reset()
    m.enable_submodels()
    with torch.inference_mode():
        m.warmup()
    return m


@torch.inference_mode()
def tpot(m, n_live, new_tokens=48, async_depth=None):
    m.reset()
    eng = ServingEngine(m, eos_token_id=None, async_depth=async_depth)
    g = torch.Generator().manual_seed(0)
    for _ in range(n_live):
        eng.add_request(torch.randint(0, 32000, (96,), generator=g),
                        max_new_tokens=new_tokens)
    # prefill everyone first (not timed)
    while any(r.prefill_done < r.prompt.shape[0] for r in eng.active.values()) \
            or eng.queue:
        eng.step()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    steps = 0
    while eng.has_work() and steps < new_tokens + 8:
        eng.step()
        steps += 1
    eng._drain_inflight(0)
    torch.cuda.synchronize()
    dt = time.monotonic() - t0
    return dt / max(steps, 1) * 1000


def main():
    m = build(64, ladder=True)
    print("engine batch 64, LADDER [8..64] captured", flush=True)
    for live in (8, 16, 32, 64):
        t = tpot(m, live)
        print(f"live={live:3d}: TPOT {t:6.2f} ms/step (async=2)", flush=True)
    t_sync = tpot(m, 16, async_depth=1)
    print(f"live= 16: TPOT {t_sync:6.2f} ms/step (async=1 control)", flush=True)
    del m
    torch.cuda.empty_cache()
    m1 = build(64, ladder=False)
    print("engine batch 64, SINGLE full-batch graph", flush=True)
    for live in (8, 64):
        t = tpot(m1, live)
        print(f"live={live:3d}: TPOT {t:6.2f} ms/step", flush=True)


if __name__ == "__main__":
    main()
