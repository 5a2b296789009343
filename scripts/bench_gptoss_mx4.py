"""GPT-OSS MoE layer: MXFP4 (packed weights, in-LDS dequant grouped
kernel) vs bf16 expert paths at decode batches — the VERDICT r01 target
is >=1.6x bf16 tok/s for the layer's weight-bound decode step.

GPT-OSS-120B geometry per layer: 128 experts top-4, H=2880, I=2880.
(gpt-oss-20b: 32 experts top-4, H=2880, I=2880.)"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from neuronx_distributed_inference_amd import ops  # noqa: E402
from neuronx_distributed_inference_amd.config import EngineConfig  # noqa: E402
from neuronx_distributed_inference_amd.models.gpt_oss.modeling_gpt_oss import (  # noqa: E402
    GptOssExperts,
    GptOssInferenceConfig,
)


def t(fn, n=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    a, b = torch.cuda.Event(True), torch.cuda.Event(True)
    a.record()
    for _ in range(n):
        fn()
    b.record()
    torch.cuda.synchronize()
    return a.elapsed_time(b) / n * 1000


def build(mx, E, H, I):
    ec = EngineConfig(batch_size=4, seq_len=64, max_length=64,
                      torch_dtype="bfloat16", tp_degree=1, moe_mxfp4=mx)
    cfg = GptOssInferenceConfig(
        neuron_config=ec, vocab_size=64, hidden_size=H, intermediate_size=I,
        num_hidden_layers=1, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=E, num_experts_per_tok=4, max_position_embeddings=64,
        rms_norm_eps=1e-5, rope_theta=10000.0,
        layer_types=["full_attention"], tie_word_embeddings=False)
    return GptOssExperts(cfg)


def main():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    E, H, I = 32, 2880, 2880   # gpt-oss-20b geometry
    wb_bf16 = E * (2 * I * H + H * I) * 2 / 1e9
    wb_mx4 = wb_bf16 / 16 * 4.25
    print(f"experts={E} H={H} I={I}: weights bf16 {wb_bf16:.2f} GB, "
          f"mxfp4 {wb_mx4:.2f} GB", flush=True)

    mx = build(True, E, H, I)
    w_gu = torch.randn(E, 2 * I, H) * 0.2
    w_dn = torch.randn(E, H, I) * 0.2
    with torch.no_grad():
        q, s = ops.mx4_quantize(w_gu)
        mx.gate_up_q.copy_(q); mx.gate_up_s.copy_(s)
        q, s = ops.mx4_quantize(w_dn)
        mx.down_q.copy_(q); mx.down_s.copy_(s)
        mx.gate_up_proj_bias.normal_(0, 0.02)
        mx.down_proj_bias.normal_(0, 0.02)
    bf = build(False, E, H, I)
    with torch.no_grad():
        bf.gate_up_proj.copy_(ops.mx4_dequant(mx.gate_up_q, mx.gate_up_s))
        bf.down_proj.copy_(ops.mx4_dequant(mx.down_q, mx.down_s))
        bf.gate_up_proj_bias.copy_(mx.gate_up_proj_bias)
        bf.down_proj_bias.copy_(mx.down_proj_bias)
    mx, bf = mx.to(dev), bf.to(dev)

    for T in (64, 256):
        x = torch.randn(T, H, dtype=torch.bfloat16, device=dev) * 0.3
        w, idx = torch.softmax(torch.randn(T, E, device=dev), -1).topk(4, dim=-1)
        # numerics sanity
        got = mx(x, w, idx)
        want = bf(x, w, idx)
        # RELATIVE error: outputs are O(100+) at this scale (clamped-GLU
        # boundary effects amplify bf16 rounding absolutely, not
        # relatively)
        err = ((got.float() - want.float()).abs().max()
               / want.float().abs().max().clamp(min=1)).item()
        t_mx = t(lambda: mx(x, w, idx))
        os.environ["NXDI_MOE_GROUPED"] = "1"
        t_bf_grouped = t(lambda: bf(x, w, idx))
        os.environ["NXDI_MOE_GROUPED"] = "0"
        t_bf_loop = t(lambda: bf(x, w, idx))
        os.environ.pop("NXDI_MOE_GROUPED")
        print(f"T={T:4d}: mxfp4 {t_mx:7.1f} us | bf16 grouped {t_bf_grouped:7.1f}"
              f" | bf16 loop {t_bf_loop:7.1f} | speedup vs best bf16 "
              f"{min(t_bf_grouped, t_bf_loop)/t_mx:.2f}x | relerr {err:.4f}",
              flush=True)


if __name__ == "__main__":
    main()
