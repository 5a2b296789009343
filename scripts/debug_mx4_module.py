"""Module-level mx4 bisect: which GPU path diverges from its own CPU
run at the gpt-oss shape?"""
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


def run(E, H, I, T):
    torch.manual_seed(0)
    dev = torch.device("cuda:0")
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
    x = torch.randn(T, H, dtype=torch.bfloat16) * 0.3
    w, idx = torch.softmax(torch.randn(T, E), -1).topk(4, dim=-1)

    cpu_mx = mx._forward_grouped(x, w, idx)          # CPU mx grouped (ref)
    cpu_bf = bf(x, w, idx)                           # CPU bf16 loop
    mx_g, bf_g = mx.to(dev), bf.to(dev)
    xg, wg, ig = x.to(dev), w.to(dev), idx.to(dev)
    gpu_mx = mx_g(xg, wg, ig).cpu()
    gpu_bf = bf_g(xg, wg, ig).cpu()

    def d(a, b):
        return float((a.float() - b.float()).abs().max())

    print(f"E={E} H={H} I={I} T={T}:")
    print(f"  cpu_mx  vs cpu_bf : {d(cpu_mx, cpu_bf):.4f}  (mx quant consistency)")
    print(f"  gpu_mx  vs cpu_mx : {d(gpu_mx, cpu_mx):.4f}  (mx GPU path)")
    print(f"  gpu_bf  vs cpu_bf : {d(gpu_bf, cpu_bf):.4f}  (bf16 GPU loop path)")
    print(f"  gpu_mx  vs gpu_bf : {d(gpu_mx, gpu_bf):.4f}", flush=True)


def main():
    run(4, 256, 256, 64)
    run(32, 256, 256, 64)
    run(4, 2880, 2880, 64)
    run(32, 2880, 2880, 64)


if __name__ == "__main__":
    main()
