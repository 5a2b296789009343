"""mx4 v3 (all-glds BK=256) vs bf16 vs fp8 grouped GEMMs at the Mixtral
decode shape."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


from neuronx_distributed_inference_amd import ops  # noqa: E402


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


def main():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    E, H, I, T, topk = 8, 4096, 14336, 256, 2
    idx = torch.randint(0, E, (T * topk,), device=dev)
    order, dest, eids, gtab, Mpad = ops.moe_grouped_tables(idx, E, topk)
    x = torch.randn(T, H, dtype=torch.bfloat16, device=dev) * 0.3
    w = torch.randn(E, I, H) * 0.05
    wb = w.to(torch.bfloat16).to(dev)
    q4, s4 = ops.mx4_quantize(w)
    q4, s4 = q4.to(dev), s4.to(dev)
    q8, s8 = ops.fp8_moe_quantize(w)
    q8, s8 = q8.to(dev), s8.to(dev)

    y_bf = ops.moe_grouped_gemm(x, wb, eids, x_gather=gtab)
    y_m4 = ops.moe_grouped_gemm_mx4(x, q4, s4, eids, x_gather=gtab)
    y_f8 = ops.moe_grouped_gemm_fp8(x, q8, s8, eids, x_gather=gtab)
    live = dest
    rel4 = ((y_m4.index_select(0, live).float()
             - y_bf.index_select(0, live).float()).abs().max()
            / y_bf.float().abs().max().clamp(min=1e-6)).item()
    rel8 = ((y_f8.index_select(0, live).float()
             - y_bf.index_select(0, live).float()).abs().max()
            / y_bf.float().abs().max().clamp(min=1e-6)).item()
    print(f"gate GEMM N={I} K={H}:  relerr mx4 {rel4:.4f}  fp8 {rel8:.4f}")
    tb = t(lambda: ops.moe_grouped_gemm(x, wb, eids, x_gather=gtab))
    t4 = t(lambda: ops.moe_grouped_gemm_mx4(x, q4, s4, eids, x_gather=gtab))
    t8 = t(lambda: ops.moe_grouped_gemm_fp8(x, q8, s8, eids, x_gather=gtab))
    wgb = E * I * H
    print(f"  bf16 {tb:7.1f} us ({wgb*2/1e6:.0f} MB)  "
          f"fp8 {t8:7.1f} us ({wgb/1e6:.0f} MB)  "
          f"mx4 {t4:7.1f} us ({wgb*0.53/1e6:.0f} MB)")


if __name__ == "__main__":
    main()
