"""Microbench: skinny_gemm vs hipBLASLt (tuned) vs torch F.linear for the
decode GEMM shapes. Prints GB/s of weight traffic (the roofline currency).
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.ops import _nxdi_C as C  # noqa


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    M = int(sys.argv[1]) if len(sys.argv) > 1 else 128
    H, I, V = 4096, 14336, 128256
    shapes = [
        ("qkv", M, 6144, H),
        ("o_proj", M, H, H),
        ("gate_up", M, 2 * I, H),
        ("down", M, H, I),
        ("lm_head", M, V, H),
    ]
    dev = "cuda"
    for name, m, n, k in shapes:
        # COLD-weight measurement: rotate through enough W copies that the
        # working set exceeds the 256 MB MALL (the real decode regime
        # cycles 32 layers of distinct weights)
        ncopy = max(2, int(512e6 / (n * k * 2)))
        x = torch.randn(m, k, dtype=torch.bfloat16, device=dev) * 0.1
        ws = [torch.randn(n, k, dtype=torch.bfloat16, device=dev) * 0.1
              for _ in range(ncopy)]
        y1 = torch.empty(m, n, dtype=torch.bfloat16, device=dev)
        C.skinny_gemm(y1, x, ws[0])
        ref = torch.nn.functional.linear(x, ws[0])
        rel = ((y1.float() - ref.float()).abs().max()
               / ref.float().abs().max()).item()
        gb = n * k * 2 / 1e9
        it = {"i": 0}

        def nxt():
            it["i"] = (it["i"] + 1) % ncopy
            return ws[it["i"]]

        # v2 blocked kernel with K-split + fp32 atomic accumulator
        y32 = torch.zeros(m, n, dtype=torch.float32, device=dev)
        ks = 1
        for cand in (1, 2, 4, 8, 16, 32):
            if k % (128 * cand) == 0 and (n // 128) * cand >= 256:
                ks = cand
                break
        if k % 128 == 0 and n % 128 == 0:
            def run2():
                if ks > 1:
                    y32.zero_()
                C.skinny_gemm2(y1, y32, x, nxt(), ksplit=ks)
            C.skinny_gemm2(y1, y32, x, ws[0], ksplit=ks)
            rel2 = ((y1.float() - ref.float()).abs().max()
                    / ref.float().abs().max()).item()
            t_sk2 = bench(run2)
        else:
            rel2, t_sk2 = float("nan"), float("inf")
        t_sk = bench(lambda: C.skinny_gemm(y1, x, nxt()))
        # stride-padded W copies: decorrelate the HBM channel mapping
        pad = 64
        wps = [torch.empty(n, k + pad, dtype=torch.bfloat16, device=dev)[:, :k]
               for _ in range(ncopy)]
        for a, b in zip(wps, ws):
            a.copy_(b)
        itp = {"i": 0}

        def nxtp():
            itp["i"] = (itp["i"] + 1) % ncopy
            return wps[itp["i"]]

        t_skp = bench(lambda: C.skinny_gemm(y1, x, nxtp()))
        del wps
        t_t = bench(lambda: torch.nn.functional.linear(x, nxt()))
        t_lt = bench(lambda: ops.lt_linear(x, nxt()))
        print(f"{name:8s} M={m:4d} N={n:6d} K={k:5d} relerr={rel:.1e} "
              f"copies={ncopy} | "
              f"sk2(ks={ks}) {t_sk2*1e6:6.1f}us {gb/t_sk2/1e3:5.2f}TB/s rel={rel2:.0e} | "
              f"skpad {t_skp*1e6:7.1f}us {gb/t_skp/1e3:5.2f}TB/s | "
              f"torch {t_t*1e6:7.1f}us {gb/t_t/1e3:5.2f}TB/s | "
              f"lt {t_lt*1e6:7.1f}us {gb/t_lt/1e3:5.2f}TB/s")


if __name__ == "__main__":
    main()
