"""mx4 v3 numerics bisect at deep K: which (row, k-chunk) pattern is
wrong vs the CPU dequant reference?"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from neuronx_distributed_inference_amd import ops  # noqa: E402


def run(K, N=128, label=""):
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    E, M = 1, 64
    eids = torch.zeros(1, dtype=torch.int32, device=dev)
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.2
    w = torch.randn(E, N, K) * 0.1
    q, sc = ops.mx4_quantize(w)
    q, sc = q.to(dev), sc.to(dev)
    os.environ["NXDI_MX4_V3"] = "1"
    got = ops.moe_grouped_gemm_mx4(x, q, sc, eids)
    del os.environ["NXDI_MX4_V3"]
    v1 = ops.moe_grouped_gemm_mx4(x, q, sc, eids)
    d = (got.float() - v1.float()).abs()
    print(f"{label} K={K} N={N}: v3-vs-v1 maxdiff {d.max().item():.4f} "
          f"(v1 absmax {v1.float().abs().max().item():.3f})")
    if d.max() > 0.05:
        bad = (d > d.max() * 0.5).nonzero()
        rows = bad[:, 0].unique()
        cols = bad[:, 1].unique()
        print(f"  bad rows m: {rows[:8].tolist()}...{len(rows)} total; "
              f"bad cols n: {cols[:12].tolist()}... {len(cols)} total")
        # per-k attribution: recompute with only one k-chunk of weights
        # nonzero
        for t in range(K // 256):
            wz = torch.zeros_like(w)
            wz[:, :, t * 256:(t + 1) * 256] = w[:, :, t * 256:(t + 1) * 256]
            qz, sz = ops.mx4_quantize(wz)
            qz, sz = qz.to(dev), sz.to(dev)
            os.environ["NXDI_MX4_V3"] = "1"
            g3 = ops.moe_grouped_gemm_mx4(x, qz, sz, eids)
            del os.environ["NXDI_MX4_V3"]
            g1 = ops.moe_grouped_gemm_mx4(x, qz, sz, eids)
            dd = (g3.float() - g1.float()).abs().max().item()
            if dd > 0.02:
                print(f"  chunk {t}: maxdiff {dd:.4f}")


def main():
    run(512, label="small")
    run(1024)
    run(2048)
    run(4096)


if __name__ == "__main__":
    main()


def perf():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    E, H, I, T, topk = 8, 4096, 14336, 256, 2
    idx = torch.randint(0, E, (T * topk,), device=dev)
    order, dest, eids, gtab, Mpad = ops.moe_grouped_tables(idx, E, topk)
    x = torch.randn(T, H, dtype=torch.bfloat16, device=dev) * 0.3
    w = torch.randn(E, I, H) * 0.05
    q4, s4 = ops.mx4_quantize(w)
    q4, s4 = q4.to(dev), s4.to(dev)

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

    os.environ.pop("NXDI_MX4_V3", None)
    t1 = t(lambda: ops.moe_grouped_gemm_mx4(x, q4, s4, eids, x_gather=gtab))
    os.environ["NXDI_MX4_V3"] = "1"
    t3 = t(lambda: ops.moe_grouped_gemm_mx4(x, q4, s4, eids, x_gather=gtab))
    del os.environ["NXDI_MX4_V3"]
    print(f"mixtral gate mx4: v1 {t1:7.1f} us | v3-hw {t3:7.1f} us")


if __name__ == "__main__":
    perf()
