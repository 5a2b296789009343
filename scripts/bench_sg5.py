"""Microbench: skinny_gemm5 splitk variants vs the autotuned hipBLASLt
path on the real decode projection shapes (Llama-3.1-8B TP=1 plus the
70B/Mixtral shapes). Cold-weight protocol from r01: rotate 10 weight
copies so the stream is MALL-cold; within-probe interleaved timing.

Usage: python scripts/bench_sg5.py [--m 256] [--iters 40]
"""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from neuronx_distributed_inference_amd import ops  # noqa: E402


def bench_shape(M, N, K, iters=40, ncopies=10):
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
    ws = [torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.1
          for _ in range(ncopies)]

    want = torch.nn.functional.linear(x.float(), ws[0].float())

    def timed(fn):
        for i in range(4):
            fn(ws[i % ncopies])
        torch.cuda.synchronize()
        t0 = torch.cuda.Event(enable_timing=True)
        t1 = torch.cuda.Event(enable_timing=True)
        t0.record()
        for i in range(iters):
            fn(ws[i % ncopies])
        t1.record()
        torch.cuda.synchronize()
        return t0.elapsed_time(t1) / iters * 1000  # us

    res = {}
    res["lt"] = timed(lambda w: ops.lt_linear(x, w))
    for mt, sk in ops._sg5_candidates(M, N, K):
        y = ops.skinny_gemm5(x, ws[0], splitk=sk, mt=mt)
        err = (y.float() - want).abs().max().item()
        res[f"sg5_m{mt}k{sk}"] = timed(
            lambda w, s=sk, m=mt: ops.skinny_gemm5(x, w, splitk=s, mt=m))
        res[f"sg5_m{mt}k{sk}_err"] = round(err, 5)
    gb = N * K * 2 / 1e9
    best = min(v for k, v in res.items() if not k.endswith("_err"))
    res["w_mb"] = round(gb * 1000, 1)
    res["best_tbs"] = round(gb / (best / 1e6) / 1000, 2)
    return res


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--m", type=int, default=0, help="0 = sweep 64/128/256")
    ap.add_argument("--iters", type=int, default=40)
    args = ap.parse_args()

    shapes = [
        ("8b_qkv", 6144, 4096),
        ("8b_o", 4096, 4096),
        ("8b_gate_up", 28672, 4096),
        ("8b_down", 4096, 14336),
        ("70b_qkv_tp8", 1280, 8192),
        ("70b_gate_up_tp8", 7168, 8192),
    ]
    ms = [args.m] if args.m else [64, 128, 256]
    out = {}
    for M in ms:
        for name, N, K in shapes:
            r = bench_shape(M, N, K, iters=args.iters)
            out[f"{name}_M{M}"] = r
            lt = r["lt"]
            best_k = min((k for k in r if k.startswith("sg5_") and not
                          k.endswith("_err")), key=r.get, default=None)
            line = f"{name:18s} M={M:3d} N={N:6d} K={K:6d}  lt={lt:7.1f}us"
            if best_k:
                line += (f"  {best_k}={r[best_k]:7.1f}us"
                         f"  ratio={r[best_k]/lt:.2f}  ({r['best_tbs']} TB/s)")
            print(line, flush=True)
    with open("gpurun_out/sg5_bench.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
