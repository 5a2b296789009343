"""Kernel perf probes (run on the GPU box).

    python scripts/perf_kernels.py [flash|decode|rmsnorm|all]

Reports TF (flash) / GB/s (memory-bound kernels) with within-run repeats.
"""

import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from neuronx_distributed_inference_amd import ops


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def flash():
    B, Hq, Hkv, S, D = 16, 64, 8, 2048, 128
    q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device="cuda")
    dt = bench(lambda: ops.attention_prefill(q, k, v, causal=True))
    flops = 2 * 2 * B * Hq * S * S * D * 0.5  # causal
    print(f"flash_prefill B{B} Hq{Hq} Hkv{Hkv} S{S} D{D}: {dt*1e3:.3f} ms  "
          f"{flops/dt/1e12:.1f} TF")


def decode():
    for B, Hkv, gq, L in [(8, 8, 4, 2048), (32, 8, 4, 2048), (8, 1, 8, 8192)]:
        D, Smax = 128, L
        Hq = Hkv * gq
        kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device="cuda")
        vc = torch.randn_like(kc)
        q = torch.randn(B, Hq, 1, D, dtype=torch.bfloat16, device="cuda")
        seq_lens = torch.full((B,), L, device="cuda", dtype=torch.int32)
        dt = bench(lambda: ops.attention_decode(q, kc, vc, seq_lens))
        bytes_ = B * Hkv * L * D * 2 * 2
        print(f"attention_decode B{B} Hkv{Hkv} gq{gq} L{L}: {dt*1e6:.1f} us  "
              f"{bytes_/dt/1e9:.0f} GB/s")


def rmsnorm():
    for rows, H in [(8, 4096), (1024, 4096), (8192, 8192)]:
        x = torch.randn(rows, H, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(H, dtype=torch.bfloat16, device="cuda")
        dt = bench(lambda: ops.rms_norm(x, w, 1e-5))
        bytes_ = rows * H * 2 * 2
        print(f"rms_norm {rows}x{H}: {dt*1e6:.1f} us  {bytes_/dt/1e9:.0f} GB/s")


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    if which in ("flash", "all"):
        flash()
    if which in ("decode", "all"):
        decode()
    if which in ("rmsnorm", "all"):
        rmsnorm()
