"""Steady-state decode-GEMM config sweep.

The cold rotating-clone microbench tops out ~2.3 TB/s on every config,
but the REAL bench shows the down proj streaming at 5.8 TB/s — cold
rankings evidently do not transfer. This sweep re-ranks (lt vs sg5
(mt, splitk)) inside a loop that mimics the real decode layer walk:
32 layers x {qkv, o, gate_up, down} with DISTINCT weights (16 GB
streamed per pass, L3-defeating, DVFS-realistic). One shape's config is
varied at a time; its per-call cost is the total-loop delta / 32.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from neuronx_distributed_inference_amd import ops  # noqa: E402

L = 32
M = 256
SHAPES = {               # name: (N, K)
    "qkv": (6144, 4096),
    "o": (4096, 4096),
    "gate_up": (28672, 4096),
    "down": (4096, 14336),
}
BASE = {"qkv": None, "o": None, "gate_up": None, "down": None}  # None = lt


def make_weights(dev):
    ws = {}
    for name, (N, K) in SHAPES.items():
        ws[name] = [
            (torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.02)
            for _ in range(L)
        ]
    return ws


def run_pass(ws, xs, cfg):
    for li in range(L):
        for name in ("qkv", "o", "gate_up", "down"):
            c = cfg[name]
            w = ws[name][li]
            x = xs[name]
            if c is None:
                ops.lt_linear(x, w)
            else:
                mt, sk = c
                ops.skinny_gemm5(x, w, splitk=sk, mt=mt)


def time_cfg(ws, xs, cfg, n=6):
    for _ in range(2):
        run_pass(ws, xs, cfg)
    torch.cuda.synchronize()
    a, b = torch.cuda.Event(True), torch.cuda.Event(True)
    a.record()
    for _ in range(n):
        run_pass(ws, xs, cfg)
    b.record()
    torch.cuda.synchronize()
    return a.elapsed_time(b) / n  # ms per pass


def main():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    print("allocating 16 GB of layer weights...", flush=True)
    ws = make_weights(dev)
    xs = {n: torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
          for n, (N, K) in SHAPES.items()}

    base = dict(BASE)
    t_base = time_cfg(ws, xs, base)
    print(f"baseline (all lt): {t_base:8.3f} ms/pass", flush=True)

    for name, (N, K) in SHAPES.items():
        print(f"--- {name} (N={N} K={K}; w {N*K*2/1e6:.0f} MB, "
              f"floor {N*K*2/6.3e12*1e6:.1f} us) ---", flush=True)
        cands = [None] + ops._sg5_candidates(M, N, K)
        best = (None, float("inf"))
        for c in cands:
            cfg = dict(base)
            cfg[name] = c
            t = time_cfg(ws, xs, cfg)
            per_call = (t - t_base) * 1000 / L  # delta us vs lt baseline
            label = "lt" if c is None else f"sg5 mt{c[0]} sk{c[1]}"
            print(f"  {label:16s}: {t:8.3f} ms/pass "
                  f"(delta {per_call:+7.1f} us/call)", flush=True)
            if t < best[1]:
                best = (c, t)
        label = "lt" if best[0] is None else f"sg5 mt{best[0][0]} sk{best[0][1]}"
        print(f"  => best: {label}", flush=True)


if __name__ == "__main__":
    main()
