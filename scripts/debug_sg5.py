"""sg5 correctness bisect on GPU: modes (0=pipelined glds, 1=serial glds,
2=serial register staging) x K sweep x chunk isolation."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from neuronx_distributed_inference_amd import ops  # noqa: E402


def err(y, want):
    return (y.float() - want.float()).abs().max().item()


def main():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    print("=== mode x K sweep (N=256) ===", flush=True)
    for M in (64, 256):
        for K in (64, 128, 192, 256, 512, 1024):
            x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
            w = torch.randn(256, K, dtype=torch.bfloat16, device=dev) * 0.1
            want = torch.nn.functional.linear(x, w)
            line = f"M={M:3d} K={K:5d}: "
            for mode in (0, 1, 2):
                e = err(ops.skinny_gemm5(x, w, mode=mode), want)
                line += f"m{mode}={e:.4f} "
            print(line, flush=True)

    print("=== chunk isolation (M=64 K=1024 N=256, mode 0) ===", flush=True)
    M, K, N = 64, 1024, 256
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.1
    for c in range(K // 64):
        x = torch.zeros(M, K, dtype=torch.bfloat16, device=dev)
        x[:, c * 64:(c + 1) * 64] = torch.randn(M, 64, device=dev) * 0.1
        want = torch.nn.functional.linear(x, w)
        e = err(ops.skinny_gemm5(x, w), want)
        print(f"chunk {c:2d} (buf {c % 3}): err={e:.4f}", flush=True)

    print("=== row/col error pattern (M=64 K=1024 N=1536, mode 0) ===", flush=True)
    M, K, N = 64, 1024, 1536
    torch.manual_seed(0)
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.1
    want = torch.nn.functional.linear(x, w)
    y = ops.skinny_gemm5(x, w)
    bad = (y.float() - want.float()).abs() > 0.02
    print("bad frac per row (first 16):",
          [round(v, 2) for v in bad.float().mean(1)[:16].tolist()], flush=True)
    bycol = bad.float().mean(0).view(-1, 64).mean(1)  # per 64-col tile
    print("bad frac per n-tile:", [round(v, 2) for v in bycol.tolist()], flush=True)
    # repeat determinism check
    y2 = ops.skinny_gemm5(x, w)
    print("deterministic:", bool((y == y2).all()), flush=True)


if __name__ == "__main__":
    main()
