"""Offline TunableOp tuning of the decode-step GEMM shapes.

Run with PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1
PYTORCH_TUNABLEOP_FILENAME=<csv>; the bench then runs with TUNING=0 and the
saved table so tuned GEMMs are used inside hipGraph capture.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    M = int(sys.argv[1]) if len(sys.argv) > 1 else 64
    H, I, V = 4096, 14336, 128256
    shapes = [
        (M, 6144, H),       # fused qkv
        (M, 2 * I, H),      # fused gate_up
        (M, H, I),          # down
        (M, H, H),          # o_proj
        (M, V, H),          # lm_head
    ]
    dev = "cuda"
    for (m, n, k) in shapes:
        x = torch.randn(m, k, dtype=torch.bfloat16, device=dev)
        w = torch.randn(n, k, dtype=torch.bfloat16, device=dev)
        for _ in range(3):
            torch.nn.functional.linear(x, w)
        torch.cuda.synchronize()
        print(f"tuned {m}x{n}x{k}")


if __name__ == "__main__":
    main()
