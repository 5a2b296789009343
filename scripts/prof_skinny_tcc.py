"""TCC/SQ counter deep-dive: why hand-written weight-streaming GEMM kernels
plateau at ~1 TB/s while hipBLASLt reaches ~3.5 TB/s on the same skinny
decode shape (M=128, N=6144, K=4096).

Modes: skinny (v1), skinny2 (v2 blocked K-split), lib (F.linear/rocBLAS),
lt (autotuned hipBLASLt). 10 rotating 50 MB weight copies (503 MB total)
keep the stream MALL-cold, matching the decode regime.
"""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from neuronx_distributed_inference_amd.ops import _nxdi_C as C

M, N, K = 128, 6144, 4096
dev = "cuda"
x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
ws = [torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.1 for _ in range(10)]
y = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
y32 = torch.empty(M, N, dtype=torch.float32, device=dev)
mode = sys.argv[1] if len(sys.argv) > 1 else "skinny"
if mode == "lt":
    C.lt_tune(M, N, K, False, x, ws[0], y, None)
for i in range(40):
    w = ws[i % 10]
    if mode == "skinny":
        C.skinny_gemm(y, x, w, None)
    elif mode.startswith("skinny2"):
        ks = int(mode[7:] or 8)
        C.skinny_gemm2(y, y32, x, w, None, ks)
    elif mode == "skinny3":
        C.skinny_gemm3(y, x, w, None)
    elif mode == "skinny4":
        C.skinny_gemm4(y, x, w, None)
    elif mode == "lt":
        C.lt_linear(y, x, w, None)
    else:
        torch.nn.functional.linear(x, w)
torch.cuda.synchronize()
print("done", mode)
