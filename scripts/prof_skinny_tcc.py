import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from neuronx_distributed_inference_amd.ops import _nxdi_C as C

M, N, K = 128, 6144, 4096
dev = "cuda"
x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
ws = [torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.1 for _ in range(10)]
y = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
mode = sys.argv[1] if len(sys.argv) > 1 else "skinny"
for i in range(40):
    w = ws[i % 10]
    if mode == "skinny":
        C.skinny_gemm(y, x, w)
    else:
        torch.nn.functional.linear(x, w)
torch.cuda.synchronize()
print("done", mode)
