import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from neuronx_distributed_inference_amd.ops import _nxdi_C as C

M, N, K = 128, 6144, 4096
dev = "cuda"
x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
ws = [torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.1 for _ in range(10)]
y = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
for i in range(60):
    C.skinny_gemm(y, x, ws[i % 10])
torch.cuda.synchronize()
print("done")
