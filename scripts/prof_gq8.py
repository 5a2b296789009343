import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from neuronx_distributed_inference_amd import ops

B, Hq, Hkv, Smax, D = 16, 64, 8, 2048, 128
dev = "cuda"
kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
q = torch.randn(B, Hq, 1, D, dtype=torch.bfloat16, device=dev)
pos = torch.full((B, 1), Smax - 2, dtype=torch.int32, device=dev)
for _ in range(5):
    ops.attention_decode(q, kc, vc, pos)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(50):
    ops.attention_decode(q, kc, vc, pos)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 50
gb = B * Hkv * Smax * D * 2 * 2 / 1e9
print(f"gq8: {dt*1e6:.1f}us {gb/dt/1e3:.2f} TB/s (KV {gb:.2f} GB)")
# gq4 comparison (Hq=32)
q4 = torch.randn(B, 32, 1, D, dtype=torch.bfloat16, device=dev)
for _ in range(5):
    ops.attention_decode(q4, kc, vc, pos)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(50):
    ops.attention_decode(q4, kc, vc, pos)
torch.cuda.synchronize()
dt4 = (time.perf_counter() - t0) / 50
print(f"gq4: {dt4*1e6:.1f}us {gb/dt4/1e3:.2f} TB/s")
