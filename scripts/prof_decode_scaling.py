import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from neuronx_distributed_inference_amd import ops

dev = "cuda"

def bench(fn, n=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n

B, Hq, Hkv, D = 16, 32, 8, 128
for Smax in (2048, 8192, 32768):
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    q = torch.randn(B, Hq, 1, D, dtype=torch.bfloat16, device=dev)
    pos = torch.full((B, 1), Smax - 2, dtype=torch.int32, device=dev)
    gb = B * Hkv * Smax * D * 2 * 2 / 1e9
    t = bench(lambda: ops.attention_decode(q, kc, vc, pos))
    ts = bench(lambda: kc.sum(dtype=torch.float32) + vc.sum(dtype=torch.float32))
    print(f"S={Smax:6d} decode {t*1e6:8.1f}us {gb/t/1e3:5.2f} TB/s | raw-sum {ts*1e6:8.1f}us {gb/ts/1e3:5.2f} TB/s")

# headline decode shape: B=256, short ctx (the bench config)
for B2, S2, ctx in ((256, 176, 128), (256, 2048, 2000)):
    kc = torch.randn(B2, 8, S2, 128, dtype=torch.bfloat16, device=dev)
    vc = torch.randn(B2, 8, S2, 128, dtype=torch.bfloat16, device=dev)
    q = torch.randn(B2, 32, 1, 128, dtype=torch.bfloat16, device=dev)
    pos = torch.full((B2, 1), ctx, dtype=torch.int32, device=dev)
    gb = B2 * 8 * (ctx + 1) * 128 * 2 * 2 / 1e9
    t = bench(lambda: ops.attention_decode(q, kc, vc, pos))
    print(f"B={B2} ctx={ctx:5d} decode {t*1e6:8.1f}us "
          f"{gb/t/1e3:5.2f} TB/s effective")
