import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

def bench(fn, n=40):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n

dev = "cuda"
M = int(sys.argv[1]) if len(sys.argv) > 1 else 256
for name, n, k in [("qkv", 6144, 4096), ("gate_up", 28672, 4096), ("down", 4096, 14336)]:
    ncopy = max(2, int(512e9 / (n * k * 8) / 128))
    ncopy = max(2, int(512e6 // (n * k)))
    ws8 = [torch.randn(n, k, device=dev).clamp(-2, 2).to(torch.float8_e4m3fn)
           for _ in range(ncopy)]
    wsb = [torch.randn(n, k, dtype=torch.bfloat16, device=dev) for _ in range(ncopy)]
    x8 = torch.randn(M, k, device=dev).clamp(-2, 2).to(torch.float8_e4m3fn)
    xb = torch.randn(M, k, dtype=torch.bfloat16, device=dev)
    one = torch.ones(1, 1, device=dev)
    it = {"i": 0}
    def nxt(ws):
        it["i"] = (it["i"] + 1) % ncopy
        return ws[it["i"]]
    t8 = bench(lambda: torch._scaled_mm(x8, nxt(ws8).t(), scale_a=one, scale_b=one,
                                        out_dtype=torch.float32))
    tb = bench(lambda: torch.nn.functional.linear(xb, nxt(wsb)))
    gb8, gbb = n * k / 1e9, n * k * 2 / 1e9
    print(f"{name:8s} M={M} fp8 {t8*1e6:7.1f}us {gb8/t8/1e3:5.2f}TB/s | "
          f"bf16 {tb*1e6:7.1f}us {gbb/tb/1e3:5.2f}TB/s")
