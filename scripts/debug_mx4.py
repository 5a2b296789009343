"""Bisect the mx4 grouped kernel at the gpt-oss shape."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from neuronx_distributed_inference_amd import ops  # noqa: E402


def check(E, N, K, Ttk, bias=False):
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    w = torch.randn(E, N, K) * 0.2
    wq, ws = ops.mx4_quantize(w)
    b = (torch.randn(E, N, dtype=torch.bfloat16) * 0.1) if bias else None
    idx = torch.randint(0, E, (Ttk,))
    order, dest, expert_ids, gtab, Mpad = ops.moe_grouped_tables(idx, E)
    x_rows = torch.randn(Ttk, K, dtype=torch.bfloat16) * 0.1
    x_sorted = torch.zeros(Mpad, K, dtype=torch.bfloat16)
    x_sorted.index_copy_(0, dest, x_rows)
    want = ops.moe_grouped_gemm_mx4(x_sorted, wq, ws, expert_ids, bias=b)
    got = ops.moe_grouped_gemm_mx4(
        x_sorted.to(dev), wq.to(dev), ws.to(dev), expert_ids.to(dev),
        bias=None if b is None else b.to(dev))
    d = (got.index_select(0, dest.to(dev)).float().cpu()
         - want.index_select(0, dest).float()).abs()
    print(f"E={E} N={N} K={K} Ttk={Ttk} bias={bias}: maxerr={d.max():.4f}",
          flush=True)
    return float(d.max())


def main():
    check(1, 64, 2880, 64)          # K=2880 alone
    check(1, 5760, 64, 64)          # N=5760 alone
    check(1, 5760, 2880, 64)        # both
    check(32, 5760, 2880, 64)       # + experts
    check(32, 5760, 2880, 1024)     # the failing config
    check(32, 5760, 2880, 1024, bias=True)
    check(1, 64, 192, 64)           # nchunks=3 (multiple of 3)
    check(1, 64, 576, 64)           # nchunks=9


if __name__ == "__main__":
    main()
