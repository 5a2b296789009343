"""Split grouped-MoE time into GEMM vs glue (tables/gather/scatter) at
the Mixtral decode shape, and time each piece."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from neuronx_distributed_inference_amd import ops  # noqa: E402
from neuronx_distributed_inference_amd.modules.moe import ExpertMLPs  # noqa: E402


def t(fn, n=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    a, b = torch.cuda.Event(True), torch.cuda.Event(True)
    a.record()
    for _ in range(n):
        fn()
    b.record()
    torch.cuda.synchronize()
    return a.elapsed_time(b) / n * 1000


def main():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    E, H, I, T, topk = 8, 4096, 14336, 256, 2
    m = ExpertMLPs(num_experts=E, hidden_size=H, intermediate_size=I,
                   tp_degree=1, dtype=torch.bfloat16).to(dev)
    with torch.no_grad():
        for p in m.parameters():
            p.normal_(0, 0.02)
    x = torch.randn(T, H, dtype=torch.bfloat16, device=dev) * 0.3
    w, idx = torch.softmax(torch.randn(T, E, device=dev), -1).topk(topk, dim=-1)

    print(f"full grouped: {t(lambda: m._forward_grouped(x, w, idx)):8.1f} us")
    print(f"full dense  : {t(lambda: m._forward_dense(x, w, idx)):8.1f} us")

    flat = idx.view(-1)
    print(f"tables(fused): {t(lambda: ops.moe_grouped_tables(flat, E)):8.1f} us")
    order, dest, expert_ids, gtab, Mpad = ops.moe_grouped_tables(flat, E)
    tok = order // topk
    x_sorted = torch.zeros(Mpad, H, dtype=torch.bfloat16, device=dev)

    def gather():
        x_sorted.index_copy_(0, dest, x.index_select(0, tok))
    print(f"gather      : {t(gather):8.1f} us")
    gather()
    wmat = m.w_gate
    # numerics + perf for each column-tile width
    ref_g = ops.moe_grouped_gemm(x_sorted, wmat, expert_ids, bn=64)
    for bn, bk in ((64, 64), (128, 64), (256, 64), (64, 128), (128, 128)):
        gg = ops.moe_grouped_gemm(x_sorted, wmat, expert_ids, bn=bn, bk=bk)
        err = (gg.index_select(0, dest).float()
               - ref_g.index_select(0, dest).float()).abs().max().item()
        print(f"gemm gate (N={I}) bn={bn} bk={bk}: "
              f"{t(lambda: ops.moe_grouped_gemm(x_sorted, wmat, expert_ids, bn=bn, bk=bk)):8.1f} us"
              f"  maxdiff {err:.4f}  [{Mpad} rows; weight {E*I*H*2/1e6:.0f} MB]")
    g = ops.moe_grouped_gemm(x_sorted, wmat, expert_ids)
    u = ops.moe_grouped_gemm(x_sorted, m.w_up, expert_ids)
    h = ops.silu_mul(g, u)
    ref_d = ops.moe_grouped_gemm(h, m.w_down, expert_ids, bn=64)
    for bn, bk in ((64, 64), (128, 64), (256, 64), (64, 128), (128, 128)):
        dd = ops.moe_grouped_gemm(h, m.w_down, expert_ids, bn=bn, bk=bk)
        err = (dd.index_select(0, dest).float()
               - ref_d.index_select(0, dest).float()).abs().max().item()
        print(f"gemm down (N={H}) bn={bn} bk={bk}: "
              f"{t(lambda: ops.moe_grouped_gemm(h, m.w_down, expert_ids, bn=bn, bk=bk)):8.1f} us"
              f"  maxdiff {err:.4f}")
    y = ops.moe_grouped_gemm(h, m.w_down, expert_ids)

    def combine():
        contrib = y.index_select(0, dest) * w.view(-1)[order].to(y.dtype).unsqueeze(-1)
        out = torch.zeros(T, H, dtype=y.dtype, device=dev)
        out.index_add_(0, tok, contrib)
        return out
    print(f"combine     : {t(combine):8.1f} us")
    # ideal: all weights once at 6.3 TB/s
    wb = E * (2 * I * H + H * I) * 2
    print(f"weight bytes/layer {wb/1e9:.2f} GB -> floor {wb/6.3e12*1e6:.0f} us")


if __name__ == "__main__":
    main()
