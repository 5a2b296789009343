"""Isolate which mx4 fusion (x_gather staging | combine epilogue) breaks
on GPU: ops-level A/B against the materialized path."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from neuronx_distributed_inference_amd import ops  # noqa: E402


def main():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    E, T, top_k, K, N = 4, 100, 2, 256, 256
    idx = torch.randint(0, E, (T * top_k,), device=dev)
    order, dest, eids, gtab, Mpad = ops.moe_grouped_tables(idx, E, top_k)
    x = torch.randn(T, K, dtype=torch.bfloat16, device=dev) * 0.3
    w = torch.randn(E, N, K) * 0.2
    q, s = ops.mx4_quantize(w)
    q, s = q.to(dev), s.to(dev)
    gatew = torch.rand(T * top_k, device=dev)[order]

    tok = order // top_k
    xs = torch.zeros(Mpad, K, dtype=torch.bfloat16, device=dev)
    xs.index_copy_(0, dest, x.index_select(0, tok))

    y_base = ops.moe_grouped_gemm_mx4(xs, q, s, eids)
    y_gather = ops.moe_grouped_gemm_mx4(x, q, s, eids, x_gather=gtab)
    d = (y_base.index_select(0, dest).float()
         - y_gather.index_select(0, dest).float()).abs().max().item()
    print(f"x_gather vs materialized: maxdiff {d:.5f}  "
          f"(y_base absmax {y_base.float().abs().max().item():.3f})")

    gw = torch.zeros(Mpad, dtype=torch.float32, device=dev)
    gw[dest] = gatew.float()
    want = torch.zeros(T, N, dtype=torch.float32, device=dev)
    contrib = y_base.index_select(0, dest).float() * gatew.float().unsqueeze(-1)
    want.index_add_(0, tok, contrib)
    out32 = torch.zeros(T, N, dtype=torch.float32, device=dev)
    got = ops.moe_grouped_gemm_mx4(xs, q, s, eids, combine_out=out32,
                                   combine_scale=gw, combine_gather=gtab)
    d2 = (got - want).abs().max().item()
    print(f"combine vs index_add: maxdiff {d2:.5f}  "
          f"(want absmax {want.abs().max().item():.3f}, "
          f"got absmax {got.abs().max().item():.3f})")


if __name__ == "__main__":
    main()
