"""Bisect the ragged ring-decode mismatch (GPU vs CPU ref)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from neuronx_distributed_inference_amd import ops  # noqa: E402


def main():
    torch.manual_seed(0)
    B, Hq, Hkv, W, D = 2, 4, 2, 32, 64
    dev = torch.device("cuda:0")
    win = W
    kc = torch.randn(B, Hkv, W, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn(B, Hkv, W, D, dtype=torch.bfloat16, device=dev)
    Sq = 4
    q = torch.randn(B, Hq, Sq, D, dtype=torch.bfloat16, device=dev)
    pos = torch.tensor([[45, 46, -1, -1], [38, -1, -1, -1]],
                       dtype=torch.int32, device=dev)
    out = ops.attention_decode(q, kc, vc, pos, sliding_window=win, ring=True)
    want = ops.attention_decode(q.float().cpu(), kc.float().cpu(),
                                vc.float().cpu(), pos.cpu(),
                                sliding_window=win, ring=True)
    d = (out.float().cpu() - want.float()).abs()
    for b in range(B):
        for h in range(Hq):
            for s in range(Sq):
                print(f"b{b} h{h} sq{s} pos={int(pos[b,s])}: maxdiff="
                      f"{d[b,h,s].max():.4f} |out|={out[b,h,s].float().abs().mean():.4f}"
                      f" |want|={want[b,h,s].abs().mean():.4f}", flush=True)
    # also: same rows WITHOUT pads (Sq matching valid count) as control
    for b, np_ in ((0, 2), (1, 1)):
        qs = q[b:b+1, :, :np_]
        ps = pos[b:b+1, :np_]
        o2 = ops.attention_decode(qs, kc[b:b+1], vc[b:b+1], ps,
                                  sliding_window=win, ring=True)
        w2 = ops.attention_decode(qs.float().cpu(), kc[b:b+1].float().cpu(),
                                  vc[b:b+1].float().cpu(), ps.cpu(),
                                  sliding_window=win, ring=True)
        print(f"control b{b} Sq={np_}: maxdiff="
              f"{(o2.float().cpu()-w2.float()).abs().max():.4f}", flush=True)


if __name__ == "__main__":
    main()
