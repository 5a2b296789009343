import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from neuronx_distributed_inference_amd import ops  # noqa: E402

dev = "cuda:0"
torch.manual_seed(0)
T, E, k = 3, 8, 4
logits = (torch.randn(T, E) * 3).to(dev)
print("logits:", logits.cpu().numpy().round(2))
for sig, pre, norm, name in ((0, 0, 0, "post"), (0, 0, 1, "postnorm"),
                             (0, 1, 0, "pre"), (1, 0, 0, "sigmoid")):
    v, i = ops.router_topk(logits.float(), k, bool(sig), bool(pre), bool(norm))
    # torch reference
    lc = logits.float().cpu()
    if sig:
        tv, ti = lc.topk(k, -1)
        tv = torch.sigmoid(tv)
    elif pre:
        p = torch.softmax(lc, -1)
        tv, ti = p.topk(k, -1)
    else:
        tv, ti = lc.topk(k, -1)
        tv = torch.softmax(tv, -1)
    if norm:
        tv = tv / tv.sum(-1, keepdim=True)
    d = (v.cpu() - tv).abs().max().item()
    di = int((i.cpu() != ti).sum())
    print(f"{name:9s} maxdiff {d:.6f} idx-mismatch {di}")
    if d > 1e-4:
        print("  got ", v.cpu().numpy().round(4))
        print("  want", tv.numpy().round(4))
