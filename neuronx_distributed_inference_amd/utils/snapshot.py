"""Input/output snapshotting for debugging.

Analog of the reference's `utils/snapshot.py` (:58-460) +
`application_base.py:423-543`: capture the full device input set (and
optionally weights) of chosen submodels at chosen request indices to .pt
files, driven by env vars or explicit API.

    NXDI_SNAPSHOT_OUTPUT_PATH=/tmp/snap NXDI_SNAPSHOT_CAPTURE_AT_REQUESTS=0,3
"""

from __future__ import annotations

import os
from typing import List, Optional

import torch


class SnapshotHook:
    def __init__(self, output_path: str, capture_at_requests: List[int],
                 save_weights: bool = False):
        self.output_path = output_path
        self.capture_at = set(capture_at_requests)
        self.save_weights = save_weights
        self.request_idx = 0

    @classmethod
    def from_env(cls) -> Optional["SnapshotHook"]:
        path = os.environ.get("NXDI_SNAPSHOT_OUTPUT_PATH")
        if not path:
            return None
        at = os.environ.get("NXDI_SNAPSHOT_CAPTURE_AT_REQUESTS", "0")
        return cls(path, [int(x) for x in at.split(",") if x != ""],
                   save_weights=os.environ.get("NXDI_SNAPSHOT_SAVE_WEIGHTS", "0") == "1")

    def maybe_capture(self, tag: str, model, inputs: dict):
        if self.request_idx in self.capture_at:
            os.makedirs(self.output_path, exist_ok=True)
            payload = {
                k: (v.detach().cpu() if isinstance(v, torch.Tensor) else v)
                for k, v in inputs.items()
            }
            if self.save_weights:
                payload["_weights"] = {
                    k: v.detach().cpu() for k, v in model.state_dict().items()
                }
            torch.save(
                payload,
                os.path.join(self.output_path, f"{tag}_req{self.request_idx}.pt"),
            )
        self.request_idx += 1


def attach_snapshot_hook(causal_lm) -> Optional[SnapshotHook]:
    """Wrap CausalLMBase.forward with env-var-driven input capture."""
    hook = SnapshotHook.from_env()
    if hook is None:
        return None
    orig = causal_lm.forward

    def wrapped(input_ids, attention_mask=None, position_ids=None, seq_ids=None,
                sampling_params=None, **kw):
        tag = "context_encoding_model" if input_ids.shape[-1] > 1 else "token_generation_model"
        hook.maybe_capture(tag, causal_lm.model, {
            "input_ids": input_ids, "attention_mask": attention_mask,
            "position_ids": position_ids, "seq_ids": seq_ids,
            "sampling_params": sampling_params,
        })
        return orig(input_ids, attention_mask, position_ids, seq_ids,
                    sampling_params, **kw)

    causal_lm.forward = wrapped
    return hook


def reconstruct_kv_cache(shards, strategy: str = "auto"):
    """Reassemble per-rank KV cache shards into the full cache for
    debugging (ref utils/kv_cache_reconstruct_utils.py:57-218).

    shards: list over TP ranks of (L, B, Hkv_local, S, D)-shaped tensors
    (or a single layer's (B, Hkv_local, S, D)). Head-sharded caches concat
    along the head dim; replicated caches (Hkv < tp, each rank holds the
    same head) verify equality and return shard 0.
    """
    import torch

    first = shards[0]
    if all(torch.equal(s, first) for s in shards[1:]):
        return first.clone()   # replicated heads
    return torch.cat(shards, dim=-3)
