"""KV-cache managers.

MI355X-native re-design of the reference's `modules/kvcache/kv_cache_manager.py`
(class at :106, shapes :152-236, get :344-367, update :369-586) — with one
genuine simplification: the reference had to alias cache tensors as graph
inputs<->outputs to keep updates on-device through the XLA trace
(`model_wrapper.py:1588-1626`); here the caches are persistent device tensors
updated in place by the HIP scatter kernel, which hipGraph capture records
directly. No aliasing machinery.

Layout: per layer K and V of shape (kv_cache_batch + pad, Hkv/rank, S_max, D)
— BHSD, row-contiguous in D so decode reads stream D-contiguous lines.
Sized against 288 GB HBM3E per GPU (capacity check in `estimate_bytes`).
"""

from __future__ import annotations

import logging
from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.parallel import state as ps

logger = logging.getLogger("nxdi_amd")


class KVCacheManager(nn.Module):
    """Contiguous (linear) per-sequence KV cache."""

    def __init__(
        self,
        num_layers: int,
        num_kv_heads: int,
        head_dim: int,
        max_len: int,
        batch_size: int,
        dtype: torch.dtype = torch.bfloat16,
        padding_rows: int = 0,
        sliding_window: Optional[int] = None,
        layer_cache_sizes: Optional[List[int]] = None,
        tp_degree: Optional[int] = None,
    ):
        super().__init__()
        self.num_layers = num_layers
        tp = tp_degree or ps.get_tensor_model_parallel_size()
        # GQA sharding: when Hkv < tp the heads are replicated so each rank
        # holds >=1 head (ref gqa.py REPLICATE_TO_TP_DEGREE).
        self.num_kv_heads_per_rank = max(num_kv_heads // tp, 1)
        self.head_dim = head_dim
        self.max_len = max_len
        self.batch_size = batch_size
        self.padding_rows = padding_rows
        self.dtype = dtype
        # sliding-window layers may use a smaller cache (ref gpt_oss mixed sizes)
        if layer_cache_sizes is None:
            size = min(sliding_window, max_len) if sliding_window else max_len
            layer_cache_sizes = [size] * num_layers
        self.layer_cache_sizes = layer_cache_sizes
        # ring layers: compact sliding-window cache (size < max_len) —
        # positions wrap modulo the layer size (ref gpt_oss per-layer KV
        # sizes)
        self.ring_layers = [sz < max_len for sz in layer_cache_sizes]

        self.k_caches = nn.ParameterList()
        self.v_caches = nn.ParameterList()
        for li in range(num_layers):
            shape = (
                batch_size + padding_rows,
                self.num_kv_heads_per_rank,
                self.layer_cache_sizes[li],
                head_dim,
            )
            self.k_caches.append(nn.Parameter(torch.zeros(shape, dtype=dtype), requires_grad=False))
            self.v_caches.append(nn.Parameter(torch.zeros(shape, dtype=dtype), requires_grad=False))

    # ------------------------------------------------------------------
    def estimate_bytes(self) -> int:
        return sum(k.numel() * k.element_size() * 2 for k in self.k_caches)

    def get_cache(self, layer_idx: int, bucket_len: Optional[int] = None):
        k = self.k_caches[layer_idx]
        v = self.v_caches[layer_idx]
        if bucket_len is not None and bucket_len < k.shape[2]:
            k = k[:, :, :bucket_len, :]
            v = v[:, :, :bucket_len, :]
        return k, v

    def update_cache(
        self,
        layer_idx: int,
        k_new: torch.Tensor,
        v_new: torch.Tensor,
        positions: torch.Tensor,
        seq_ids: Optional[torch.Tensor] = None,
    ) -> None:
        """In-place scatter of (B,Hkv,S_new,D) K/V at `positions` (B,S_new);
        `seq_ids` maps batch lines to cache rows (continuous batching)."""
        kc = self.k_caches[layer_idx].data
        vc = self.v_caches[layer_idx].data
        if self.ring_layers[layer_idx]:
            positions = self.wrap_positions(layer_idx, positions)
        ops.kv_cache_update(kc, vc, k_new, v_new, positions, seq_ids)

    def is_ring(self, layer_idx: int) -> bool:
        return self.ring_layers[layer_idx]

    def wrap_positions(self, layer_idx: int, positions: torch.Tensor) -> torch.Tensor:
        """Ring (compact SWA) write positions: slot = pos % size, and any
        position whose slot is ALSO claimed by a newer position in the
        SAME batch is dropped (-1). Without the drop, a prefill longer
        than the window issues two racing writes to one slot in a single
        scatter — the GPU writer's order is nondeterministic (caught by
        the mistral ring e2e: ~0.9 token agreement, serialized runs
        clean), while the CPU reference writes sequentially."""
        sz = self.layer_cache_sizes[layer_idx]
        pmax = positions.amax(dim=-1, keepdim=True)
        newest = positions > pmax - sz
        valid = (positions >= 0) & newest
        return torch.where(valid, positions % sz,
                           torch.full_like(positions, -1))

    def compact_rows(self, seq_ids: torch.Tensor, src: torch.Tensor,
                     dst: torch.Tensor):
        """Move cache rows src[b, j] -> dst[b, j] for every layer (tree
        speculation compaction: accepted tree nodes live at scattered
        rows L+node_idx and must become the contiguous committed prefix —
        ref medusa/tree KV gather-scatter, kv_cache_manager.py:265-321).
        Advanced indexing reads the full source before writing, so
        overlapping src/dst are safe."""
        if src.numel() == 0:
            return
        rows = seq_ids.long().clamp_min(0).view(-1, 1)
        src = src.long()
        dst = dst.long()
        for li in range(self.num_layers):
            kc = self.k_caches[li].data
            vc = self.v_caches[li].data
            kc[rows, :, dst] = kc[rows, :, src]
            vc[rows, :, dst] = vc[rows, :, src]

    def reset(self):
        for k in self.k_caches:
            k.data.zero_()
        for v in self.v_caches:
            v.data.zero_()

    def to_device(self, device):
        for p in self.parameters():
            p.data = p.data.to(device)
        return self
