"""Paged (block) KV cache for paged attention / prefix caching.

Analog of the reference's `modules/kvcache/block_kv_cache_manager.py`
(layout :40-91, gather :150-182, scatter :268-375). Layout here is
(num_blocks, block_size, Hkv/rank, D) so a block's tokens are contiguous —
the decode kernel walks the block table per sequence.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.parallel import state as ps


class BlockKVCacheManager(nn.Module):
    is_paged = True

    def __init__(
        self,
        num_layers: int,
        num_kv_heads: int,
        head_dim: int,
        num_blocks: int,
        block_size: int = 128,
        dtype: torch.dtype = torch.bfloat16,
        tp_degree: Optional[int] = None,
    ):
        super().__init__()
        tp = tp_degree or ps.get_tensor_model_parallel_size()
        self.num_kv_heads_per_rank = max(num_kv_heads // tp, 1)
        self.head_dim = head_dim
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.num_layers = num_layers
        self.dtype = dtype
        shape = (num_blocks, block_size, self.num_kv_heads_per_rank, head_dim)
        self.k_caches = nn.ParameterList(
            nn.Parameter(torch.zeros(shape, dtype=dtype), requires_grad=False)
            for _ in range(num_layers)
        )
        self.v_caches = nn.ParameterList(
            nn.Parameter(torch.zeros(shape, dtype=dtype), requires_grad=False)
            for _ in range(num_layers)
        )

    def estimate_bytes(self) -> int:
        return sum(k.numel() * k.element_size() * 2 for k in self.k_caches)

    # per-forward step context set by DecoderModel (block table + slots)
    def set_step_context(self, block_table: torch.Tensor, slot_mapping: torch.Tensor):
        self._block_table = block_table
        self._slot_mapping = slot_mapping

    @property
    def block_table(self):
        return self._block_table

    def get_cache(self, layer_idx: int, bucket_len=None):
        return self.k_caches[layer_idx], self.v_caches[layer_idx]

    def update_cache(self, layer_idx: int, k_new, v_new, positions, seq_ids=None):
        """Same contract as the linear manager: k_new/v_new (B,Hkv,S,D);
        routes through the slot mapping installed by set_step_context."""
        B, Hkv, S, D = k_new.shape
        k_flat = k_new.transpose(1, 2).reshape(B * S, Hkv, D)
        v_flat = v_new.transpose(1, 2).reshape(B * S, Hkv, D)
        self.update(layer_idx, k_flat, v_flat, self._slot_mapping)

    def update(self, layer_idx: int, k_new: torch.Tensor, v_new: torch.Tensor,
               slot_mapping: torch.Tensor) -> None:
        """k_new/v_new: (N_tokens, Hkv, D); slot_mapping: (N_tokens,) flat
        slot = block_id * block_size + offset, or -1 to drop."""
        ops.block_kv_cache_update(
            self.k_caches[layer_idx].data, self.v_caches[layer_idx].data,
            k_new, v_new, slot_mapping,
        )

    def gather(self, layer_idx: int, block_table: torch.Tensor, max_len: int
               ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Contiguous (B,Hkv,S,D) K/V view for prefix-prefill paths."""
        from neuronx_distributed_inference_amd.ops import ref
        return ref.paged_gather_kv(
            self.k_caches[layer_idx].data, self.v_caches[layer_idx].data,
            block_table, max_len,
        )

    def reset(self):
        for k in self.k_caches:
            k.data.zero_()
        for v in self.v_caches:
            v.data.zero_()


def generate_slot_mapping(block_table: torch.Tensor, positions: torch.Tensor,
                          block_size: int) -> torch.Tensor:
    """In-graph slot-mapping computation (ref
    `block_kv_cache_manager.py:376-430`): positions (B,S) + per-sequence
    block table (B, max_blocks) -> flat slots (B*S,). Negative positions
    (padding) map to slot -1 (dropped by the scatter kernel)."""
    valid = positions >= 0
    pos = positions.clamp(min=0)
    blk = (pos // block_size).clamp(max=block_table.shape[1] - 1)
    off = pos % block_size
    blocks = torch.gather(block_table.long(), 1, blk.long())
    slots = blocks * block_size + off
    slots = torch.where(valid & (blocks >= 0), slots, torch.full_like(slots, -1))
    return slots.flatten()
