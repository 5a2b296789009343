"""The attention module every model subclasses.

MI355X-native analog of the reference's `NeuronAttentionBase`
(`modules/attention/attention_base.py:134-2488`): one nn.Module serving both
phases — context encoding (flash prefill kernel, K1-analog) and token
generation (decode kernel against the KV cache, K2-analog) — selected by the
active-token count. KV-cache writes go through the in-place HIP scatter
before attention, so the captured decode graph reads what it just wrote with
no aliasing machinery.

Design difference vs the reference: no trace boundary means no dual
CTE/TKG projection sets unless CP != DP meshes demand it; the same
Column/Row-parallel projections serve both phases.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.modules.attention import utils as attn_utils
from neuronx_distributed_inference_amd.modules.attention.gqa import (
    GroupQueryAttentionO,
    GroupQueryAttentionQKV,
)
from neuronx_distributed_inference_amd.modules.rope import RotaryEmbedding
from neuronx_distributed_inference_amd.parallel import state as ps

PHASE_CTE = "context_encoding"
PHASE_TKG = "token_generation"


class AttentionBase(nn.Module):
    def __init__(self, config, layer_idx: int = 0, rotary: Optional[RotaryEmbedding] = None):
        super().__init__()
        ec = config.neuron_config
        self.config = config
        self.layer_idx = layer_idx
        self.hidden_size = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.num_kv_heads = getattr(config, "num_key_value_heads", self.num_heads)
        self.head_dim = getattr(config, "head_dim", None) or self.hidden_size // self.num_heads
        self.tp_degree = ec.tp_degree
        self.torch_dtype = ec.torch_dtype
        self.sliding_window = getattr(config, "sliding_window", None) or 0
        self.attn_bias = bool(getattr(config, "attention_bias", False))
        self.clip_qkv = getattr(config, "clip_qkv", None)

        self.qkv_proj = GroupQueryAttentionQKV(
            self.hidden_size, self.num_heads, self.num_kv_heads, self.head_dim,
            self.tp_degree, self.torch_dtype, bias=self.attn_bias,
            fused=ec.fused_qkv,
        )
        self.o_proj = GroupQueryAttentionO(
            self.hidden_size, self.num_heads, self.head_dim, self.tp_degree,
            self.torch_dtype, bias=bool(getattr(config, "attention_o_bias", False)),
        )
        self.num_heads_per_rank = self.qkv_proj.num_heads_per_rank
        self.num_kv_heads_per_rank = self.qkv_proj.num_kv_heads_per_rank

        # Context parallelism (ref §2.8 CP + dual projection sets,
        # attention_base.py:326-421): prefill attention runs in tp/cp-sized
        # TP subgroups over a SECOND projection set sharded over the prefill
        # group; each CP rank computes its S/cp query slice against the full
        # K/V (activations are TP-replicated, so no KV all-gather is needed)
        # and the outputs are all-gathered along S over the CP group.
        # Attention data parallelism for decode (ref §2.8 DP +
        # DataParallelKVCacheManager): TKG attention + KV cache sharded by
        # batch across DP subgroups, each running reduced-TP (tp/dp)
        # projections; out-of-shard rows write to nowhere (seq_id -1) and the
        # outputs are all-gathered along batch over the DP group.
        self.dp_degree = ec.attention_dp_degree
        if self.dp_degree > 1:
            tpd = self.tp_degree // self.dp_degree
            self.tkg_qkv_proj = GroupQueryAttentionQKV(
                self.hidden_size, self.num_heads, self.num_kv_heads, self.head_dim,
                tpd, self.torch_dtype, bias=self.attn_bias, fused=False,
                group=ps.get_attention_dp_decode_tp_group,
            )
            self.tkg_o_proj = GroupQueryAttentionO(
                self.hidden_size, self.num_heads, self.head_dim, tpd,
                self.torch_dtype, bias=False,
                group=ps.get_attention_dp_decode_tp_group,
            )

        self.cp_degree = ec.cp_degree
        if self.cp_degree > 1:
            tpc = self.tp_degree // self.cp_degree
            self.cte_qkv_proj = GroupQueryAttentionQKV(
                self.hidden_size, self.num_heads, self.num_kv_heads, self.head_dim,
                tpc, self.torch_dtype, bias=self.attn_bias, fused=False,
                group=ps.get_prefill_tp_group,
            )
            self.cte_o_proj = GroupQueryAttentionO(
                self.hidden_size, self.num_heads, self.head_dim, tpc,
                self.torch_dtype, bias=False, group=ps.get_prefill_tp_group,
            )
        self.scale = 1.0 / math.sqrt(self.head_dim)
        self.rotary = rotary
        # optional per-head q/k norms (Qwen3 PRE_ROPE qk-norm); models set these
        self.q_norm: Optional[nn.Module] = None
        self.k_norm: Optional[nn.Module] = None
        self.sinks: Optional[nn.Parameter] = None  # GPT-OSS learned sinks
        # Llama4 hooks (ref llama4 text attention): NoPE layers, complex-pair
        # (interleaved) rope, weightless L2 qk-norm AFTER rope, and the
        # log-temperature query scaling of NoPE layers
        self.use_rope = True
        self.rope_interleaved = False
        # flash decoding: decode KV S-sharded over the
        # kv-shard group; requires no sinks and plain causal masking
        self.kv_shard_degree = (
            ps.get_kv_shard_degree()
            if getattr(ec, "flash_decoding_enabled", False) else 1
        )
        self.qk_norm_post_rope: Optional[nn.Module] = None
        self.q_scale_fn = None  # fn(position_ids (B,S)) -> (B,S) scale

    # ------------------------------------------------------------------
    def prep_qkv(
        self, hidden: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
        sequence_parallel: bool = False,
        position_ids: Optional[torch.Tensor] = None,
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        """Projections + BSHD->BHSD + optional qk-norm + RoPE
        (ref prep_qkv_tensors attention_base.py:489-563)."""
        q, k, v = self.qkv_proj(hidden, sequence_parallel=sequence_parallel)
        B, S = q.shape[0], q.shape[1]
        if self.clip_qkv is not None:  # DBRX clamp (ref modeling_dbrx.py)
            q = q.clamp(-self.clip_qkv, self.clip_qkv)
            k = k.clamp(-self.clip_qkv, self.clip_qkv)
            v = v.clamp(-self.clip_qkv, self.clip_qkv)
        q = q.view(B, S, self.num_heads_per_rank, self.head_dim)
        k = k.view(B, S, self.num_kv_heads_per_rank, self.head_dim)
        v = v.view(B, S, self.num_kv_heads_per_rank, self.head_dim)
        if self.q_norm is not None:
            q = self.q_norm(q)
        if self.k_norm is not None:
            k = self.k_norm(k)
        q = q.transpose(1, 2)  # BHSD
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        if cos is not None and self.use_rope:
            if self.rope_interleaved:
                from neuronx_distributed_inference_amd.modules.rope import (
                    rope_apply_interleaved,
                )

                q, k = rope_apply_interleaved(q, k, cos, sin)
            else:
                q, k = ops.rope_apply(q, k, cos, sin)
        if self.qk_norm_post_rope is not None:
            q = self.qk_norm_post_rope(q)
            k = self.qk_norm_post_rope(k)
        if self.q_scale_fn is not None and position_ids is not None:
            q = (q * self.q_scale_fn(position_ids).unsqueeze(1).unsqueeze(-1)).to(q.dtype)
        return q, k, v

    # ------------------------------------------------------------------
    def forward(
        self,
        hidden: torch.Tensor,
        cos: torch.Tensor,
        sin: torch.Tensor,
        kv_mgr,
        position_ids: torch.Tensor,
        phase: str,
        attention_mask: Optional[torch.Tensor] = None,
        seq_ids: Optional[torch.Tensor] = None,
        active_mask: Optional[torch.Tensor] = None,
        sequence_parallel: bool = False,
    ) -> torch.Tensor:
        B, S, _ = hidden.shape
        if sequence_parallel:
            S = S * self.tp_degree  # hidden is S-sharded; attention runs full-S
        if not self.use_rope:  # NoPE layer (Llama4 global-attention layers)
            cos, sin = None, None
        # a 4-D mask in TKG is an EXPLICIT attention mask (tree-structured
        # speculation); 2-D masks stay padding masks
        if (active_mask is None and attention_mask is not None
                and attention_mask.dim() == 4 and phase == PHASE_TKG):
            active_mask = attention_mask

        # Fused TKG front end (K2-analog): one GEMM -> one kernel doing
        # RoPE + cache write + q relayout. Skipped when per-head qk-norms or
        # clip_qkv need the unfused path.
        if (
            phase == PHASE_TKG
            and self.qkv_proj.fused
            and self.q_norm is None
            and self.clip_qkv is None
            and cos is not None
            and not getattr(kv_mgr, "is_paged", False)
            and not self.rope_interleaved
            and self.qk_norm_post_rope is None
            and self.q_scale_fn is None
        ):
            y = self.qkv_proj.qkv(hidden)
            kc, vc = kv_mgr.get_cache(self.layer_idx)
            import neuronx_distributed_inference_amd.ops as _ops

            wpos = position_ids
            if self.kv_shard_degree > 1:
                from neuronx_distributed_inference_amd.modules.attention import (
                    flash_decoding as _fdm,
                )

                wpos = _fdm.shard_write_positions(
                    position_ids, self.kv_shard_degree, ps.get_kv_shard_rank())
            elif getattr(kv_mgr, "ring_layers", None) and kv_mgr.is_ring(self.layer_idx):
                wpos = kv_mgr.wrap_positions(self.layer_idx, position_ids)
            q = _ops.qkv_rope_update(
                y, cos, sin, kc.data if hasattr(kc, "data") else kc,
                vc.data if hasattr(vc, "data") else vc,
                wpos, seq_ids,
                self.num_heads_per_rank, self.num_kv_heads_per_rank,
            )
            out = self._decode(q, kv_mgr, position_ids, seq_ids, active_mask)
            out = out.transpose(1, 2).reshape(B, S, -1)
            return self.o_proj(out)

        # Fused CTE front end: one GEMM -> qkv_rope_update (RoPE + cache
        # scatter + q relayout) -> flash attention reading K/V straight from
        # the cache rows just written. Saves 3 transpose-copies + rope + kv
        # kernels per layer vs the unfused path. Right-padded batches are
        # handled without a mask: pad positions (-1) were skipped by the
        # scatter, so causal-by-row-index only ever sees the real prefix.
        if (
            phase == PHASE_CTE
            and self.qkv_proj.fused
            and self.q_norm is None
            and self.clip_qkv is None
            and cos is not None
            and seq_ids is not None
            and not self.rope_interleaved
            and self.qk_norm_post_rope is None
            and self.q_scale_fn is None
            and not sequence_parallel
            and self.cp_degree == 1
            and self.dp_degree == 1
            and not getattr(kv_mgr, "is_paged", False)
            and self.kv_shard_degree == 1
            and not (getattr(kv_mgr, "ring_layers", None)
                     and kv_mgr.is_ring(self.layer_idx))
            and (not hidden.is_cuda or self.head_dim in (64, 128))
        ):
            kc, vc = kv_mgr.get_cache(self.layer_idx)
            kc = kc.data if hasattr(kc, "data") else kc
            vc = vc.data if hasattr(vc, "data") else vc
            if kc.dtype == hidden.dtype:  # fp8 cache: unfused CTE path below
                import neuronx_distributed_inference_amd.ops as _ops

                y = self.qkv_proj.qkv(hidden)
                q = _ops.qkv_rope_update(
                    y, cos, sin, kc, vc, position_ids, seq_ids,
                    self.num_heads_per_rank, self.num_kv_heads_per_rank,
                )
                out = _ops.attention_prefill_cached(
                    q, kc, vc, seq_ids, scale=self.scale, sinks=self.sinks,
                    sliding_window=self.sliding_window,
                )
                out = out.transpose(1, 2).reshape(B, S, -1)
                return self.o_proj(out)

        if (
            phase == PHASE_CTE and self.cp_degree > 1
            and S % self.cp_degree == 0 and not sequence_parallel
        ):
            return self._cp_prefill(hidden, cos, sin, kv_mgr, position_ids,
                                    attention_mask, seq_ids)
        if phase == PHASE_TKG and self.dp_degree > 1 and B % self.dp_degree == 0:
            return self._dp_decode(hidden, cos, sin, kv_mgr, position_ids, seq_ids)

        q, k, v = self.prep_qkv(hidden, cos, sin, sequence_parallel=sequence_parallel,
                                position_ids=position_ids)

        # write this step's K/V into the cache first (in-place scatter)
        if self.dp_degree > 1 and phase == PHASE_CTE:
            # attention-DP: the cache lives in DECODE layout (tp/dp sharding,
            # B/dp local rows) — write it from the TKG projection set
            self._dp_write_cte_cache(hidden, cos, sin, kv_mgr, position_ids, seq_ids)
        else:
            wpos = position_ids
            if self.kv_shard_degree > 1:
                from neuronx_distributed_inference_amd.modules.attention import (
                    flash_decoding as _fdm,
                )

                wpos = _fdm.shard_write_positions(
                    position_ids, self.kv_shard_degree, ps.get_kv_shard_rank())
            kv_mgr.update_cache(self.layer_idx, k, v, wpos, seq_ids)

        if phase == PHASE_CTE:
            out = self._prefill(q, k, v, attention_mask)
        else:
            out = self._decode(q, kv_mgr, position_ids, seq_ids, active_mask)

        out = out.transpose(1, 2).reshape(B, S, -1)  # BHSD -> BS(H*D)
        return self.o_proj(out, sequence_parallel=sequence_parallel)

    # ------------------------------------------------------------------
    def _cp_prefill(self, hidden, cos, sin, kv_mgr, position_ids,
                    attention_mask, seq_ids) -> torch.Tensor:
        from neuronx_distributed_inference_amd.parallel import mappings

        B, S, _ = hidden.shape
        cp = self.cp_degree
        cp_rank = ps.get_context_parallel_rank()

        # decode-sharded K/V into the cache (standard projections; the q
        # output of this pass is discarded — the ref instead reorders kv
        # heads between meshes, attention/utils.py:455-623). With
        # flash decoding the decode cache is S-SHARDED: map positions to
        # this rank's span exactly like the non-CP write path (the
        # CP+flash-decode seam bug the combined gloo test caught).
        _, k_full, v_full = self.prep_qkv(hidden, cos, sin, position_ids=position_ids)
        wpos = position_ids
        if self.kv_shard_degree > 1:
            from neuronx_distributed_inference_amd.modules.attention import (
                flash_decoding as _fdm,
            )

            wpos = _fdm.shard_write_positions(
                position_ids, self.kv_shard_degree, ps.get_kv_shard_rank())
        kv_mgr.update_cache(self.layer_idx, k_full, v_full, wpos, seq_ids)

        # CTE-sharded q/k/v over the prefill-TP subgroup
        qc, kc, vc = self.cte_qkv_proj(hidden)
        hc = self.cte_qkv_proj.num_heads_per_rank
        hkvc = self.cte_qkv_proj.num_kv_heads_per_rank
        qc = qc.view(B, S, hc, self.head_dim).transpose(1, 2)
        kc = kc.view(B, S, hkvc, self.head_dim).transpose(1, 2)
        vc = vc.view(B, S, hkvc, self.head_dim).transpose(1, 2)
        if cos is not None:
            from neuronx_distributed_inference_amd import ops as _ops

            qc, kc = _ops.rope_apply(qc, kc, cos, sin)

        # this rank's S/cp query rows vs the full K/V, offset-causal mask.
        # ZIGZAG assignment when S % 2cp == 0 (rank r takes chunks r and
        # 2cp-1-r of 2cp): causal work per rank is constant, where
        # contiguous slices leave the first rank nearly idle (the ref's
        # strided CP query assignment, attention/utils.py CP paths).
        ssl = S // cp
        zig = S % (2 * cp) == 0
        dev = hidden.device
        if zig:
            half = S // (2 * cp)
            c0a, c0b = cp_rank * half, (2 * cp - 1 - cp_rank) * half
            q_slice = torch.cat(
                [qc[:, :, c0a:c0a + half], qc[:, :, c0b:c0b + half]], dim=2
            ).contiguous()
            qpos = torch.cat([c0a + torch.arange(half, device=dev),
                              c0b + torch.arange(half, device=dev)])
            qpos = qpos.view(1, 1, ssl, 1)
        else:
            c0 = cp_rank * ssl
            q_slice = qc[:, :, c0:c0 + ssl].contiguous()
            qpos = (c0 + torch.arange(ssl, device=dev)).view(1, 1, ssl, 1)
        ar = torch.arange(S, device=dev).view(1, 1, 1, S)
        mask = ar <= qpos
        if attention_mask is not None:
            mask = mask & attention_mask.view(B, 1, 1, S).bool()
        if self.sliding_window:
            mask = mask & (ar > qpos - self.sliding_window)
        out = ops.attention_prefill(q_slice, kc, vc, mask=mask, scale=self.scale,
                                    sinks=self.sinks, causal=False)
        out = out.transpose(1, 2).reshape(B, ssl, -1)
        out = self.cte_o_proj(out)                   # reduce over prefill group
        out = mappings.gather_along_dim(out, dim=1,
                                        group=ps.get_context_parallel_group())
        if zig:
            # gathered order is [r0:(chunk 0, chunk 2cp-1), r1:(1, 2cp-2)...];
            # permute back to natural chunk order
            half = S // (2 * cp)
            chunks = out.view(B, 2 * cp, half, -1)
            order = [min(c, 2 * cp - 1 - c) * 2 + (0 if c < cp else 1)
                     for c in range(2 * cp)]
            idx = torch.tensor(order, device=out.device)
            out = chunks.index_select(1, idx).reshape(B, S, -1)
        return out

    def _dp_local_rows(self, seq_ids, B, device):
        dp_rank = ps.get_attention_dp_rank()
        per = B // self.dp_degree
        if seq_ids is None:
            seq_ids = torch.arange(B, device=device)
        local = seq_ids - dp_rank * per
        return torch.where((local >= 0) & (local < per), local,
                           torch.full_like(local, -1))

    def _dp_write_cte_cache(self, hidden, cos, sin, kv_mgr, position_ids, seq_ids):
        B, S, _ = hidden.shape
        _, k, v = self.tkg_qkv_proj(hidden)
        hkvc = self.tkg_qkv_proj.num_kv_heads_per_rank
        k = k.view(B, S, hkvc, self.head_dim).transpose(1, 2)
        v = v.view(B, S, hkvc, self.head_dim).transpose(1, 2)
        if cos is not None:
            # rope applies to k only; reuse the paired op with a dummy q
            dummy_q = torch.empty_like(k)
            _, k = ops.rope_apply(dummy_q, k, cos, sin)
        local_rows = self._dp_local_rows(seq_ids, B, hidden.device)
        kv_mgr.update_cache(self.layer_idx, k, v, position_ids, local_rows)

    def _dp_decode(self, hidden, cos, sin, kv_mgr, position_ids, seq_ids):
        from neuronx_distributed_inference_amd.parallel import mappings

        B, S, _ = hidden.shape
        dp = self.dp_degree
        dp_rank = ps.get_attention_dp_rank()
        per = B // dp
        if seq_ids is None:
            seq_ids = torch.arange(B, device=hidden.device)

        q, k, v = self.tkg_qkv_proj(hidden)
        hc = self.tkg_qkv_proj.num_heads_per_rank
        hkvc = self.tkg_qkv_proj.num_kv_heads_per_rank
        q = q.view(B, S, hc, self.head_dim).transpose(1, 2)
        k = k.view(B, S, hkvc, self.head_dim).transpose(1, 2)
        v = v.view(B, S, hkvc, self.head_dim).transpose(1, 2)
        if cos is not None:
            q, k = ops.rope_apply(q, k, cos, sin)

        # cache rows are LOCAL to this DP shard (kv_cache_batch_size = B/dp);
        # out-of-shard rows map to -1 and the scatter kernel skips them
        local_rows = self._dp_local_rows(seq_ids, B, hidden.device)
        kv_mgr.update_cache(self.layer_idx, k, v, position_ids, local_rows)

        lo = dp_rank * per
        q_shard = q[lo:lo + per].contiguous()
        k_cache, v_cache = kv_mgr.get_cache(self.layer_idx)
        out = ops.attention_decode(
            q_shard, k_cache, v_cache, position_ids[lo:lo + per],
            seq_ids=local_rows[lo:lo + per], scale=self.scale, sinks=self.sinks,
            sliding_window=self.sliding_window,
        )
        out = out.transpose(1, 2).reshape(per, S, -1)
        out = self.tkg_o_proj(out)                  # reduce over decode-TP
        # all-gather the batch shards over the DP group (ref AG_AR option)
        return mappings.gather_along_dim(out, dim=0, group=ps.get_attention_dp_group())

    def _prefill(self, q, k, v, attention_mask) -> torch.Tensor:
        """Causal flash attention over the fresh K/V (the cache holds the
        same values; using the projections directly avoids a re-read).

        attention_mask is None for dense (un-padded) batches — the caller
        drops an all-ones mask on the host so no per-layer sync happens.
        """
        mask = None
        if attention_mask is not None:
            mask = attn_utils.padded_causal_mask(attention_mask, q.shape[2])
            if self.sliding_window > 0:
                mask = mask & attn_utils.sliding_window_mask(
                    q.shape[2], k.shape[2], self.sliding_window, device=q.device
                )
            elif self.sliding_window < 0:  # chunked attention (Llama4)
                mask = mask & attn_utils.chunked_attention_mask(
                    q.shape[2], k.shape[2], -self.sliding_window, device=q.device
                )
            return ops.attention_prefill(
                q, k, v, mask=mask, scale=self.scale, sinks=self.sinks, causal=False
            )
        return ops.attention_prefill(
            q, k, v, scale=self.scale, sinks=self.sinks, causal=True,
            sliding_window=self.sliding_window,
        )

    def _decode(self, q, kv_mgr, position_ids, seq_ids, active_mask) -> torch.Tensor:
        k_cache, v_cache = kv_mgr.get_cache(self.layer_idx)
        if active_mask is not None and active_mask.dim() == 4:
            # EXPLICIT-mask decode (tree-structured speculation: a node
            # attends the prior plus its ancestor chain only — position
            # causality is wrong for sibling nodes; ref tree TKG
            # model_base.py:2094 + token_tree masks). Runs masked
            # attention against the cache prefix the mask covers.
            kc = k_cache.data if hasattr(k_cache, "data") else k_cache
            vc = v_cache.data if hasattr(v_cache, "data") else v_cache
            B = q.shape[0]
            rows = (seq_ids.long().clamp_min(0) if seq_ids is not None
                    else torch.arange(B, device=q.device))
            Sm = active_mask.shape[-1]
            k = kc[rows][:, :, :Sm].to(q.dtype)
            v = vc[rows][:, :, :Sm].to(q.dtype)
            return ops.attention_prefill(q, k, v, mask=active_mask,
                                         scale=self.scale, sinks=self.sinks,
                                         causal=False)
        # per-query positions (B,Sq): exact for ragged chunks (-1 pads)
        if getattr(kv_mgr, "is_paged", False):
            return ops.attention_decode_paged(
                q, k_cache.data, v_cache.data, position_ids, kv_mgr.block_table,
                scale=self.scale, sliding_window=self.sliding_window,
            )
        if self.kv_shard_degree > 1 and self.sinks is None and self.sliding_window == 0:
            from neuronx_distributed_inference_amd.modules.attention import (
                flash_decoding as _fdm,
            )

            B, _, Sq, _ = q.shape
            pos = position_ids if position_ids.dim() == 2 else (
                position_ids.view(B, 1) - Sq + torch.arange(
                    Sq, device=q.device).view(1, Sq))
            return _fdm.flash_decoding_attention(
                q, k_cache, v_cache, pos, seq_ids, self.scale)
        ring = bool(getattr(kv_mgr, "ring_layers", None)) and kv_mgr.is_ring(self.layer_idx)
        return ops.attention_decode(
            q, k_cache, v_cache, position_ids, seq_ids=seq_ids, scale=self.scale,
            sinks=self.sinks, sliding_window=self.sliding_window, ring=ring,
        )
