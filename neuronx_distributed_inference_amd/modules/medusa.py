"""Medusa speculation.

Analog of the reference's Medusa support (ResBlock + N
column-parallel head MLPs `modeling_llama.py:1059-1187`, `_medusa_forward`
`model_base.py:393-508`, HF-side `_medusa_assisted_decoding`
`hf_adapter.py:799`). This is the linear-chain variant: each head predicts
the token i+1 positions ahead from the last hidden state; the chain is
verified with the same multi-token speculation forward the two-model path
uses, so acceptance semantics are shared.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd.parallel.layers import ColumnParallelLinear
from neuronx_distributed_inference_amd.parallel.ops import distributed_argmax
from neuronx_distributed_inference_amd.utils.hf_adapter import GenerationConfig


class ResBlock(nn.Module):
    """x + silu(Wx) (ref ResBlock, modeling_llama.py:1059)."""

    def __init__(self, hidden_size: int, dtype: torch.dtype):
        super().__init__()
        self.linear = nn.Linear(hidden_size, hidden_size, dtype=dtype)

    def forward(self, x):
        return x + torch.nn.functional.silu(self.linear(x))


class MedusaHeads(nn.Module):
    def __init__(self, hidden_size: int, vocab_size: int, num_heads: int,
                 tp_degree: int = 1, dtype: torch.dtype = torch.float32):
        super().__init__()
        self.blocks = nn.ModuleList(
            ResBlock(hidden_size, dtype) for _ in range(num_heads)
        )
        self.heads = nn.ModuleList(
            ColumnParallelLinear(hidden_size, vocab_size, bias=False, dtype=dtype,
                                 tp_degree=tp_degree)
            for _ in range(num_heads)
        )
        self.num_heads = num_heads

    @torch.inference_mode()
    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        """hidden (B, H) -> greedy head predictions (B, num_heads)."""
        preds = []
        for blk, head in zip(self.blocks, self.heads):
            logits = head(blk(hidden))
            preds.append(distributed_argmax(logits, dim=-1).view(-1, 1))
        return torch.cat(preds, dim=1)


class MedusaGenerationAdapter:
    """Chain-medusa assisted decoding: heads draft, target verifies in one
    multi-token forward, longest matching prefix + bonus accepted."""

    def __init__(self, model, medusa_heads: MedusaHeads):
        self.model = model
        self.heads = medusa_heads
        self.k = medusa_heads.num_heads

    @torch.inference_mode()
    def generate(self, input_ids, attention_mask=None, generation_config=None, **kw):
        gc = generation_config or GenerationConfig(**{
            k: v for k, v in kw.items() if k in GenerationConfig.__dataclass_fields__})
        B, S0 = input_ids.shape
        device = input_ids.device
        if attention_mask is None:
            attention_mask = torch.ones_like(input_ids)
        max_len = gc.max_length or min(self.model.ec.max_length, S0 + gc.max_new_tokens)
        max_len = min(max_len, self.model.ec.max_length - self.k - 2)
        k = self.k

        self.model.reset()
        seq_ids = torch.arange(B, device=device)
        cum = attention_mask.long().cumsum(-1) - 1
        position_ids = torch.where(attention_mask.bool(), cum, torch.full_like(cum, -1))
        t_out = self.model(input_ids=input_ids, attention_mask=attention_mask,
                           position_ids=position_ids, seq_ids=seq_ids)
        lengths = attention_mask.sum(-1).long()
        next_tok = t_out.tokens.view(B)
        h_last = self.model.model.last_hidden[:, -1, :]   # (B, H)
        out_rows = [input_ids[i][attention_mask[i].bool()].tolist() for i in range(B)]
        for i in range(B):
            out_rows[i].append(int(next_tok[i]))
        finished = torch.zeros(B, dtype=torch.bool, device=device)

        while True:
            if gc.eos_token_id is not None:
                finished |= next_tok == gc.eos_token_id
            if bool(finished.all()) or max(len(r) for r in out_rows) >= max_len:
                break
            cand = self.heads(h_last.to(self.heads.blocks[0].linear.weight.dtype))  # (B, k)
            verify_in = torch.cat([next_tok.view(B, 1), cand], dim=1)
            verify_pos = lengths.view(B, 1) + torch.arange(k + 1, device=device).view(1, -1)
            v_out = self.model(input_ids=verify_in, position_ids=verify_pos, seq_ids=seq_ids)
            target_toks = v_out.tokens.view(B, k + 1)
            hidden_all = self.model.model.last_hidden       # (B, k+1, H)

            match = (cand == target_toks[:, :k])
            n_accept = match.cumprod(dim=1).sum(dim=1)
            for i in range(B):
                if finished[i]:
                    continue
                m = int(n_accept[i])
                row = out_rows[i]
                for tok in cand[i, :m].tolist() + [int(target_toks[i, m])]:
                    if len(row) >= max_len:
                        break
                    row.append(tok)
                    if gc.eos_token_id is not None and tok == gc.eos_token_id:
                        finished[i] = True
                        break
            h_last = hidden_all.gather(
                1, n_accept.view(B, 1, 1).expand(-1, 1, hidden_all.shape[-1])
            ).squeeze(1)
            lengths = lengths + n_accept + 1
            next_tok = torch.tensor([r[-1] for r in out_rows], device=device, dtype=torch.long)

        longest = max(len(r) for r in out_rows)
        out = torch.full((B, longest), gc.pad_token_id, dtype=torch.long, device=device)
        for i, r in enumerate(out_rows):
            out[i, : len(r)] = torch.tensor(r, device=device)
        return out


class MedusaTreeAdapter:
    """Medusa TREE decoding (ref medusa tree attention: medusa_mask /
    accepted-index inputs, modeling_llama.py:1059-1187 + KV compaction
    kv_cache_manager.py:265-321): each head proposes top-k candidates for
    its position; the static tree's depth-d nodes draw from head d-1's
    candidate list by child rank; the target verifies ALL nodes in one
    masked pass (the same tree-verify machinery as EAGLE tree TKG) and
    acceptance walks the deepest matching path."""

    def __init__(self, model, medusa_heads: MedusaHeads, tree):
        from neuronx_distributed_inference_amd.modules.token_tree import (
            ancestor_matrix,
        )

        self.model = model
        self.heads = medusa_heads
        self.tree = tree
        assert len(tree.branching) <= medusa_heads.num_heads, \
            "tree depth exceeds medusa head count"
        self.anc = ancestor_matrix(tree)
        # child rank within its parent (which top-k slot feeds the node)
        self.child_rank = [0] * tree.num_nodes
        seen = {}
        for n in range(1, tree.num_nodes):
            p = tree.parent[n]
            self.child_rank[n] = seen.get(p, 0)
            seen[p] = self.child_rank[n] + 1

    def _head_topk(self, hidden: torch.Tensor) -> torch.Tensor:
        """(B, depth, kmax) per-head top-k candidate tokens."""
        from neuronx_distributed_inference_amd.parallel.ops import distributed_topk

        kmax = max(self.tree.branching)
        outs = []
        for d in range(len(self.tree.branching)):
            blk, head = self.heads.blocks[d], self.heads.heads[d]
            logits = head(blk(hidden))
            _, ids = distributed_topk(logits.unsqueeze(1), kmax, dim=-1)
            outs.append(ids.view(-1, 1, kmax))
        return torch.cat(outs, dim=1)

    @torch.inference_mode()
    def _tree_verify(self, node_tokens, lengths, seq_ids):
        """Target verifies all tree nodes in one masked multi-token pass
        (rows = L + node index, rope = L + depth)."""
        from neuronx_distributed_inference_amd.modules.attention.attention_base import (
            PHASE_TKG,
        )
        from neuronx_distributed_inference_amd.modules.token_tree import (
            tree_attention_mask_batch,
        )
        from neuronx_distributed_inference_amd.parallel.ops import distributed_argmax

        t = self.model.model
        tree = self.tree
        B = node_tokens.shape[0]
        N = tree.num_nodes
        device = node_tokens.device
        Smax = t.kv_mgr.max_len
        rows = lengths.view(B, 1) + torch.arange(N, device=device).view(1, N)
        rope = lengths.view(B, 1) + torch.tensor(tree.depth, device=device).view(1, N)
        mask = tree_attention_mask_batch(tree, self.anc, list(range(N)),
                                         lengths, Smax, device)
        emb = t.embed_tokens(node_tokens)
        cos, sin = t.rotary.get_cos_sin(rope, device=device)
        h = emb
        residual = None
        for layer in t.layers:
            h, residual = layer(h, residual, cos, sin, t.kv_mgr, rows,
                                PHASE_TKG, mask, seq_ids)
        h_final = t.norm(h, residual)[0] if residual is not None else t.norm(h)
        logits = t.lm_head(h_final)
        return distributed_argmax(logits, dim=-1), h_final

    @torch.inference_mode()
    def generate(self, input_ids, attention_mask=None, generation_config=None, **kw):
        gc = generation_config or GenerationConfig(**{
            k: v for k, v in kw.items() if k in GenerationConfig.__dataclass_fields__})
        B, S0 = input_ids.shape
        device = input_ids.device
        tree = self.tree
        N = tree.num_nodes
        if attention_mask is None:
            attention_mask = torch.ones_like(input_ids)
        max_len = gc.max_length or min(self.model.ec.max_length, S0 + gc.max_new_tokens)
        max_len = min(max_len, self.model.ec.max_length - N - 2)

        self.model.reset()
        seq_ids = torch.arange(B, device=device)
        cum = attention_mask.long().cumsum(-1) - 1
        position_ids = torch.where(attention_mask.bool(), cum, torch.full_like(cum, -1))
        t_out = self.model(input_ids=input_ids, attention_mask=attention_mask,
                           position_ids=position_ids, seq_ids=seq_ids)
        lengths = attention_mask.sum(-1).long()
        next_tok = t_out.tokens.view(B)
        hidden = self.model.model.last_hidden[:, -1, :]

        out_rows = [input_ids[i][attention_mask[i].bool()].tolist() for i in range(B)]
        for i in range(B):
            out_rows[i].append(int(next_tok[i]))
        finished = torch.zeros(B, dtype=torch.bool, device=device)

        while True:
            if gc.eos_token_id is not None:
                finished |= next_tok == gc.eos_token_id
            if bool(finished.all()) or max(len(r) for r in out_rows) >= max_len:
                break
            cand = self._head_topk(hidden)              # (B, depth, kmax)
            node_tokens = torch.zeros(B, N, dtype=torch.long, device=device)
            node_tokens[:, 0] = next_tok
            for n in range(1, N):
                d = tree.depth[n] - 1                   # head index
                node_tokens[:, n] = cand[:, d, self.child_rank[n]]
            tgt_toks, h_all = self._tree_verify(node_tokens, lengths, seq_ids)
            nt, tt = node_tokens.cpu(), tgt_toks.cpu()
            max_m = len(tree.branching)
            src = torch.zeros(B, max_m, dtype=torch.long)
            dst = torch.zeros(B, max_m, dtype=torch.long)
            bonus_nodes = torch.zeros(B, dtype=torch.long)
            n_acc = torch.zeros(B, dtype=torch.long)
            for b in range(B):
                path, bonus = tree.accept(nt[b], tt[b])
                L = int(lengths[b])
                for j in range(max_m):
                    src[b, j] = L + path[j] if j < len(path) else L
                    dst[b, j] = L + j + 1 if j < len(path) else L
                bonus_nodes[b] = bonus
                n_acc[b] = len(path)
                if finished[b]:
                    n_acc[b] = 0
                    continue
                row = out_rows[b]
                emitted = [int(nt[b, p]) for p in path] + [int(tt[b, bonus])]
                for tok in emitted:
                    if len(row) >= max_len:
                        break
                    row.append(tok)
                    if gc.eos_token_id is not None and tok == gc.eos_token_id:
                        finished[b] = True
                        break
            self.model.model.kv_mgr.compact_rows(
                seq_ids, src.to(device), dst.to(device))
            hidden = h_all.gather(
                1, bonus_nodes.to(device).view(B, 1, 1).expand(
                    -1, 1, h_all.shape[-1])).squeeze(1)
            lengths = lengths + n_acc.to(device) + 1
            next_tok = torch.tensor([r[-1] for r in out_rows], device=device,
                                    dtype=torch.long)

        longest = max(len(r) for r in out_rows)
        out = torch.full((B, longest), gc.pad_token_id, dtype=torch.long, device=device)
        for i, r in enumerate(out_rows):
            out[i, : len(r)] = torch.tensor(r, device=device)
        return out
