"""EAGLE speculation.

Analog of the reference's EAGLE support (draft conditioned on
target hidden states via an `fc` combining (embed, prev_hidden)
`modeling_llama.py:1161-1167` / `model_base.py:1526-1592`, hidden-state
rolling buffer `modules/eagle/hidden_state.py`).

EAGLE-1 chain variant: the draft is a small llama-style stack whose layer-0
input is fc(cat(embed(token_{i+1}), target_hidden_i)); during drafting the
draft's own output hidden feeds the next step. Verification and acceptance
reuse the multi-token speculation forward.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd.modules.attention.attention_base import (
    PHASE_TKG,
)
from neuronx_distributed_inference_amd.parallel.ops import distributed_argmax
from neuronx_distributed_inference_amd.utils.hf_adapter import GenerationConfig


class EagleDraftModel(nn.Module):
    """fc + decoder layers + (shared or own) lm_head over a DecoderModel's
    building blocks. `base` is a DecoderModel (the draft's own tiny config)
    whose embed/lm_head/layers/kv_mgr are reused; the only structural
    addition is the fc combiner."""

    def __init__(self, base_model, hidden_size: int, dtype: torch.dtype):
        super().__init__()
        self.base = base_model  # DecoderModel (draft-sized)
        self.fc = nn.Linear(2 * hidden_size, hidden_size, bias=False, dtype=dtype)

    @torch.inference_mode()
    def forward(self, input_ids, prev_hidden, position_ids, seq_ids, phase,
                attention_mask=None, rope_position_ids=None, topk: int = 0):
        """input_ids (B,S): tokens t_{i+1}; prev_hidden (B,S,H): target (or
        draft) hidden at i. Returns (greedy tokens (B,S), draft hidden
        (B,S,H)[, top-k token ids (B,S,topk)]).

        Tree mode (ref tree TKG, model_base.py:2094): `position_ids`
        carries the CACHE-ROW positions (distinct per node) while
        `rope_position_ids` carries base+depth rotary positions, and a
        4-D `attention_mask` restricts each node to prior + ancestors."""
        b = self.base
        emb = b.embed_tokens(input_ids)
        h = self.fc(torch.cat([emb, prev_hidden.to(emb.dtype)], dim=-1))
        rp = rope_position_ids if rope_position_ids is not None else position_ids
        cos, sin = b.rotary.get_cos_sin(rp, device=h.device)
        residual = None
        for layer in b.layers:
            h, residual = layer(h, residual, cos, sin, b.kv_mgr, position_ids,
                                phase, attention_mask, seq_ids)
        if residual is not None:
            h_final, _ = b.norm(h, residual)
        else:
            h_final = b.norm(h)
        logits = b.lm_head(h_final)
        tokens = distributed_argmax(logits, dim=-1)
        if topk > 0:
            from neuronx_distributed_inference_amd.parallel.ops import (
                distributed_topk,
            )

            _, ids = distributed_topk(logits, topk, dim=-1)
            return tokens, h_final, ids
        return tokens, h_final




def eagle_draft_prefill(target, draft: "EagleDraftModel", input_ids,
                        attention_mask, position_ids, seq_ids, next_tok):
    """EAGLE draft CONTEXT ENCODING (ref EAGLE CTE, model_base.py:
    1931-2093): feed (token_{i+1}, target_hidden_i) for the whole prompt
    so the draft's KV cache holds a real prefix before the first
    speculation round. Requires target.model.eagle_capture_hidden=True
    during the target prefill (full_hidden kept)."""
    from neuronx_distributed_inference_amd.modules.attention.attention_base import (
        PHASE_CTE,
    )

    h_full = getattr(target.model, "full_hidden", None)
    if h_full is None:
        return  # capture flag was off: draft starts cold (still correct)
    B, S = input_ids.shape
    # the app may have padded CTE to a context bucket: captured hiddens
    # then carry the padded S — the prompt occupies the first S positions
    # (right padding)
    h_full = h_full[:, :S]
    # shifted tokens: t_1..t_{S-1}, next_tok; hiddens: h_0..h_{S-1}
    shifted = torch.cat([input_ids[:, 1:], next_tok.view(B, 1)], dim=1)
    draft(shifted, h_full, position_ids, seq_ids, PHASE_CTE,
          attention_mask=attention_mask)


class EagleGenerationAdapter:
    """EAGLE chain decoding: draft k tokens conditioned on target hiddens,
    verify with one multi-token target forward, accept the matching prefix."""

    def __init__(self, target, draft: EagleDraftModel, speculation_length: int):
        self.target = target
        self.draft = draft
        self.k = speculation_length

    @torch.inference_mode()
    def generate(self, input_ids, attention_mask=None, generation_config=None, **kw):
        gc = generation_config or GenerationConfig(**{
            k: v for k, v in kw.items() if k in GenerationConfig.__dataclass_fields__})
        B, S0 = input_ids.shape
        device = input_ids.device
        if attention_mask is None:
            attention_mask = torch.ones_like(input_ids)
        max_len = gc.max_length or min(self.target.ec.max_length, S0 + gc.max_new_tokens)
        max_len = min(max_len, self.target.ec.max_length - self.k - 2)
        k = self.k

        self.target.reset()
        self.draft.base.reset_cache()
        seq_ids = torch.arange(B, device=device)
        cum = attention_mask.long().cumsum(-1) - 1
        position_ids = torch.where(attention_mask.bool(), cum, torch.full_like(cum, -1))
        self.target.model.eagle_capture_hidden = True
        t_out = self.target(input_ids=input_ids, attention_mask=attention_mask,
                            position_ids=position_ids, seq_ids=seq_ids)
        self.target.model.eagle_capture_hidden = False
        lengths = attention_mask.sum(-1).long()
        next_tok = t_out.tokens.view(B)
        # target hidden of the LAST prompt token conditions the draft
        # (model_base already gathered CTE last_hidden at the true last
        # prompt position, mask-aware)
        h_tgt = self.target.model.last_hidden[:, -1, :]     # (B, H)

        # draft prefill: (token_{i+1}, target_hidden_i) over the prompt
        # fills the draft's KV cache (ref EAGLE CTE)
        eagle_draft_prefill(self.target, self.draft, input_ids,
                            attention_mask, position_ids, seq_ids, next_tok)
        out_rows = [input_ids[i][attention_mask[i].bool()].tolist() for i in range(B)]
        for i in range(B):
            out_rows[i].append(int(next_tok[i]))
        finished = torch.zeros(B, dtype=torch.bool, device=device)

        while True:
            if gc.eos_token_id is not None:
                finished |= next_tok == gc.eos_token_id
            if bool(finished.all()) or max(len(r) for r in out_rows) >= max_len:
                break

            # ---- draft k chained tokens conditioned on hiddens ----
            cand = []
            d_tok = next_tok.view(B, 1)
            d_hid = h_tgt.view(B, 1, -1)
            d_pos = lengths.view(B, 1)
            for j in range(k):
                t_j, h_j = self.draft(d_tok, d_hid, d_pos + j, seq_ids, PHASE_TKG)
                d_tok = t_j.view(B, 1)
                d_hid = h_j[:, -1:, :]
                cand.append(d_tok)
            cand = torch.cat(cand, dim=1)

            # ---- target verify ----
            verify_in = torch.cat([next_tok.view(B, 1), cand], dim=1)
            verify_pos = lengths.view(B, 1) + torch.arange(k + 1, device=device).view(1, -1)
            v_out = self.target(input_ids=verify_in, position_ids=verify_pos, seq_ids=seq_ids)
            target_toks = v_out.tokens.view(B, k + 1)
            hidden_all = self.target.model.last_hidden[:, : k + 1]  # (B, k+1, H)

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
            h_tgt = hidden_all.gather(
                1, n_accept.view(B, 1, 1).expand(-1, 1, hidden_all.shape[-1])
            ).squeeze(1)
            lengths = lengths + n_accept + 1
            next_tok = torch.tensor([r[-1] for r in out_rows], device=device, dtype=torch.long)

        longest = max(len(r) for r in out_rows)
        out = torch.full((B, longest), gc.pad_token_id, dtype=torch.long, device=device)
        for i, r in enumerate(out_rows):
            out[i, : len(r)] = torch.tensor(r, device=device)
        return out


class EagleTreeSpeculationModel(nn.Module):
    """EAGLE tree-structured token generation (ref
    `_eagle_tree_token_gen_forward`, model_base.py:2094 + the
    token_tree masks/positions). One forward = draft tree expansion
    (level by level, top-k children per node) + target verify of ALL
    nodes in one masked pass — every step is tensor ops with STATIC
    shapes from the static tree, so the round is hipGraph-capturable.

    Node n occupies cache row L+n (distinct rows; siblings share a depth
    hence a ROPE position but never a row); attention uses the explicit
    tree mask (prior + ancestor chain) via the 4-D-mask decode path.
    """

    def __init__(self, target_model, draft: "EagleDraftModel", tree):
        super().__init__()
        self.target = target_model  # DecoderModel
        self.draft = draft
        self.tree = tree
        from neuronx_distributed_inference_amd.modules.token_tree import (
            ancestor_matrix,
        )

        self.register_buffer("anc", ancestor_matrix(tree), persistent=False)
        self.register_buffer("depth_t", torch.tensor(tree.depth), persistent=False)

    def _mask(self, nodes, lengths, Smax, device):
        from neuronx_distributed_inference_amd.modules.token_tree import (
            tree_attention_mask_batch,
        )

        return tree_attention_mask_batch(self.tree, self.anc, nodes, lengths,
                                         Smax, device)

    @torch.inference_mode()
    def forward(self, input_ids, prev_hidden, lengths, seq_ids):
        """input_ids (B,1) committed-unfed token at position lengths;
        prev_hidden (B,1,H) target hidden conditioning the draft.
        Returns (node_tokens (B,N), target_tokens (B,N),
        target_hidden (B,N,H)) — the host walks acceptance."""
        from neuronx_distributed_inference_amd.modules.attention.attention_base import (
            PHASE_TKG,
        )

        tree = self.tree
        B = input_ids.shape[0]
        device = input_ids.device
        N = tree.num_nodes
        Smax = min(self.target.kv_mgr.max_len,
                   int(self.draft.base.kv_mgr.max_len))
        node_tokens = torch.zeros(B, N, dtype=torch.long, device=device)
        node_tokens[:, 0] = input_ids.view(B)
        node_hidden = [None] * N            # draft conditioning per node
        node_hidden[0] = prev_hidden.view(B, 1, -1)

        # ---- draft expansion, level by level ----
        for d, k_d in enumerate(tree.branching):
            nodes = tree.level_nodes[d]
            w = len(nodes)
            tok_lvl = node_tokens[:, nodes]                    # (B, w)
            hid_lvl = torch.cat([node_hidden[n] for n in nodes], dim=1)
            rows = lengths.view(B, 1) + torch.tensor(
                nodes, device=device).view(1, w)
            rope = lengths.view(B, 1) + d
            mask = self._mask(nodes, lengths, Smax, device)
            _, h_lvl, topk_ids = self.draft(
                tok_lvl, hid_lvl, rows, seq_ids, PHASE_TKG,
                attention_mask=mask, rope_position_ids=rope.expand(B, w),
                topk=k_d)
            # children of nodes[j] get its top-k_d tokens, in rank order
            child_iter = {n: 0 for n in nodes}
            for child in tree.level_nodes[d + 1]:
                p = tree.parent[child]
                j = nodes.index(p)
                r = child_iter[p]
                child_iter[p] += 1
                node_tokens[:, child] = topk_ids[:, j, r]
                node_hidden[child] = h_lvl[:, j:j + 1]

        # ---- target verify: all N nodes, one masked pass ----
        t = self.target
        rows = lengths.view(B, 1) + torch.arange(N, device=device).view(1, N)
        rope = lengths.view(B, 1) + self.depth_t.view(1, N).to(device)
        mask = self._mask(list(range(N)), lengths, Smax, device)
        emb = t.embed_tokens(node_tokens)
        cos, sin = t.rotary.get_cos_sin(rope, device=device)
        h = emb
        residual = None
        for layer in t.layers:
            h, residual = layer(h, residual, cos, sin, t.kv_mgr, rows,
                                PHASE_TKG, mask, seq_ids)
        if residual is not None:
            h_final, _ = t.norm(h, residual)
        else:
            h_final = t.norm(h)
        logits = t.lm_head(h_final)
        target_tokens = distributed_argmax(logits, dim=-1)     # (B, N)
        return node_tokens, target_tokens, h_final


class EagleTreeAdapter:
    """Host loop around EagleTreeSpeculationModel: per round, one tree
    forward; acceptance walks the deepest target-matching path, the KV
    caches compact the accepted rows into the committed prefix."""

    def __init__(self, target, draft: "EagleDraftModel", tree):
        self.target = target
        self.draft = draft
        self.tree = tree
        self.model = EagleTreeSpeculationModel(target.model, draft, tree)

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
        max_len = gc.max_length or min(self.target.ec.max_length, S0 + gc.max_new_tokens)
        max_len = min(max_len, self.target.ec.max_length - N - 2)

        self.target.reset()
        self.draft.base.reset_cache()
        seq_ids = torch.arange(B, device=device)
        cum = attention_mask.long().cumsum(-1) - 1
        position_ids = torch.where(attention_mask.bool(), cum, torch.full_like(cum, -1))
        self.target.model.eagle_capture_hidden = True
        t_out = self.target(input_ids=input_ids, attention_mask=attention_mask,
                            position_ids=position_ids, seq_ids=seq_ids)
        self.target.model.eagle_capture_hidden = False
        lengths = attention_mask.sum(-1).long()
        next_tok = t_out.tokens.view(B)
        h_tgt = self.target.model.last_hidden[:, -1, :]
        eagle_draft_prefill(self.target, self.draft, input_ids,
                            attention_mask, position_ids, seq_ids, next_tok)

        out_rows = [input_ids[i][attention_mask[i].bool()].tolist() for i in range(B)]
        for i in range(B):
            out_rows[i].append(int(next_tok[i]))
        finished = torch.zeros(B, dtype=torch.bool, device=device)

        while True:
            if gc.eos_token_id is not None:
                finished |= next_tok == gc.eos_token_id
            if bool(finished.all()) or max(len(r) for r in out_rows) >= max_len:
                break
            node_toks, tgt_toks, tgt_hidden = self.model(
                next_tok.view(B, 1), h_tgt.view(B, 1, -1), lengths, seq_ids)
            nt = node_toks.cpu()
            tt = tgt_toks.cpu()
            max_m = len(tree.branching)
            src = torch.zeros(B, max_m, dtype=torch.long)
            dst = torch.zeros(B, max_m, dtype=torch.long)
            bonus_nodes = torch.zeros(B, dtype=torch.long)
            n_acc = torch.zeros(B, dtype=torch.long)
            for b in range(B):
                path, bonus_node = tree.accept(nt[b], tt[b])
                L = int(lengths[b])
                for j in range(max_m):
                    node = path[j] if j < len(path) else 0
                    src[b, j] = L + node if j < len(path) else L
                    dst[b, j] = L + j + 1 if j < len(path) else L
                bonus_nodes[b] = bonus_node
                n_acc[b] = len(path)
                if finished[b]:
                    n_acc[b] = 0
                    continue
                row = out_rows[b]
                emitted = [int(nt[b, p]) for p in path] + [int(tt[b, bonus_node])]
                for tok in emitted:
                    if len(row) >= max_len:
                        break
                    row.append(tok)
                    if gc.eos_token_id is not None and tok == gc.eos_token_id:
                        finished[b] = True
                        break
            # compact accepted rows into the contiguous prefix (both caches)
            src_d, dst_d = src.to(device), dst.to(device)
            self.target.model.kv_mgr.compact_rows(seq_ids, src_d, dst_d)
            self.draft.base.kv_mgr.compact_rows(seq_ids, src_d, dst_d)
            h_tgt = tgt_hidden.gather(
                1, bonus_nodes.to(device).view(B, 1, 1).expand(
                    -1, 1, tgt_hidden.shape[-1])).squeeze(1)
            lengths = lengths + n_acc.to(device) + 1
            next_tok = torch.tensor([r[-1] for r in out_rows], device=device,
                                    dtype=torch.long)

        longest = max(len(r) for r in out_rows)
        out = torch.full((B, longest), gc.pad_token_id, dtype=torch.long, device=device)
        for i, r in enumerate(out_rows):
            out[i, : len(r)] = torch.tensor(r, device=device)
        return out


class Eagle3DraftModel(nn.Module):
    """EAGLE-3 draft (ref model_base.py:1402-1437): the combiner takes
    the token embedding plus THREE target layers' hidden states
    concatenated — fc: (emb, h_low, h_mid, h_high) -> H. The target
    exposes them via `eagle3_capture_layers`/`eagle3_hiddens`."""

    def __init__(self, base_model, hidden_size: int, dtype: torch.dtype,
                 capture_layers):
        super().__init__()
        self.base = base_model          # DecoderModel (draft-sized)
        self.capture_layers = list(capture_layers)
        n = 1 + len(self.capture_layers)
        self.fc = nn.Linear(n * hidden_size, hidden_size, bias=False, dtype=dtype)

    @torch.inference_mode()
    def forward(self, input_ids, prev_hiddens, position_ids, seq_ids, phase,
                attention_mask=None, rope_position_ids=None):
        """prev_hiddens: (B, S, 3H) — the concatenated capture-layer
        hiddens at the conditioning positions."""
        b = self.base
        emb = b.embed_tokens(input_ids)
        h = self.fc(torch.cat([emb, prev_hiddens.to(emb.dtype)], dim=-1))
        rp = rope_position_ids if rope_position_ids is not None else position_ids
        cos, sin = b.rotary.get_cos_sin(rp, device=h.device)
        residual = None
        for layer in b.layers:
            h, residual = layer(h, residual, cos, sin, b.kv_mgr, position_ids,
                                phase, attention_mask, seq_ids)
        h_final = b.norm(h, residual)[0] if residual is not None else b.norm(h)
        logits = b.lm_head(h_final)
        tokens = distributed_argmax(logits, dim=-1)
        return tokens, h_final


class Eagle3GenerationAdapter:
    """EAGLE-3 chain decoding: like EAGLE-1 but the draft's conditioning
    vector is cat of 3 target layers' hiddens; after the first step the
    draft chains on its OWN hidden replicated across the 3 slots (the
    reference's draft-feedback convention)."""

    def __init__(self, target, draft: Eagle3DraftModel, speculation_length: int):
        self.target = target
        self.draft = draft
        self.k = speculation_length
        target.model.eagle3_capture_layers = draft.capture_layers

    def _cat_capture(self, B):
        hs = self.target.model.eagle3_hiddens
        return torch.cat([hs[li] for li in self.draft.capture_layers], dim=-1)

    @torch.inference_mode()
    def generate(self, input_ids, attention_mask=None, generation_config=None, **kw):
        gc = generation_config or GenerationConfig(**{
            k: v for k, v in kw.items() if k in GenerationConfig.__dataclass_fields__})
        B, S0 = input_ids.shape
        device = input_ids.device
        if attention_mask is None:
            attention_mask = torch.ones_like(input_ids)
        max_len = gc.max_length or min(self.target.ec.max_length, S0 + gc.max_new_tokens)
        max_len = min(max_len, self.target.ec.max_length - self.k - 2)
        k = self.k
        nh = len(self.draft.capture_layers)

        self.target.reset()
        self.draft.base.reset_cache()
        seq_ids = torch.arange(B, device=device)
        cum = attention_mask.long().cumsum(-1) - 1
        position_ids = torch.where(attention_mask.bool(), cum, torch.full_like(cum, -1))
        t_out = self.target(input_ids=input_ids, attention_mask=attention_mask,
                            position_ids=position_ids, seq_ids=seq_ids)
        lengths = attention_mask.sum(-1).long()
        next_tok = t_out.tokens.view(B)
        cap = self._cat_capture(B)[:, :S0]            # (B, S, 3H)
        # draft context encoding over the prompt (shifted tokens vs the
        # captured hiddens)
        from neuronx_distributed_inference_amd.modules.attention.attention_base import (
            PHASE_CTE,
        )

        shifted = torch.cat([input_ids[:, 1:], next_tok.view(B, 1)], dim=1)
        self.draft(shifted, cap, position_ids, seq_ids, PHASE_CTE,
                   attention_mask=attention_mask)
        h_cond = cap.gather(
            1, (lengths - 1).clamp(min=0).view(B, 1, 1).expand(-1, 1, cap.shape[-1])
        )                                             # (B, 1, 3H)

        out_rows = [input_ids[i][attention_mask[i].bool()].tolist() for i in range(B)]
        for i in range(B):
            out_rows[i].append(int(next_tok[i]))
        finished = torch.zeros(B, dtype=torch.bool, device=device)

        while True:
            if gc.eos_token_id is not None:
                finished |= next_tok == gc.eos_token_id
            if bool(finished.all()) or max(len(r) for r in out_rows) >= max_len:
                break
            cand = []
            d_tok = next_tok.view(B, 1)
            d_hid = h_cond
            d_pos = lengths.view(B, 1)
            for j in range(k):
                t_j, h_j = self.draft(d_tok, d_hid, d_pos + j, seq_ids, PHASE_TKG)
                d_tok = t_j.view(B, 1)
                d_hid = h_j[:, -1:, :].repeat(1, 1, nh)   # draft feedback
                cand.append(d_tok)
            cand = torch.cat(cand, dim=1)

            verify_in = torch.cat([next_tok.view(B, 1), cand], dim=1)
            verify_pos = lengths.view(B, 1) + torch.arange(k + 1, device=device).view(1, -1)
            v_out = self.target(input_ids=verify_in, position_ids=verify_pos,
                                seq_ids=seq_ids)
            target_toks = v_out.tokens.view(B, k + 1)
            cap_all = self._cat_capture(B)[:, : k + 1]  # (B, k+1, 3H)

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
            h_cond = cap_all.gather(
                1, n_accept.view(B, 1, 1).expand(-1, 1, cap_all.shape[-1]))
            lengths = lengths + n_accept + 1
            next_tok = torch.tensor([r[-1] for r in out_rows], device=device,
                                    dtype=torch.long)

        longest = max(len(r) for r in out_rows)
        out = torch.full((B, longest), gc.pad_token_id, dtype=torch.long, device=device)
        for i, r in enumerate(out_rows):
            out[i, : len(r)] = torch.tensor(r, device=device)
        return out
