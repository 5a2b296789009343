"""Speculative decoding.

Vanilla two-model speculation (ref `utils/hf_adapter.py:439-798`
_standard_assisted_decoding + the `speculation_model` submodel,
model_base.py:3155-3179): the draft decodes `speculation_length` tokens
greedily, the target verifies all of them in ONE forward (the speculation
submodel graph, S = spec_len + 1), and the host accepts the longest
matching prefix plus the target's bonus token. Per-row positions, so
ragged batches work; rejected cache rows are overwritten as positions
advance (the same garbage-overwrite invariant the prefill padding uses).
"""

from __future__ import annotations

from typing import Optional

import torch

from neuronx_distributed_inference_amd.utils.hf_adapter import GenerationConfig


class SpeculativeGenerationAdapter:
    def __init__(self, model, draft_model, speculation_length: int):
        self.model = model
        self.draft = draft_model
        self.k = speculation_length

    @torch.inference_mode()
    def generate(
        self,
        input_ids: torch.Tensor,
        attention_mask: Optional[torch.Tensor] = None,
        generation_config: Optional[GenerationConfig] = None,
        **kwargs,
    ) -> torch.Tensor:
        gc = generation_config or GenerationConfig(**{
            k: v for k, v in kwargs.items() if k in GenerationConfig.__dataclass_fields__
        })
        B, S0 = input_ids.shape
        device = input_ids.device
        if attention_mask is None:
            attention_mask = torch.ones_like(input_ids)
        max_len = gc.max_length or min(self.model.ec.max_length, S0 + gc.max_new_tokens)
        max_len = min(max_len, self.model.ec.max_length - self.k - 2)
        k = self.k

        self.model.reset()
        self.draft.reset()
        seq_ids = torch.arange(B, device=device)

        cum = attention_mask.long().cumsum(-1) - 1
        position_ids = torch.where(attention_mask.bool(), cum, torch.full_like(cum, -1))
        t_out = self.model(input_ids=input_ids, attention_mask=attention_mask,
                           position_ids=position_ids, seq_ids=seq_ids)
        self.draft(input_ids=input_ids, attention_mask=attention_mask,
                   position_ids=position_ids, seq_ids=seq_ids)

        lengths = attention_mask.sum(-1).long()            # (B,) committed length
        next_tok = t_out.tokens.view(B)                    # last committed token, not yet in cache
        # per-row output buffers
        out_rows = [input_ids[i][attention_mask[i].bool()].tolist() for i in range(B)]
        for i in range(B):
            out_rows[i].append(int(next_tok[i]))
        finished = torch.zeros(B, dtype=torch.bool, device=device)

        while True:
            if gc.eos_token_id is not None:
                finished |= next_tok == gc.eos_token_id
            if bool(finished.all()) or max(len(r) for r in out_rows) >= max_len:
                break

            # ---- draft: k greedy decode steps ----
            drafts = []
            d_in = next_tok.view(B, 1)
            d_pos = lengths.view(B, 1)
            for j in range(k):
                d_out = self.draft(input_ids=d_in, position_ids=d_pos + j, seq_ids=seq_ids)
                d_in = d_out.tokens.view(B, 1)
                drafts.append(d_in)
            cand = torch.cat(drafts, dim=1)                 # (B, k)

            # ---- target verify: one forward over [t, d1..dk] ----
            verify_in = torch.cat([next_tok.view(B, 1), cand], dim=1)   # (B, k+1)
            verify_pos = lengths.view(B, 1) + torch.arange(k + 1, device=device).view(1, -1)
            v_out = self.model(input_ids=verify_in, position_ids=verify_pos, seq_ids=seq_ids)
            target_toks = v_out.tokens.view(B, k + 1)       # g_0..g_k

            # ---- accept longest matching prefix + bonus ----
            match = (cand == target_toks[:, :k])            # d_{j+1} vs g_j
            n_accept = (match.cumprod(dim=1)).sum(dim=1)    # (B,) 0..k
            for i in range(B):
                if finished[i]:
                    continue
                m = int(n_accept[i])
                accepted = cand[i, :m].tolist()
                bonus = int(target_toks[i, m])
                row = out_rows[i]
                for tok in accepted + [bonus]:
                    if len(row) >= max_len:
                        break
                    row.append(tok)
                    if gc.eos_token_id is not None and tok == gc.eos_token_id:
                        finished[i] = True
                        break
            lengths = lengths + n_accept + 1
            next_tok = torch.tensor(
                [r[-1] for r in out_rows], device=device, dtype=torch.long
            )

        # pad rows to a rectangle
        longest = max(len(r) for r in out_rows)
        out = torch.full((B, longest), gc.pad_token_id, dtype=torch.long, device=device)
        for i, r in enumerate(out_rows):
            out[i, : len(r)] = torch.tensor(r, device=device)
        return out
