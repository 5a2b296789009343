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


class FusedSpeculationModel(torch.nn.Module):
    """Draft unroll + target verify + accept logic in ONE module, fully
    tensor-side (no host sync), so one hipGraph covers a whole speculation
    round (ref NeuronFusedSpecModel, model_base.py:1598-3021: draft loop
    :1839-1888, verify :1898-1914, in-graph accept `_tkg_postprocessor`
    :2799-2855, acceptance_padding_token = -1)."""

    ACCEPT_PAD = -1

    def __init__(self, target, draft, speculation_length: int):
        super().__init__()
        self.target = target
        self.draft = draft
        self.k = speculation_length

    def forward(self, input_ids: torch.Tensor, position_ids: torch.Tensor,
                seq_ids: torch.Tensor):
        """input_ids (B,1): last committed (unfed) token; position_ids (B,1)
        its position. Returns (accepted_tokens (B,k+1) padded with -1,
        n_accept (B,))."""
        from neuronx_distributed_inference_amd.modules.attention.attention_base import (
            PHASE_TKG,
        )

        B = input_ids.shape[0]
        k = self.k
        device = input_ids.device
        d_in = input_ids
        cand = []
        for i in range(k):
            d_tok, _ = self.draft(
                d_in, None, position_ids + i, seq_ids, None, phase=PHASE_TKG
            )
            d_in = d_tok.view(B, 1)
            cand.append(d_in)
        cand = torch.cat(cand, dim=1)                       # (B, k)

        verify_in = torch.cat([input_ids, cand], dim=1)     # (B, k+1)
        verify_pos = position_ids + torch.arange(k + 1, device=device).view(1, -1)
        t_toks, _ = self.target(
            verify_in, None, verify_pos, seq_ids, None, phase=PHASE_TKG
        )
        t_toks = t_toks.view(B, k + 1)

        match = (cand == t_toks[:, :k]).long()
        n_accept = match.cumprod(dim=1).sum(dim=1)          # (B,)
        idx = torch.arange(k + 1, device=device).view(1, -1)
        bonus = t_toks.gather(1, n_accept.view(B, 1))       # (B,1)
        cand_ext = torch.cat([cand, torch.zeros(B, 1, dtype=cand.dtype, device=device)], 1)
        accepted = torch.where(
            idx < n_accept.view(B, 1), cand_ext,
            torch.where(idx == n_accept.view(B, 1), bonus.expand(B, k + 1),
                        torch.full_like(cand_ext, self.ACCEPT_PAD)),
        )
        return accepted, n_accept


class FusedSpeculativeAdapter:
    """Host loop around FusedSpeculationModel: one device dispatch per
    speculation round (ref _fused_assisted_decoding, hf_adapter.py:495)."""

    def __init__(self, target, draft, speculation_length: int, use_graphs: bool = True):
        self.target = target
        self.draft = draft
        self.k = speculation_length
        self.fused = FusedSpeculationModel(target.model, draft.model, speculation_length)
        self.use_graphs = use_graphs
        self._graph = None

    def _maybe_capture(self, B, device):
        if not (self.use_graphs and device.type == "cuda"):
            return
        from neuronx_distributed_inference_amd.runtime.graph_engine import GraphRunner

        ids = torch.zeros(B, 1, dtype=torch.long, device=device)
        pos = torch.zeros(B, 1, dtype=torch.long, device=device)
        sid = torch.arange(B, device=device)
        try:
            self._graph = GraphRunner(self.fused, (ids, pos, sid))
        except Exception:  # noqa: BLE001
            import logging

            logging.getLogger("nxdi_amd").exception(
                "fused-spec capture failed; running eager")
            self._graph = None
            torch.cuda.synchronize()

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

        self.target.reset()
        self.draft.reset()
        seq_ids = torch.arange(B, device=device)
        cum = attention_mask.long().cumsum(-1) - 1
        position_ids = torch.where(attention_mask.bool(), cum, torch.full_like(cum, -1))
        t_out = self.target(input_ids=input_ids, attention_mask=attention_mask,
                            position_ids=position_ids, seq_ids=seq_ids)
        self.draft(input_ids=input_ids, attention_mask=attention_mask,
                   position_ids=position_ids, seq_ids=seq_ids)
        if self._graph is None:
            self._maybe_capture(B, device)

        lengths = attention_mask.sum(-1).long()
        next_tok = t_out.tokens.view(B)
        out_rows = [input_ids[i][attention_mask[i].bool()].tolist() for i in range(B)]
        for i in range(B):
            out_rows[i].append(int(next_tok[i]))
        finished = torch.zeros(B, dtype=torch.bool, device=device)

        while True:
            if gc.eos_token_id is not None:
                finished |= next_tok == gc.eos_token_id
            if bool(finished.all()) or max(len(r) for r in out_rows) >= max_len:
                break
            fn = self._graph if self._graph is not None else self.fused
            accepted, n_accept = fn(next_tok.view(B, 1), lengths.view(B, 1), seq_ids)
            acc = accepted.cpu()
            nacc = n_accept.cpu()
            for i in range(B):
                if finished[i]:
                    continue
                row = out_rows[i]
                for j in range(int(nacc[i]) + 1):
                    tok = int(acc[i, j])
                    if len(row) >= max_len:
                        break
                    row.append(tok)
                    if gc.eos_token_id is not None and tok == gc.eos_token_id:
                        finished[i] = True
                        break
            lengths = lengths + n_accept + 1
            next_tok = torch.tensor([r[-1] for r in out_rows], device=device, dtype=torch.long)

        longest = max(len(r) for r in out_rows)
        out = torch.full((B, longest), gc.pad_token_id, dtype=torch.long, device=device)
        for i, r in enumerate(out_rows):
            out[i, : len(r)] = torch.tensor(r, device=device)
        return out
