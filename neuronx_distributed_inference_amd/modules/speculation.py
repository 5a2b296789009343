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


class DraftTPWrapper:
    """Proxy that runs every draft call inside the reduced-TP group
    override (ref speculative-draft group, application_base.py:61-66):
    the draft's parallel layers and distributed argmax/topk then
    collect over draft_tp ranks instead of the full TP group. Weights
    are replicated across subgroups, so every rank computes the draft
    locally — no token broadcast."""

    def __init__(self, inner):
        object.__setattr__(self, "_inner", inner)

    def __call__(self, *a, **kw):
        from neuronx_distributed_inference_amd.parallel import state as ps

        with ps.draft_tp_context():
            return self._inner(*a, **kw)

    def generate(self, *a, **kw):
        from neuronx_distributed_inference_amd.parallel import state as ps

        with ps.draft_tp_context():
            return self._inner.generate(*a, **kw)

    def reset(self, *a, **kw):
        from neuronx_distributed_inference_amd.parallel import state as ps

        with ps.draft_tp_context():
            return self._inner.reset(*a, **kw)

    def __getattr__(self, name):
        return getattr(object.__getattribute__(self, "_inner"), name)


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

    def __init__(self, target, draft, speculation_length: int,
                 rejection_config=None):
        """`rejection_config`: an OnDeviceSamplingConfig -> true rejection
        sampling (accept d_j iff rand < p(d_j)/q(d_j), resample from the
        residual; ref model_base.py:1678-1759), so speculation works with
        top-k/top-p sampling. None -> exact greedy token match."""
        super().__init__()
        self.target = target
        self.draft = draft
        self.k = speculation_length
        self.rejection = None
        if rejection_config is not None:
            from neuronx_distributed_inference_amd.modules.generation.sampling import (
                RejectionSampler,
            )

            vocab = getattr(target, "vocab_size", None) or target.config.vocab_size
            self.rejection = RejectionSampler(rejection_config, vocab)
            target.spec_capture_topk = self.rejection.k_g
            draft.spec_capture_topk = self.rejection.k_g

    def forward(self, input_ids: torch.Tensor, position_ids: torch.Tensor,
                seq_ids: torch.Tensor,
                sampling_params: Optional[torch.Tensor] = None):
        """input_ids (B,1): last committed (unfed) token; position_ids (B,1)
        its position. Returns (accepted_tokens (B,k+1) padded with -1,
        n_accept (B,))."""
        from neuronx_distributed_inference_amd.modules.attention.attention_base import (
            PHASE_TKG,
        )

        B = input_ids.shape[0]
        k = self.k
        device = input_ids.device
        if self.rejection is not None:
            return self._forward_rejection(
                input_ids, position_ids, seq_ids, sampling_params)
        d_in = input_ids
        cand = []
        from neuronx_distributed_inference_amd.parallel import state as _ps

        for i in range(k):
            # draft collectives run over the draft sub-communicator when
            # one was initialized (no-op context otherwise)
            with _ps.draft_tp_context():
                d_tok, _ = self.draft(
                    d_in, None, position_ids + i, seq_ids, None,
                    phase=PHASE_TKG)
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

    def _forward_rejection(self, input_ids, position_ids, seq_ids,
                           sampling_params):
        """Sampled draft + rejection accept + residual resample, all
        tensor-side (one hipGraph covers the round)."""
        from neuronx_distributed_inference_amd.modules.attention.attention_base import (
            PHASE_TKG,
        )
        from neuronx_distributed_inference_amd.modules.generation.sampling import (
            prepare_sampling_params,
        )

        B = input_ids.shape[0]
        k = self.k
        device = input_ids.device
        rj = self.rejection
        if sampling_params is None:
            c = rj.config
            sampling_params = prepare_sampling_params(
                B, c.top_k if c.do_sample else 1, c.top_p, c.temperature,
                device=device)

        d_in = input_ids
        cand = []
        q_probs, q_idx = [], []
        for i in range(k):
            # the draft model's own sampler output is ignored: the round
            # samples d_j itself from the SAME filtered distribution it
            # uses for q(d_j), so accept ratios are exact by construction
            from neuronx_distributed_inference_amd.parallel import (
                state as _ps,
            )

            with _ps.draft_tp_context():
                self.draft(d_in, None, position_ids + i, seq_ids,
                           sampling_params, phase=PHASE_TKG)
            vals, idxs = self.draft.spec_topk           # (B,1,Kg)
            from neuronx_distributed_inference_amd.modules.generation.sampling import (
                filtered_probs,
            )

            qp = filtered_probs(vals, sampling_params)
            d_tok = rj.sample(qp, idxs)                 # (B,1)
            d_in = d_tok.view(B, 1)
            cand.append(d_in)
            q_probs.append(qp)
            q_idx.append(idxs)
        cand = torch.cat(cand, dim=1)                   # (B,k)
        q_probs = torch.cat(q_probs, dim=1)             # (B,k,Kg)
        q_idx = torch.cat(q_idx, dim=1)

        verify_in = torch.cat([input_ids, cand], dim=1)
        verify_pos = position_ids + torch.arange(k + 1, device=device).view(1, -1)
        self.target(verify_in, None, verify_pos, seq_ids,
                    sampling_params, phase=PHASE_TKG)
        t_vals, t_idx = self.target.spec_topk           # (B,k+1,Kg)
        from neuronx_distributed_inference_amd.modules.generation.sampling import (
            filtered_probs,
        )

        p_probs = filtered_probs(t_vals, sampling_params)
        return rj.accept(cand, q_probs, q_idx, p_probs, t_idx,
                         pad_id=self.ACCEPT_PAD)


class FusedSpeculativeAdapter:
    """Host loop around FusedSpeculationModel: one device dispatch per
    speculation round (ref _fused_assisted_decoding, hf_adapter.py:495).

    The loop is de-Pythonized: accepted tokens are compacted into a
    device-side (B, max_len) buffer with tensor ops (no per-row Python
    appends), and the host syncs only one all-finished flag every
    `sync_every` rounds — batch-256 friendly."""

    SYNC_EVERY = 4

    def __init__(self, target, draft, speculation_length: int,
                 use_graphs: bool = True, rejection_config=None):
        self.target = target
        self.draft = draft
        self.k = speculation_length
        self.fused = FusedSpeculationModel(
            target.model, draft.model, speculation_length,
            rejection_config=rejection_config)
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
        k = self.k

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

        # device-side state: output buffer, per-row lengths, finished flags
        pad = gc.pad_token_id if gc.pad_token_id is not None else 0
        out_buf = torch.full((B, max_len + k + 1), pad,
                             dtype=torch.long, device=device)
        lengths = attention_mask.sum(-1).long()             # committed length
        am = attention_mask.bool()
        # left-align the prompt rows into the buffer
        src_pos = (cum.clamp(min=0))
        out_buf.scatter_(1, torch.where(am, src_pos, torch.full_like(src_pos, max_len + k)),
                         torch.where(am, input_ids, torch.full_like(input_ids, pad)))
        out_buf[:, max_len + k] = pad
        next_tok = t_out.tokens.view(B)
        out_buf.scatter_(1, lengths.view(B, 1), next_tok.view(B, 1))
        lengths = lengths + 1
        finished = torch.zeros(B, dtype=torch.bool, device=device)
        if gc.eos_token_id is not None:
            finished |= next_tok == gc.eos_token_id

        idxpos = torch.arange(k + 1, device=device).view(1, -1)
        rounds = 0
        max_rounds = max_len  # worst case: 1 token per round
        while rounds < max_rounds:
            fn = self._graph if self._graph is not None else self.fused
            # next_tok sits at position lengths-1 (committed but not yet
            # fed); the fused round feeds it there and drafts from there
            accepted, n_accept = fn(next_tok.view(B, 1),
                                    (lengths - 1).view(B, 1), seq_ids)
            # --- device-side emission (no per-row python) ---
            n_eff = torch.where(finished, torch.full_like(n_accept, -1), n_accept)
            span = idxpos <= n_eff.view(B, 1)               # candidate slots
            if gc.eos_token_id is not None:
                is_eos = span & (accepted == gc.eos_token_id)
                # first eos position (k+1 if none)
                eos_pos = torch.where(
                    is_eos.any(1),
                    is_eos.float().argmax(1).long(),
                    torch.full((B,), k + 1, dtype=torch.long, device=device))
            else:
                eos_pos = torch.full((B,), k + 1, dtype=torch.long, device=device)
            emit = torch.minimum(n_eff + 1, eos_pos + 1)
            emit = torch.minimum(emit, (max_len - lengths).clamp(min=0))
            valid = idxpos < emit.view(B, 1)
            tgt = lengths.view(B, 1) + idxpos               # write positions
            tgt = torch.where(valid, tgt, torch.full_like(tgt, max_len + k))
            out_buf.scatter_(1, tgt, torch.where(valid, accepted,
                                                 torch.full_like(accepted, pad)))
            out_buf[:, max_len + k] = pad
            lengths = lengths + emit
            finished = finished | (eos_pos <= n_eff) | (lengths >= max_len)
            next_tok = out_buf.gather(1, (lengths - 1).clamp(min=0).view(B, 1)).view(B)
            rounds += 1
            # ONE host sync every SYNC_EVERY rounds
            if rounds % self.SYNC_EVERY == 0 and bool(finished.all()):
                break

        longest = int(lengths.max())
        return out_buf[:, :longest]
