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
