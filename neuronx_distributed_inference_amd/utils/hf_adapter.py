"""HuggingFace-style generation adapter.

Analog of the reference's `utils/hf_adapter.py` (HuggingFaceGenerationAdapter
:133, token loop :139-405): the host-side generate loop — one device
dispatch per step (prefill once, then decode), tokens appended on host,
stopping on eos/max_length. Speculative assisted decoding plugs in here.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch

from neuronx_distributed_inference_amd.models.model_base import CausalLMBase
from neuronx_distributed_inference_amd.modules.generation.sampling import (
    prepare_sampling_params,
)


@dataclass
class GenerationConfig:
    max_new_tokens: int = 32
    max_length: Optional[int] = None
    do_sample: bool = False
    top_k: int = 1
    top_p: float = 1.0
    temperature: float = 1.0
    eos_token_id: Optional[int] = None
    pad_token_id: int = 0


class HuggingFaceGenerationAdapter:
    def __init__(self, model: CausalLMBase):
        self.model = model

    @torch.inference_mode()
    def generate(
        self,
        input_ids: torch.Tensor,
        attention_mask: Optional[torch.Tensor] = None,
        generation_config: Optional[GenerationConfig] = None,
        **kwargs,
    ) -> torch.Tensor:
        if generation_config is not None:
            gc = generation_config
        else:
            fields = {
                k: v for k, v in kwargs.items()
                if k in GenerationConfig.__dataclass_fields__
            }
            gc = GenerationConfig(**fields)
        model = self.model
        ec = model.ec
        B, S0 = input_ids.shape
        device = input_ids.device
        if attention_mask is None:
            attention_mask = torch.ones_like(input_ids)
        max_len = gc.max_length or min(ec.max_length, S0 + gc.max_new_tokens)
        max_len = min(max_len, ec.max_length)

        sampling_params = None
        if ec.on_device_sampling_config is not None:
            sampling_params = prepare_sampling_params(
                B,
                gc.top_k if gc.do_sample else 1,
                gc.top_p if gc.do_sample else 1.0,
                gc.temperature,
                device=device,
            )

        model.reset()
        seq_ids = torch.arange(B, device=device)
        out_ids = input_ids
        cur_mask = attention_mask
        finished = torch.zeros(B, dtype=torch.bool, device=device)

        # ---- prefill ---- (padded entries get position -1: skipped by the
        # KV scatter, masked out of attention)
        cum = cur_mask.long().cumsum(-1) - 1
        position_ids = torch.where(cur_mask.bool(), cum, torch.full_like(cum, -1))
        out = model(
            input_ids=input_ids,
            attention_mask=cur_mask,
            position_ids=position_ids,
            seq_ids=seq_ids,
            sampling_params=sampling_params,
        )
        next_tokens = self._next_token(out, gc)
        lengths = cur_mask.sum(-1)  # (B,) true lengths

        # eos handling: the `finished.all()` readback is the only per-step
        # host sync; skip it entirely when no eos is requested (the analog of
        # the reference's async mode hiding the token readback,
        # modules/async_execution.py)
        check_eos = gc.eos_token_id is not None
        while True:
            next_tokens = torch.where(
                finished, torch.full_like(next_tokens, gc.pad_token_id), next_tokens
            )
            out_ids = torch.cat([out_ids, next_tokens.view(B, 1)], dim=-1)
            if check_eos:
                finished |= next_tokens.view(-1) == gc.eos_token_id
                if bool(finished.all()):
                    break
            if out_ids.shape[1] >= max_len:
                break
            position_ids = lengths.view(B, 1).long()
            lengths = lengths + 1
            cur_mask = torch.cat(
                [cur_mask, (~finished).view(B, 1).to(cur_mask.dtype)], dim=-1
            )
            out = model(
                input_ids=next_tokens.view(B, 1),
                attention_mask=cur_mask,
                position_ids=position_ids,
                seq_ids=seq_ids,
                sampling_params=sampling_params,
            )
            next_tokens = self._next_token(out, gc)

        return out_ids

    @staticmethod
    def _next_token(out, gc: GenerationConfig) -> torch.Tensor:
        if out.tokens is not None:
            return out.tokens.view(-1)
        logits = out.logits[:, -1, :]
        if gc.do_sample:
            probs = torch.softmax(logits.float() / max(gc.temperature, 1e-6), dim=-1)
            if gc.top_k and gc.top_k > 0:
                v, i = probs.topk(gc.top_k, dim=-1)
                v = v / v.sum(-1, keepdim=True)
                pick = torch.multinomial(v, 1).squeeze(-1)
                return i.gather(1, pick.view(-1, 1)).view(-1)
            return torch.multinomial(probs, 1).view(-1)
        return logits.argmax(dim=-1).view(-1)
