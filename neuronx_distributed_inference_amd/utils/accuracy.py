"""Accuracy harness.

Analog of the reference's `utils/accuracy.py` (check_accuracy :244-343 —
exact token match vs a CPU/HF oracle; check_accuracy_logits :478 —
per-position logit tolerance with divergence-index reporting via
LogitMatchingValidationError).
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional

import torch

from neuronx_distributed_inference_amd.utils.hf_adapter import (
    GenerationConfig,
    HuggingFaceGenerationAdapter,
)

logger = logging.getLogger("nxdi_amd")


class AccuracyValidationError(AssertionError):
    pass


class LogitMatchingValidationError(AccuracyValidationError):
    def __init__(self, message, divergence_index: int, results: Dict):
        super().__init__(message)
        self.divergence_index = divergence_index
        self.results = results


def check_accuracy(
    model,
    input_ids: torch.Tensor,
    attention_mask: Optional[torch.Tensor] = None,
    expected_token_ids: Optional[torch.Tensor] = None,
    hf_model=None,
    max_new_tokens: int = 32,
    generation_config: Optional[GenerationConfig] = None,
    divergence_tolerance: int = 0,
) -> torch.Tensor:
    """Greedy-generate and require exact token match with the oracle
    (`expected_token_ids` or an HF model run on the same inputs)."""
    gc = generation_config or GenerationConfig(max_new_tokens=max_new_tokens)
    got = HuggingFaceGenerationAdapter(model).generate(
        input_ids, attention_mask=attention_mask, generation_config=gc
    )
    got = got.cpu()
    S0 = input_ids.shape[1]
    if expected_token_ids is not None:
        expected_token_ids = expected_token_ids.cpu()
        n = min(got.shape[1], expected_token_ids.shape[1])
        mism = (got[:, :n] != expected_token_ids[:, :n]).sum().item()
        if mism > divergence_tolerance:
            first = int((got[:, :n] != expected_token_ids[:, :n]).any(0).float().argmax())
            raise AccuracyValidationError(
                f"token mismatch: {mism} positions differ (first at {first});\n"
                f"got      {got[:, :n].tolist()}\n"
                f"expected {expected_token_ids[:, :n].tolist()}"
            )
        logger.info("check_accuracy passed (%d tokens)", n)
        return got

    # HF oracle: generate each row UNPADDED (HF generate assumes left padding
    # for batches; per-row solo runs are the honest reference for our
    # right-padded continuous-batching semantics)
    assert hf_model is not None, "need expected_token_ids or hf_model"
    mask = attention_mask.cpu() if attention_mask is not None else torch.ones_like(input_ids.cpu())
    ids_cpu = input_ids.cpu()
    for i in range(ids_cpu.shape[0]):
        real = ids_cpu[i][mask[i].bool()].view(1, -1)
        with torch.no_grad():
            gen = hf_model.generate(real, max_new_tokens=gc.max_new_tokens, do_sample=False)
        exp_new = gen[0, real.shape[1]:]
        got_new = got[i, S0:]
        n = min(len(exp_new), len(got_new))
        if not torch.equal(got_new[:n], exp_new[:n]):
            raise AccuracyValidationError(
                f"row {i} token mismatch:\n got      {got_new[:n].tolist()}\n"
                f" expected {exp_new[:n].tolist()}"
            )
    logger.info("check_accuracy passed (%d rows)", ids_cpu.shape[0])
    return got


def check_accuracy_logits(
    model,
    input_ids: torch.Tensor,
    golden_logits: torch.Tensor,
    attention_mask: Optional[torch.Tensor] = None,
    atol: float = 1e-2,
    rtol: float = 1e-2,
    tol_map: Optional[Dict[int, tuple]] = None,
) -> Dict:
    """Compare the model's last-token logits against goldens position by
    position during greedy decode; on failure raise with the divergence
    index (ref check_accuracy_logits + LogitMatchingValidationError)."""
    model.reset()
    B, S0 = input_ids.shape
    n_positions = golden_logits.shape[0]
    results = {"max_abs_diff": [], "checked": 0}
    if attention_mask is None:
        attention_mask = torch.ones_like(input_ids)
    out = model(input_ids=input_ids, attention_mask=attention_mask)
    lengths = attention_mask.sum(-1)
    cur_mask = attention_mask
    for i in range(n_positions):
        logits = out.logits[:, -1, :].float()
        gold = golden_logits[i].float()
        a, r = (tol_map or {}).get(i, (atol, rtol))
        diff = (logits - gold).abs().max().item()
        results["max_abs_diff"].append(diff)
        if not torch.allclose(logits, gold, atol=a, rtol=r):
            raise LogitMatchingValidationError(
                f"logits diverge at position {i}: max abs diff {diff:.5f} "
                f"(atol={a}, rtol={r})", i, results,
            )
        results["checked"] += 1
        if i == n_positions - 1:
            break
        tok = logits.argmax(-1).view(B, 1)
        pos = lengths.view(B, 1).long()
        lengths = lengths + 1
        cur_mask = torch.cat([cur_mask, torch.ones(B, 1, dtype=cur_mask.dtype,
                                                   device=cur_mask.device)], dim=-1)
        out = model(input_ids=tok, attention_mask=cur_mask, position_ids=pos)
    return results
