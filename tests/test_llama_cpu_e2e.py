"""Tier-1/oracle test: tiny random-weight Llama on CPU vs HF transformers.

Mirrors the reference's accuracy strategy (`utils/accuracy.py`
check_accuracy :244-343 = exact token match vs HF CPU generate;
check_accuracy_logits :478 = per-token logit tolerance).
"""

import pytest
import torch

from neuronx_distributed_inference_amd.models.llama.modeling_llama import LlamaForCausalLM
from neuronx_distributed_inference_amd.utils.hf_adapter import (
    GenerationConfig,
    HuggingFaceGenerationAdapter,
)


@pytest.fixture
def models(tiny_llama_config):
    from transformers import LlamaConfig, LlamaForCausalLM as HFLlama

    torch.manual_seed(0)
    cfg = tiny_llama_config
    hf_cfg = LlamaConfig(
        vocab_size=cfg.vocab_size, hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        max_position_embeddings=256, rms_norm_eps=cfg.rms_norm_eps,
        tie_word_embeddings=False,
    )
    hf = HFLlama(hf_cfg).eval()
    ours = LlamaForCausalLM(cfg).eval()
    sd = LlamaForCausalLM.convert_hf_to_native_state_dict(hf.state_dict(), cfg)
    missing, unexpected = ours.model.load_state_dict(sd, strict=False)
    assert not [m for m in missing if not m.startswith("kv_mgr")]
    assert not unexpected
    return ours, hf


@torch.inference_mode()
def test_prefill_logits_match_hf(models):
    ours, hf = models
    ids = torch.randint(0, 128, (2, 12))
    mask = torch.ones_like(ids)
    ref = hf(ids, attention_mask=mask).logits
    out = ours(input_ids=ids, attention_mask=mask)
    # our prefill returns only the last position's logits
    torch.testing.assert_close(out.logits[:, -1, :], ref[:, -1, :], atol=1e-5, rtol=1e-5)


@torch.inference_mode()
def test_greedy_generation_matches_hf(models):
    ours, hf = models
    torch.manual_seed(1)
    ids = torch.randint(0, 128, (2, 10))
    mask = torch.ones_like(ids)
    adap = HuggingFaceGenerationAdapter(ours)
    got = adap.generate(ids, attention_mask=mask, generation_config=GenerationConfig(max_new_tokens=16))
    want = hf.generate(ids, attention_mask=mask, max_new_tokens=16, do_sample=False)
    assert torch.equal(got, want)


@torch.inference_mode()
def test_decode_after_prefill_uses_kv_cache(models):
    ours, _ = models
    ids = torch.randint(0, 128, (2, 8))
    mask = torch.ones_like(ids)
    ours.reset()
    out = ours(input_ids=ids, attention_mask=mask)
    assert ours.kv_cache_populated
    # cache rows for positions 0..7 must be non-zero
    k0 = ours.model.kv_mgr.k_caches[0]
    assert k0[:, :, :8, :].abs().sum() > 0
    assert k0[:, :, 8:, :].abs().sum() == 0


@torch.inference_mode()
def test_ragged_batch_generation(models):
    """Right-padded ragged batch: each row's continuation must equal the
    row generated alone (continuous-batching semantics)."""
    ours, hf = models
    torch.manual_seed(2)
    ids = torch.randint(0, 128, (2, 10))
    mask = torch.ones_like(ids)
    mask[0, 7:] = 0  # row 0 has only 7 real tokens
    adap = HuggingFaceGenerationAdapter(ours)
    got = adap.generate(ids, attention_mask=mask, generation_config=GenerationConfig(max_new_tokens=6))
    solo = adap.generate(
        ids[0:1, :7], attention_mask=torch.ones(1, 7, dtype=mask.dtype),
        generation_config=GenerationConfig(max_new_tokens=6),
    )
    assert torch.equal(got[0, 10:], solo[0, 7:])
