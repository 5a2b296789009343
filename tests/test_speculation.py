"""Speculative decoding: greedy spec output must equal target-only greedy
output (the correctness invariant of speculative decoding), on CPU."""

import pytest
import torch

from neuronx_distributed_inference_amd.config import EngineConfig, OnDeviceSamplingConfig
from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
    LlamaForCausalLM,
    LlamaInferenceConfig,
)
from neuronx_distributed_inference_amd.modules.speculation import (
    SpeculativeGenerationAdapter,
)
from neuronx_distributed_inference_amd.utils.hf_adapter import (
    GenerationConfig,
    HuggingFaceGenerationAdapter,
)


def _build(hidden, layers, seed, spec_len=0, vocab=64):
    ec = EngineConfig(
        batch_size=2, max_batch_size=2, seq_len=96, max_length=96,
        torch_dtype="float32", tp_degree=1,
        on_device_sampling_config=OnDeviceSamplingConfig(top_k=1),
        speculation_length=spec_len,
    )
    cfg = LlamaInferenceConfig(
        neuron_config=ec, vocab_size=vocab, hidden_size=hidden,
        intermediate_size=hidden * 2, num_hidden_layers=layers,
        num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
        hidden_act="silu", tie_word_embeddings=False,
    )
    torch.manual_seed(seed)
    m = LlamaForCausalLM(cfg).eval()
    with torch.no_grad():
        for p in m.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    return m


@torch.inference_mode()
@pytest.mark.parametrize("spec_len", [3, 5])
def test_greedy_spec_matches_target_only(spec_len):
    target = _build(64, 2, seed=1, spec_len=spec_len)
    draft = _build(32, 1, seed=2)
    ids = torch.randint(0, 64, (2, 8))

    want = HuggingFaceGenerationAdapter(_build(64, 2, seed=1)).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=24))

    spec = SpeculativeGenerationAdapter(target, draft, speculation_length=spec_len)
    got = spec.generate(ids, generation_config=GenerationConfig(max_new_tokens=24))

    n = min(got.shape[1], want.shape[1])
    assert torch.equal(got[:, :n], want[:, :n]), (got.tolist(), want.tolist())


@torch.inference_mode()
def test_spec_with_identical_draft():
    """Draft == target: rows may stop at different lengths (per-row
    acceptance), but every generated prefix must match the target-only
    stream exactly."""
    target = _build(64, 2, seed=3, spec_len=4)
    draft = _build(64, 2, seed=3)
    ids = torch.randint(0, 64, (2, 6))
    spec = SpeculativeGenerationAdapter(target, draft, speculation_length=4)
    got = spec.generate(ids, generation_config=GenerationConfig(max_new_tokens=20))
    want = HuggingFaceGenerationAdapter(_build(64, 2, seed=3)).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=20))
    for i in range(2):
        row = got[i]
        # non-pad length (prompt has no zeros in this seed beyond real pads)
        n = int((row != 0).float().flip(0).argmax())
        n = row.shape[0] - n
        n = min(n, want.shape[1])
        assert n >= ids.shape[1] + 10  # made real progress
        assert torch.equal(row[:n], want[i, :n]), (row.tolist(), want[i].tolist())


@torch.inference_mode()
def test_spec_through_submodel_wrappers():
    """Speculation routed through the speculation_model wrapper (bucketed)
    must agree with the direct path."""
    target = _build(64, 2, seed=4, spec_len=3)
    draft = _build(32, 1, seed=5)
    ids = torch.randint(0, 64, (2, 8))
    direct = SpeculativeGenerationAdapter(target, draft, 3).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=16))

    target2 = _build(64, 2, seed=4, spec_len=3)
    draft2 = _build(32, 1, seed=5)
    target2.enable_submodels()
    draft2.enable_submodels()
    assert target2.speculation_model is not None
    wrapped = SpeculativeGenerationAdapter(target2, draft2, 3).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=16))
    n = min(direct.shape[1], wrapped.shape[1])
    assert torch.equal(direct[:, :n], wrapped[:, :n])


@torch.inference_mode()
def test_fused_spec_matches_target_only():
    from neuronx_distributed_inference_amd.modules.speculation import (
        FusedSpeculativeAdapter,
    )

    target = _build(64, 2, seed=1, spec_len=4)
    draft = _build(32, 1, seed=2)
    ids = torch.randint(0, 64, (2, 8))
    want = HuggingFaceGenerationAdapter(_build(64, 2, seed=1)).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=24))
    got = FusedSpeculativeAdapter(target, draft, 4, use_graphs=False).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=24))
    n = min(got.shape[1], want.shape[1])
    assert torch.equal(got[:, :n], want[:, :n]), (got.tolist(), want.tolist())


@torch.inference_mode()
def test_medusa_matches_target_only():
    """Chain-medusa (random heads => mostly rejected) must still equal the
    target-only greedy stream."""
    from neuronx_distributed_inference_amd.modules.medusa import (
        MedusaGenerationAdapter,
        MedusaHeads,
    )

    target = _build(64, 2, seed=1, spec_len=3)
    torch.manual_seed(9)
    heads = MedusaHeads(64, 64, num_heads=3, dtype=torch.float32)
    with torch.no_grad():
        for p in heads.parameters():
            p.normal_(0, 0.05)
    ids = torch.randint(0, 64, (2, 8))
    want = HuggingFaceGenerationAdapter(_build(64, 2, seed=1)).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=20))
    got = MedusaGenerationAdapter(target, heads).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=20))
    for i in range(2):
        row = got[i]
        n = int((row != 0).float().flip(0).argmax())
        n = row.shape[0] - n
        n = min(n, want.shape[1])
        assert n >= ids.shape[1] + 8
        assert torch.equal(row[:n], want[i, :n]), (row.tolist(), want[i].tolist())


@torch.inference_mode()
def test_eagle_matches_target_only():
    """EAGLE chain (random draft => mostly rejected) must equal the
    target-only greedy stream."""
    from neuronx_distributed_inference_amd.modules.eagle import (
        EagleDraftModel,
        EagleGenerationAdapter,
    )

    target = _build(64, 2, seed=1, spec_len=3)
    draft_lm = _build(64, 1, seed=7)
    torch.manual_seed(8)
    draft = EagleDraftModel(draft_lm.model, hidden_size=64, dtype=torch.float32)
    with torch.no_grad():
        draft.fc.weight.normal_(0, 0.05)
    ids = torch.randint(0, 64, (2, 8))
    want = HuggingFaceGenerationAdapter(_build(64, 2, seed=1)).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=20))
    got = EagleGenerationAdapter(target, draft, speculation_length=3).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=20))
    for i in range(2):
        row = got[i]
        n = int((row != 0).float().flip(0).argmax())
        n = row.shape[0] - n
        n = min(n, want.shape[1])
        assert n >= ids.shape[1] + 8
        assert torch.equal(row[:n], want[i, :n]), (row.tolist(), want[i].tolist())
