"""Model-hub oracle tests: every model family vs its HF transformers
implementation on a tiny random-weight config (CPU fp32, exact greedy
token match — the reference's check_accuracy strategy)."""

import pytest
import torch

from neuronx_distributed_inference_amd.config import EngineConfig
from neuronx_distributed_inference_amd.utils.hf_adapter import (
    GenerationConfig,
    HuggingFaceGenerationAdapter,
)

COMMON = dict(vocab_size=128, hidden_size=64, intermediate_size=128,
              num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
              max_position_embeddings=128, tie_word_embeddings=False)


def _engine():
    return EngineConfig(batch_size=2, max_batch_size=2, seq_len=48, max_length=48,
                        torch_dtype="float32", tp_degree=1)


def _check_match(ours, hf, n_new=10, vocab=128):
    torch.manual_seed(5)
    ids = torch.randint(0, vocab, (2, 8))
    mask = torch.ones_like(ids)
    got = HuggingFaceGenerationAdapter(ours).generate(
        ids, attention_mask=mask, generation_config=GenerationConfig(max_new_tokens=n_new))
    want = hf.generate(ids, attention_mask=mask, max_new_tokens=n_new, do_sample=False)
    assert torch.equal(got, want), (got.tolist(), want.tolist())


@torch.inference_mode()
def test_mistral():
    from transformers import MistralConfig, MistralForCausalLM as HF

    from neuronx_distributed_inference_amd.models.mistral import (
        MistralForCausalLM, MistralInferenceConfig,
    )

    torch.manual_seed(0)
    hf_cfg = MistralConfig(**COMMON, rms_norm_eps=1e-5, sliding_window=None)
    hf = HF(hf_cfg).eval()
    cfg = MistralInferenceConfig(neuron_config=_engine(), **hf_cfg.to_dict())
    ours = MistralForCausalLM(cfg).eval()
    sd = MistralForCausalLM.convert_hf_to_native_state_dict(hf.state_dict(), cfg)
    missing, unexpected = ours.model.load_state_dict(sd, strict=False)
    assert not [m for m in missing if not m.startswith("kv_mgr")] and not unexpected
    _check_match(ours, hf)


@torch.inference_mode()
def test_qwen2():
    from transformers import Qwen2Config, Qwen2ForCausalLM as HF

    from neuronx_distributed_inference_amd.models.qwen2 import (
        Qwen2ForCausalLM, Qwen2InferenceConfig,
    )

    torch.manual_seed(0)
    hf_cfg = Qwen2Config(**COMMON, rms_norm_eps=1e-5)
    hf = HF(hf_cfg).eval()
    cfg = Qwen2InferenceConfig(neuron_config=_engine(), **hf_cfg.to_dict())
    ours = Qwen2ForCausalLM(cfg).eval()
    sd = Qwen2ForCausalLM.convert_hf_to_native_state_dict(hf.state_dict(), cfg)
    missing, unexpected = ours.model.load_state_dict(sd, strict=False)
    assert not [m for m in missing if not m.startswith("kv_mgr")] and not unexpected
    _check_match(ours, hf)


@torch.inference_mode()
def test_qwen3():
    from transformers import Qwen3Config, Qwen3ForCausalLM as HF

    from neuronx_distributed_inference_amd.models.qwen3 import (
        Qwen3ForCausalLM, Qwen3InferenceConfig,
    )

    torch.manual_seed(0)
    hf_cfg = Qwen3Config(**COMMON, rms_norm_eps=1e-5, head_dim=16)
    hf = HF(hf_cfg).eval()
    cfg = Qwen3InferenceConfig(neuron_config=_engine(), **hf_cfg.to_dict())
    ours = Qwen3ForCausalLM(cfg).eval()
    sd = Qwen3ForCausalLM.convert_hf_to_native_state_dict(hf.state_dict(), cfg)
    missing, unexpected = ours.model.load_state_dict(sd, strict=False)
    assert not [m for m in missing if not m.startswith("kv_mgr")] and not unexpected
    _check_match(ours, hf)


@torch.inference_mode()
def test_mixtral():
    from transformers import MixtralConfig, MixtralForCausalLM as HF

    from neuronx_distributed_inference_amd.models.mixtral import (
        MixtralForCausalLM, MixtralInferenceConfig,
    )

    torch.manual_seed(0)
    hf_cfg = MixtralConfig(**COMMON, rms_norm_eps=1e-5, num_local_experts=4,
                           num_experts_per_tok=2)
    hf = HF(hf_cfg).eval()
    cfg = MixtralInferenceConfig(neuron_config=_engine(), **hf_cfg.to_dict())
    ours = MixtralForCausalLM(cfg).eval()
    sd = MixtralForCausalLM.convert_hf_to_native_state_dict(hf.state_dict(), cfg)
    missing, unexpected = ours.model.load_state_dict(sd, strict=False)
    assert not [m for m in missing if not m.startswith("kv_mgr")] and not unexpected
    _check_match(ours, hf)


@torch.inference_mode()
def test_qwen3_moe():
    from transformers import Qwen3MoeConfig, Qwen3MoeForCausalLM as HF

    from neuronx_distributed_inference_amd.models.qwen3_moe import (
        Qwen3MoeForCausalLM, Qwen3MoeInferenceConfig,
    )

    torch.manual_seed(0)
    hf_cfg = Qwen3MoeConfig(**COMMON, rms_norm_eps=1e-5, head_dim=16,
                            num_experts=4, num_experts_per_tok=2,
                            moe_intermediate_size=64, decoder_sparse_step=1,
                            norm_topk_prob=True)
    hf = HF(hf_cfg).eval()
    cfg = Qwen3MoeInferenceConfig(neuron_config=_engine(), **hf_cfg.to_dict())
    ours = Qwen3MoeForCausalLM(cfg).eval()
    sd = Qwen3MoeForCausalLM.convert_hf_to_native_state_dict(hf.state_dict(), cfg)
    missing, unexpected = ours.model.load_state_dict(sd, strict=False)
    assert not [m for m in missing if not m.startswith("kv_mgr")] and not unexpected
    _check_match(ours, hf)


@torch.inference_mode()
def test_dbrx():
    from transformers import DbrxConfig, DbrxForCausalLM as HF

    from neuronx_distributed_inference_amd.models.dbrx import (
        DbrxForCausalLM, DbrxInferenceConfig,
    )

    torch.manual_seed(0)
    hf_cfg = DbrxConfig(
        d_model=64, n_heads=4, n_layers=2, max_seq_len=128, vocab_size=128,
        attn_config={"kv_n_heads": 2, "clip_qkv": 8.0, "rope_theta": 10000.0},
        ffn_config={"ffn_hidden_size": 96, "moe_num_experts": 4, "moe_top_k": 2},
        resid_pdrop=0.0, emb_pdrop=0.0,
    )
    hf = HF(hf_cfg).eval()
    d = hf_cfg.to_dict()
    cfg = DbrxInferenceConfig(neuron_config=_engine(), **d)
    ours = DbrxForCausalLM(cfg).eval()
    sd = DbrxForCausalLM.convert_hf_to_native_state_dict(hf.state_dict(), cfg)
    missing, unexpected = ours.model.load_state_dict(sd, strict=False)
    assert not [m for m in missing if not m.startswith("kv_mgr")], missing
    assert not unexpected, unexpected
    _check_match(ours, hf)


def test_registry_resolves_all():
    from neuronx_distributed_inference_amd.registry import (
        available_model_types,
        get_model_classes,
    )

    for mt in available_model_types():
        cfg_cls, model_cls = get_model_classes(mt)
        assert cfg_cls is not None and model_cls is not None


@torch.inference_mode()
def test_gemma3():
    from transformers import Gemma3TextConfig, Gemma3ForCausalLM as HF

    from neuronx_distributed_inference_amd.models.gemma3 import (
        Gemma3ForCausalLM, Gemma3InferenceConfig,
    )

    torch.manual_seed(0)
    hf_cfg = Gemma3TextConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, sliding_window=8, sliding_window_pattern=2,
        max_position_embeddings=128, rms_norm_eps=1e-6, tie_word_embeddings=True,
    )
    hf = HF(hf_cfg).eval()
    d = hf_cfg.to_dict()
    cfg = Gemma3InferenceConfig(neuron_config=_engine(), **d)
    ours = Gemma3ForCausalLM(cfg).eval()
    sd = Gemma3ForCausalLM.convert_hf_to_native_state_dict(hf.state_dict(), cfg)
    missing, unexpected = ours.model.load_state_dict(sd, strict=False)
    assert not [m for m in missing if not m.startswith("kv_mgr") and ".rotary." not in m], missing
    assert not unexpected, unexpected
    _check_match(ours, hf)


@torch.inference_mode()
def test_gpt_oss():
    from transformers import GptOssConfig, GptOssForCausalLM as HF

    from neuronx_distributed_inference_amd.models.gpt_oss import (
        GptOssForCausalLM, GptOssInferenceConfig,
    )

    torch.manual_seed(0)
    hf_cfg = GptOssConfig(
        vocab_size=128, hidden_size=64, intermediate_size=64,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, num_local_experts=4, num_experts_per_tok=2,
        sliding_window=8, max_position_embeddings=128,
    )
    hf = HF(hf_cfg).eval()
    d = hf_cfg.to_dict()
    cfg = GptOssInferenceConfig(neuron_config=_engine(), **d)
    ours = GptOssForCausalLM(cfg).eval()
    sd = GptOssForCausalLM.convert_hf_to_native_state_dict(hf.state_dict(), cfg)
    missing, unexpected = ours.model.load_state_dict(sd, strict=False)
    assert not [m for m in missing if not m.startswith("kv_mgr")], missing
    assert not unexpected, unexpected
    _check_match(ours, hf)


@torch.inference_mode()
def test_deepseek_v3():
    from transformers import DeepseekV3Config, DeepseekV3ForCausalLM as HF

    from neuronx_distributed_inference_amd.models.deepseek import (
        DeepseekV3ForCausalLM, DeepseekV3InferenceConfig,
    )

    torch.manual_seed(0)
    hf_cfg = DeepseekV3Config(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        moe_intermediate_size=32, num_hidden_layers=3,
        num_attention_heads=4, num_key_value_heads=4,
        n_routed_experts=4, num_experts_per_tok=2, n_shared_experts=1,
        n_group=2, topk_group=1, routed_scaling_factor=2.5,
        q_lora_rank=32, kv_lora_rank=16, qk_rope_head_dim=8,
        qk_nope_head_dim=16, v_head_dim=16, first_k_dense_replace=1,
        max_position_embeddings=128,
    )
    hf = HF(hf_cfg).eval()
    d = hf_cfg.to_dict()
    cfg = DeepseekV3InferenceConfig(neuron_config=_engine(), **d)
    ours = DeepseekV3ForCausalLM(cfg).eval()
    sd = DeepseekV3ForCausalLM.convert_hf_to_native_state_dict(hf.state_dict(), cfg)
    missing, unexpected = ours.model.load_state_dict(sd, strict=False)
    assert not [m for m in missing if not m.startswith("kv_mgr")], missing
    assert not unexpected, unexpected
    _check_match(ours, hf)


@torch.inference_mode()
def test_llama4():
    """Llama4 text: interleaved rope, NoPE + temperature tuning, chunked
    local attention, sigmoid top-1 input-scaled MoE + shared expert."""
    from transformers.models.llama4 import Llama4TextConfig
    from transformers.models.llama4.modeling_llama4 import Llama4ForCausalLM as HF

    from neuronx_distributed_inference_amd.models.llama4 import (
        Llama4ForCausalLM, Llama4InferenceConfig,
    )

    torch.manual_seed(0)
    hf_cfg = Llama4TextConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        intermediate_size_mlp=160, num_hidden_layers=4,
        num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128, tie_word_embeddings=False,
        num_local_experts=4, num_experts_per_tok=1,
        interleave_moe_layer_step=2, no_rope_layer_interval=2,
        attention_chunk_size=8, use_qk_norm=True,
        attn_temperature_tuning=True, floor_scale=8, attn_scale=0.1,
        rms_norm_eps=1e-5,
    )
    hf = HF(hf_cfg).eval()
    cfg = Llama4InferenceConfig(neuron_config=_engine(), **hf_cfg.to_dict())
    ours = Llama4ForCausalLM(cfg).eval()
    sd = Llama4ForCausalLM.convert_hf_to_native_state_dict(hf.state_dict(), cfg)
    missing, unexpected = ours.model.load_state_dict(sd, strict=False)
    assert not [m for m in missing if not m.startswith("kv_mgr")], missing
    assert not unexpected, unexpected
    _check_match(ours, hf, n_new=12)


@torch.inference_mode()
def test_whisper():
    """Whisper encoder-decoder: greedy tokens must match a stepwise
    teacher-forced HF forward (cache-free oracle)."""
    from transformers import WhisperConfig
    from transformers import WhisperForConditionalGeneration as HF

    from neuronx_distributed_inference_amd.models.whisper import (
        WhisperForConditionalGeneration, WhisperInferenceConfig,
    )

    torch.manual_seed(0)
    hf_cfg = WhisperConfig(
        vocab_size=100, num_mel_bins=8, d_model=32, encoder_layers=2,
        decoder_layers=2, encoder_attention_heads=4, decoder_attention_heads=4,
        encoder_ffn_dim=64, decoder_ffn_dim=64, max_source_positions=24,
        max_target_positions=40, decoder_start_token_id=1, eos_token_id=2,
        pad_token_id=0, bos_token_id=None,
    )
    hf = HF(hf_cfg).eval()
    cfg = WhisperInferenceConfig(neuron_config=_engine(), **hf_cfg.to_dict())
    ours = WhisperForConditionalGeneration(cfg).eval()
    sd = WhisperForConditionalGeneration.convert_hf_to_native_state_dict(
        hf.state_dict(), cfg)
    missing, unexpected = ours.load_state_dict(sd, strict=False)
    assert not missing, missing
    assert not unexpected, unexpected

    feats = torch.randn(2, 8, 48)
    n_new = 10
    # oracle: stepwise cache-free HF greedy
    ids = torch.full((2, 1), 1, dtype=torch.long)
    for _ in range(n_new):
        lg = hf(input_features=feats, decoder_input_ids=ids).logits
        ids = torch.cat([ids, lg[:, -1].argmax(-1, keepdim=True)], dim=1)
    got = ours.generate(feats, max_new_tokens=n_new)
    assert torch.equal(got[:, : ids.shape[1]], ids), (got.tolist(), ids.tolist())


@torch.inference_mode()
def test_mllama():
    """Mllama (Llama-3.2 Vision): tiled ViT + gated cross-attention text
    model; greedy tokens vs stepwise cache-free HF forward with an image."""
    from transformers import MllamaConfig
    from transformers import MllamaForConditionalGeneration as HF
    from transformers.models.mllama import MllamaTextConfig, MllamaVisionConfig

    from neuronx_distributed_inference_amd.models.mllama import (
        MllamaForConditionalGeneration, MllamaInferenceConfig,
    )

    torch.manual_seed(0)
    tc = MllamaTextConfig(
        vocab_size=120, hidden_size=32, intermediate_size=64,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        cross_attention_layers=[1, 3], rope_theta=10000.0,
        max_position_embeddings=64, bos_token_id=3, eos_token_id=2,
        pad_token_id=0, rope_scaling=None, tie_word_embeddings=False)
    vc = MllamaVisionConfig(
        hidden_size=32, intermediate_size=64, num_hidden_layers=3,
        num_global_layers=1, attention_heads=4, image_size=32, patch_size=16,
        max_num_tiles=2, supported_aspect_ratios=[[1, 1], [1, 2], [2, 1]],
        intermediate_layers_indices=[1, 2], vision_output_dim=96, num_channels=3)
    hf_cfg = MllamaConfig(vision_config=vc.to_dict(), text_config=tc.to_dict())
    hf = HF(hf_cfg).eval()

    ec = EngineConfig(batch_size=1, max_batch_size=1, seq_len=48, max_length=48,
                      torch_dtype="float32", tp_degree=1)
    cfg = MllamaInferenceConfig(neuron_config=ec,
                                text_config=tc.to_dict(),
                                vision_config=vc.to_dict())
    ours = MllamaForConditionalGeneration(cfg).eval()
    sd = MllamaForConditionalGeneration.convert_hf_to_native_state_dict(
        hf.state_dict(), cfg)
    missing, unexpected = ours.load_state_dict(sd, strict=False)
    real_missing = [m for m in missing if ".kv_mgr." not in m and not m.startswith("kv_mgr")]
    assert not real_missing, real_missing
    assert not unexpected, unexpected

    px = torch.randn(1, 1, 2, 3, 32, 32)
    ar_ids = torch.tensor([[2]])           # aspect ratio [1,2] -> id 2
    ar_mask = torch.tensor([[[1, 1]]])
    ids = torch.randint(4, 120, (1, 6))
    n_new = 8
    ref_ids = ids.clone()
    for _ in range(n_new):
        lg = hf(input_ids=ref_ids, pixel_values=px, aspect_ratio_ids=ar_ids,
                aspect_ratio_mask=ar_mask, use_cache=False).logits
        ref_ids = torch.cat([ref_ids, lg[:, -1].argmax(-1, keepdim=True)], dim=1)
    got = ours.generate(ids, pixel_values=px, aspect_ratio_ids=ar_ids,
                        aspect_ratio_mask=ar_mask, max_new_tokens=n_new)
    assert torch.equal(got[:, : ref_ids.shape[1]], ref_ids), (
        got.tolist(), ref_ids.tolist())

    # text-only path must also match (cross layers skipped)
    ref_ids = ids.clone()
    for _ in range(4):
        lg = hf(input_ids=ref_ids, use_cache=False).logits
        ref_ids = torch.cat([ref_ids, lg[:, -1].argmax(-1, keepdim=True)], dim=1)
    got = ours.generate(ids, max_new_tokens=4)
    assert torch.equal(got[:, : ref_ids.shape[1]], ref_ids)

    # per-token tile masks (cross_attention_mask), incl. a fully-masked row
    # and a partially-masked tile; HF repeats the last row per new token
    xm = torch.ones(1, 6, 1, 2, dtype=torch.long)
    xm[0, 0] = 0          # row 0: no vision at all (full-row masked out)
    xm[0, 2, 0, 1] = 0    # row 2: second tile hidden
    ref_ids, ref_xm = ids.clone(), xm.clone()
    for _ in range(6):
        lg = hf(input_ids=ref_ids, pixel_values=px, aspect_ratio_ids=ar_ids,
                aspect_ratio_mask=ar_mask, cross_attention_mask=ref_xm,
                use_cache=False).logits
        ref_ids = torch.cat([ref_ids, lg[:, -1].argmax(-1, keepdim=True)], dim=1)
        ref_xm = torch.cat([ref_xm, ref_xm[:, -1:]], dim=1)
    got = ours.generate(ids, pixel_values=px, aspect_ratio_ids=ar_ids,
                        aspect_ratio_mask=ar_mask, cross_attention_mask=xm,
                        max_new_tokens=6)
    assert torch.equal(got[:, : ref_ids.shape[1]], ref_ids), (
        got.tolist(), ref_ids.tolist())


@torch.inference_mode()
def test_pixtral():
    """Pixtral/llava: 2D-rope block-diagonal vision encoder + inline image
    splice into the Mistral text model; greedy vs stepwise HF."""
    from transformers import LlavaConfig
    from transformers import LlavaForConditionalGeneration as HF
    from transformers.models.mistral import MistralConfig
    from transformers.models.pixtral import PixtralVisionConfig

    from neuronx_distributed_inference_amd.models.pixtral import (
        PixtralForConditionalGeneration, PixtralInferenceConfig,
    )

    torch.manual_seed(0)
    vc = PixtralVisionConfig(
        hidden_size=32, intermediate_size=64, num_hidden_layers=2,
        num_attention_heads=4, image_size=32, patch_size=8, num_channels=3,
        rope_theta=10000.0)
    tc = MistralConfig(
        vocab_size=140, hidden_size=48, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rms_norm_eps=1e-5, sliding_window=None,
        tie_word_embeddings=False)
    hf_cfg = LlavaConfig(vision_config=vc.to_dict(), text_config=tc.to_dict(),
                         image_token_id=139, projector_hidden_act="gelu",
                         vision_feature_layer=-1,
                         vision_feature_select_strategy="full")
    hf = HF(hf_cfg).eval()

    ec = EngineConfig(batch_size=1, max_batch_size=1, seq_len=64, max_length=64,
                      torch_dtype="float32", tp_degree=1)
    cfg = PixtralInferenceConfig(
        neuron_config=ec, text_config=tc.to_dict(), vision_config=vc.to_dict(),
        image_token_id=139, projector_hidden_act="gelu",
        multimodal_projector_bias=True)
    ours = PixtralForConditionalGeneration(cfg).eval()
    sd = PixtralForConditionalGeneration.convert_hf_to_native_state_dict(
        hf.state_dict(), cfg)
    missing, unexpected = ours.load_state_dict(sd, strict=False)
    real_missing = [m for m in missing if ".kv_mgr." not in m]
    assert not real_missing, real_missing
    assert not unexpected, unexpected

    px = torch.randn(1, 3, 32, 32)           # 4x4 = 16 patches
    ids = torch.randint(0, 139, (1, 22))
    ids[0, 2:18] = 139                        # 16 image placeholders
    n_new = 8
    ref_ids = ids.clone()
    for _ in range(n_new):
        lg = hf(input_ids=ref_ids, pixel_values=px, use_cache=False).logits
        ref_ids = torch.cat([ref_ids, lg[:, -1].argmax(-1, keepdim=True)], dim=1)
    got = ours.generate(ids, pixel_values=px, max_new_tokens=n_new)
    assert torch.equal(got[:, : ref_ids.shape[1]], ref_ids), (
        got.tolist(), ref_ids.tolist())


@torch.inference_mode()
def test_qwen2_vl():
    """Qwen2-VL: Conv3d patchify ViT + m-rope text model; greedy vs
    stepwise HF with an image and text-only."""
    from transformers import Qwen2VLConfig
    from transformers import Qwen2VLForConditionalGeneration as HF

    from neuronx_distributed_inference_amd.models.qwen2_vl import (
        Qwen2VLForConditionalGeneration, Qwen2VLInferenceConfig,
    )

    torch.manual_seed(0)
    vc = dict(depth=2, embed_dim=32, hidden_size=48, num_heads=4,
              in_channels=3, patch_size=4, temporal_patch_size=2,
              spatial_merge_size=2, mlp_ratio=2, hidden_act="quick_gelu")
    tc = dict(vocab_size=160, hidden_size=48, intermediate_size=96,
              num_hidden_layers=2, num_attention_heads=4,
              num_key_value_heads=2, max_position_embeddings=128,
              rms_norm_eps=1e-5, rope_theta=10000.0, tie_word_embeddings=False,
              rope_scaling={"rope_type": "default",
                            "mrope_section": [2, 2, 2]})
    hf_cfg = Qwen2VLConfig(vision_config=vc, text_config=tc,
                           image_token_id=159, vision_start_token_id=158,
                           vision_end_token_id=157, video_token_id=156)
    hf = HF(hf_cfg).eval()

    ec = EngineConfig(batch_size=1, max_batch_size=1, seq_len=64, max_length=64,
                      torch_dtype="float32", tp_degree=1)
    cfg = Qwen2VLInferenceConfig(
        neuron_config=ec, text_config=dict(tc), vision_config=dict(vc),
        image_token_id=159, video_token_id=156)
    ours = Qwen2VLForConditionalGeneration(cfg).eval()
    sd = Qwen2VLForConditionalGeneration.convert_hf_to_native_state_dict(
        hf.state_dict(), cfg)
    missing, unexpected = ours.load_state_dict(sd, strict=False)
    real_missing = [m for m in missing if ".kv_mgr." not in m]
    assert not real_missing, real_missing
    assert not unexpected, unexpected

    # one image: grid (t=1, h=4, w=4) -> 16 patches -> 4 merged tokens
    grid = torch.tensor([[1, 4, 4]])
    px = torch.randn(16, 3 * 2 * 4 * 4)
    ids = torch.randint(0, 156, (1, 14))
    ids[0, 3:7] = 159                     # 4 merged-image placeholders
    n_new = 8
    ref_ids = ids.clone()
    for _ in range(n_new):
        lg = hf(input_ids=ref_ids, pixel_values=px, image_grid_thw=grid,
                mm_token_type_ids=(ref_ids == 159).int(),
                use_cache=False).logits
        ref_ids = torch.cat([ref_ids, lg[:, -1].argmax(-1, keepdim=True)], dim=1)
    got = ours.generate(ids, pixel_values=px, image_grid_thw=grid,
                        max_new_tokens=n_new)
    assert torch.equal(got[:, : ref_ids.shape[1]], ref_ids), (
        got.tolist(), ref_ids.tolist())

    # text-only
    ids2 = torch.randint(0, 156, (1, 10))
    ref_ids = ids2.clone()
    for _ in range(4):
        lg = hf(input_ids=ref_ids, use_cache=False).logits
        ref_ids = torch.cat([ref_ids, lg[:, -1].argmax(-1, keepdim=True)], dim=1)
    got = ours.generate(ids2, max_new_tokens=4)
    assert torch.equal(got[:, : ref_ids.shape[1]], ref_ids)

    # video (t=2): temporal m-rope axis advances per frame; 2*4*4 patches
    # -> 8 merged video placeholders, plus an image in the same prompt
    vgrid = torch.tensor([[2, 4, 4]])
    vpx = torch.randn(32, 3 * 2 * 4 * 4)
    ids3 = torch.randint(0, 156, (1, 20))
    ids3[0, 2:10] = 156                   # 8 video placeholders
    ids3[0, 12:16] = 159                  # 4 image placeholders
    mm_tt = torch.zeros_like(ids3)
    mm_tt[ids3 == 159] = 1
    mm_tt[ids3 == 156] = 2
    ref_ids = ids3.clone()
    for _ in range(6):
        lg = hf(input_ids=ref_ids, pixel_values=px, image_grid_thw=grid,
                pixel_values_videos=vpx, video_grid_thw=vgrid,
                mm_token_type_ids=torch.cat(
                    [mm_tt, torch.zeros(1, ref_ids.shape[1] - 20,
                                        dtype=mm_tt.dtype)], dim=1),
                use_cache=False).logits
        ref_ids = torch.cat([ref_ids, lg[:, -1].argmax(-1, keepdim=True)], dim=1)
    got = ours.generate(ids3, pixel_values=px, image_grid_thw=grid,
                        pixel_values_videos=vpx, video_grid_thw=vgrid,
                        max_new_tokens=6)
    assert torch.equal(got[:, : ref_ids.shape[1]], ref_ids), (
        got.tolist(), ref_ids.tolist())


@torch.inference_mode()
def test_qwen3_vl():
    """Qwen3-VL: DeepStack ViT (learned interpolated pos grid) + Qwen3 text
    with interleaved m-rope; greedy vs stepwise HF."""
    from transformers import Qwen3VLForConditionalGeneration as HF
    from transformers.models.qwen3_vl import Qwen3VLConfig

    from neuronx_distributed_inference_amd.models.qwen3_vl import (
        Qwen3VLForConditionalGeneration, Qwen3VLInferenceConfig,
    )

    torch.manual_seed(0)
    vc = dict(depth=3, hidden_size=32, intermediate_size=64, num_heads=4,
              in_channels=3, patch_size=4, temporal_patch_size=2,
              spatial_merge_size=2, out_hidden_size=48,
              num_position_embeddings=16, deepstack_visual_indexes=[0, 1],
              hidden_act="gelu_pytorch_tanh")
    tc = dict(vocab_size=160, hidden_size=48, intermediate_size=96,
              num_hidden_layers=3, num_attention_heads=4,
              num_key_value_heads=2, head_dim=12, max_position_embeddings=128,
              rms_norm_eps=1e-5, rope_theta=10000.0, tie_word_embeddings=False,
              rope_scaling={"rope_type": "default", "mrope_section": [2, 2, 2]})
    hf_cfg = Qwen3VLConfig(vision_config=vc, text_config=tc,
                           image_token_id=159, vision_start_token_id=158,
                           vision_end_token_id=157, video_token_id=156)
    hf = HF(hf_cfg).eval()

    ec = EngineConfig(batch_size=1, max_batch_size=1, seq_len=64, max_length=64,
                      torch_dtype="float32", tp_degree=1)
    cfg = Qwen3VLInferenceConfig(
        neuron_config=ec, text_config=dict(tc), vision_config=dict(vc),
        image_token_id=159, video_token_id=156)
    ours = Qwen3VLForConditionalGeneration(cfg).eval()
    sd = Qwen3VLForConditionalGeneration.convert_hf_to_native_state_dict(
        hf.state_dict(), cfg)
    missing, unexpected = ours.load_state_dict(sd, strict=False)
    real_missing = [m for m in missing if ".kv_mgr." not in m]
    assert not real_missing, real_missing
    assert not unexpected, unexpected

    grid = torch.tensor([[1, 4, 4]])
    px = torch.randn(16, 3 * 2 * 4 * 4)
    ids = torch.randint(0, 156, (1, 14))
    ids[0, 3:7] = 159
    n_new = 8
    ref_ids = ids.clone()
    for _ in range(n_new):
        lg = hf(input_ids=ref_ids, pixel_values=px, image_grid_thw=grid,
                mm_token_type_ids=(ref_ids == 159).int(),
                use_cache=False).logits
        ref_ids = torch.cat([ref_ids, lg[:, -1].argmax(-1, keepdim=True)], dim=1)
    got = ours.generate(ids, pixel_values=px, image_grid_thw=grid,
                        max_new_tokens=n_new)
    assert torch.equal(got[:, : ref_ids.shape[1]], ref_ids), (
        got.tolist(), ref_ids.tolist())

    # video: HF splits each video into per-frame (1,h,w) grids (timestamp
    # text tokens separate the frame runs); DeepStack injects at video
    # positions too
    vgrid = torch.tensor([[2, 4, 4]])
    vpx = torch.randn(32, 3 * 2 * 4 * 4)
    ids3 = torch.randint(0, 156, (1, 16))
    ids3[0, 2:6] = 156                    # frame 1 (4 merged tokens)
    ids3[0, 7:11] = 156                   # frame 2 (timestamp text between)
    mm_tt = torch.zeros_like(ids3)
    mm_tt[ids3 == 156] = 2
    ref_ids = ids3.clone()
    for _ in range(6):
        lg = hf(input_ids=ref_ids, pixel_values_videos=vpx,
                video_grid_thw=vgrid,
                mm_token_type_ids=torch.cat(
                    [mm_tt, torch.zeros(1, ref_ids.shape[1] - 16,
                                        dtype=mm_tt.dtype)], dim=1),
                use_cache=False).logits
        ref_ids = torch.cat([ref_ids, lg[:, -1].argmax(-1, keepdim=True)], dim=1)
    got = ours.generate(ids3, pixel_values_videos=vpx, video_grid_thw=vgrid,
                        max_new_tokens=6)
    assert torch.equal(got[:, : ref_ids.shape[1]], ref_ids), (
        got.tolist(), ref_ids.tolist())


@torch.inference_mode()
def test_llama4_vision():
    """Llama4 conditional generation: unfold-conv ViT + complex 2D rope +
    pixel-shuffle adapter + inline splice; greedy vs stepwise HF."""
    from transformers import Llama4Config
    from transformers.models.llama4.modeling_llama4 import (
        Llama4ForConditionalGeneration as HF,
    )

    from neuronx_distributed_inference_amd.models.llama4.modeling_llama4 import (
        Llama4ForConditionalGeneration, Llama4VisionInferenceConfig,
    )

    torch.manual_seed(0)
    # HF invariant: intermediate_size == hidden_size / pixel_shuffle_ratio^2
    vc = dict(hidden_size=32, intermediate_size=128, num_hidden_layers=2,
              num_attention_heads=4, image_size=16, patch_size=4,
              num_channels=3, pixel_shuffle_ratio=0.5,
              projector_input_dim=128, projector_output_dim=128,
              vision_output_dim=128, rope_theta=10000)
    tc = dict(vocab_size=150, hidden_size=48, intermediate_size=96,
              intermediate_size_mlp=96, num_hidden_layers=2,
              num_attention_heads=4, num_key_value_heads=2, head_dim=12,
              max_position_embeddings=128, num_local_experts=2,
              num_experts_per_tok=1, interleave_moe_layer_step=2,
              no_rope_layer_interval=2, attention_chunk_size=16,
              use_qk_norm=True, attn_temperature_tuning=True, floor_scale=16,
              attn_scale=0.1, rms_norm_eps=1e-5, tie_word_embeddings=False)
    hf_cfg = Llama4Config(vision_config=vc, text_config=tc, image_token_id=149)
    hf = HF(hf_cfg).eval()

    ec = EngineConfig(batch_size=1, max_batch_size=1, seq_len=64, max_length=64,
                      torch_dtype="float32", tp_degree=1)
    cfg = Llama4VisionInferenceConfig(
        neuron_config=ec, text_config=dict(tc), vision_config=dict(vc),
        image_token_id=149)
    ours = Llama4ForConditionalGeneration(cfg).eval()
    sd = Llama4ForConditionalGeneration.convert_hf_to_native_state_dict(
        hf.state_dict(), cfg)
    missing, unexpected = ours.load_state_dict(sd, strict=False)
    real_missing = [m for m in missing if ".kv_mgr." not in m]
    assert not real_missing, real_missing
    assert not unexpected, unexpected

    px = torch.randn(1, 3, 16, 16)   # 4x4 patches -> shuffle 0.5 -> 4 tokens
    ids = torch.randint(0, 149, (1, 12))
    ids[0, 2:6] = 149
    n_new = 6
    ref_ids = ids.clone()
    for _ in range(n_new):
        lg = hf(input_ids=ref_ids, pixel_values=px, use_cache=False).logits
        ref_ids = torch.cat([ref_ids, lg[:, -1].argmax(-1, keepdim=True)], dim=1)
    got = ours.generate(ids, pixel_values=px, max_new_tokens=n_new)
    assert torch.equal(got[:, : ref_ids.shape[1]], ref_ids), (
        got.tolist(), ref_ids.tolist())
