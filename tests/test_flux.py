"""Flux MMDiT structural tests.

No diffusers golden is available in this image (GAPS.md), so these verify
the structural invariants the architecture guarantees: AdaLN-Zero gating
(zero gates => transformer reduces to the analytic out-path), rope pair
isometry, scheduler boundary conditions, and deterministic denoising.
"""

import math

import pytest
import torch

from neuronx_distributed_inference_amd.config import EngineConfig
from neuronx_distributed_inference_amd.models.flux import (
    FlowMatchEulerScheduler,
    FluxInferenceConfig,
    FluxPipeline,
    FluxTransformer2DModel,
)


def _cfg():
    ec = EngineConfig(batch_size=1, max_batch_size=1, seq_len=32, max_length=32,
                      torch_dtype="float32", tp_degree=1)
    return FluxInferenceConfig(
        neuron_config=ec, num_layers=2, num_single_layers=2,
        attention_head_dim=32, num_attention_heads=2, in_channels=16,
        joint_attention_dim=48, pooled_projection_dim=24,
        axes_dims_rope=[8, 12, 12], guidance_embeds=True)


def _model():
    torch.manual_seed(0)
    m = FluxTransformer2DModel(_cfg()).eval()
    with torch.no_grad():
        for p in m.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.05)
            else:
                p.zero_()   # parallel-layer biases are torch.empty
    return m


@torch.inference_mode()
def test_adaln_zero_gates_give_analytic_output():
    """With every gate chunk zeroed, all blocks are identities and the
    output must equal proj_out(norm_out(x_embedder(x))) exactly — verifies
    the residual/gating wiring of double AND single blocks."""
    m = _model()
    d = m.inner_dim
    with torch.no_grad():
        for blk in m.transformer_blocks:
            for ln in (blk.norm1.linear, blk.norm1_context.linear):
                ln.weight[2 * d:3 * d].zero_()   # gate_msa
                ln.bias[2 * d:3 * d].zero_()
                ln.weight[5 * d:6 * d].zero_()   # gate_mlp
                ln.bias[5 * d:6 * d].zero_()
        for blk in m.single_transformer_blocks:
            blk.norm_linear.weight[2 * d:3 * d].zero_()
            blk.norm_linear.bias[2 * d:3 * d].zero_()
    x = torch.randn(1, 16, 16)
    txt = torch.randn(1, 5, 48)
    pooled = torch.randn(1, 24)
    t = torch.tensor([0.7])
    g = torch.tensor([3.5])
    img_ids, txt_ids = FluxPipeline.prepare_ids(8, 8, 5, x.device)
    out = m(x, txt, pooled, t, img_ids, txt_ids, guidance=g)
    temb = m.time_text_embed(t * 1000, g * 1000, pooled)
    want = m.proj_out(m.norm_out(m.x_embedder(x), temb))
    torch.testing.assert_close(out, want, atol=1e-5, rtol=1e-5)


def test_rope_pair_isometry():
    from neuronx_distributed_inference_amd.models.flux.modeling_flux import (
        apply_rope_pairs, rope_3axis,
    )

    torch.manual_seed(1)
    ids = torch.randint(0, 50, (6, 3)).float()
    cos, sin = rope_3axis(ids, [8, 12, 12], 10000.0)
    assert cos.shape == (6, 32)
    x = torch.randn(1, 2, 6, 32)
    y = apply_rope_pairs(x, cos, sin)
    # rotation preserves the norm of each (2i, 2i+1) pair
    xn = x.reshape(1, 2, 6, -1, 2).norm(dim=-1)
    yn = y.reshape(1, 2, 6, -1, 2).norm(dim=-1)
    torch.testing.assert_close(xn, yn, atol=1e-5, rtol=1e-5)
    # position 0 is the identity
    ids0 = torch.zeros(1, 3)
    c0, s0 = rope_3axis(ids0, [8, 12, 12], 10000.0)
    torch.testing.assert_close(c0, torch.ones_like(c0))
    torch.testing.assert_close(s0, torch.zeros_like(s0))


def test_scheduler_shift_and_step():
    s = FlowMatchEulerScheduler()
    sig = s.sigmas_for(8, 1024)
    assert sig.shape == (9,) and sig[-1] == 0
    assert (sig[:-1] > sig[1:]).all()          # strictly decreasing
    assert 0 < sig[0] <= 1.0
    x = torch.randn(2, 4)
    v = torch.randn(2, 4)
    # integrating to sigma=0 with the true velocity (x1 - x0 form) is exact
    out = s.step(x, v, 0.3, 0.0)
    torch.testing.assert_close(out, x - 0.3 * v)


@torch.inference_mode()
def test_pipeline_deterministic_denoise():
    m = _model()
    pipe = FluxPipeline(m)
    txt = torch.randn(1, 5, 48)
    pooled = torch.randn(1, 24)
    g1 = torch.Generator().manual_seed(42)
    g2 = torch.Generator().manual_seed(42)
    a = pipe.denoise(txt, pooled, height_lat=8, width_lat=8, num_steps=3,
                     generator=g1)
    b = pipe.denoise(txt, pooled, height_lat=8, width_lat=8, num_steps=3,
                     generator=g2)
    assert a.shape == (1, 16, 16)
    assert torch.isfinite(a).all()
    torch.testing.assert_close(a, b)


@torch.inference_mode()
def test_clip_text_encoder_matches_hf():
    from transformers import CLIPTextConfig, CLIPTextModel

    from neuronx_distributed_inference_amd.models.flux.text_encoders import (
        ClipTextEncoder,
    )

    torch.manual_seed(0)
    cfg = CLIPTextConfig(vocab_size=100, hidden_size=32, intermediate_size=64,
                         num_hidden_layers=2, num_attention_heads=4,
                         max_position_embeddings=24, eos_token_id=2,
                         bos_token_id=1, pad_token_id=0, hidden_act="quick_gelu")
    hf = CLIPTextModel(cfg).eval()
    ours = ClipTextEncoder(100, 32, 4, 64, 2, 24, eos_token_id=2).eval()
    sd = ClipTextEncoder.convert_hf_to_native_state_dict(hf.state_dict())
    missing, unexpected = ours.load_state_dict(sd, strict=False)
    assert not missing, missing
    ids = torch.randint(3, 100, (2, 10))
    ids[:, -1] = 2
    want = hf(input_ids=ids)
    h, pooled = ours(ids)
    torch.testing.assert_close(h, want.last_hidden_state, atol=1e-4, rtol=1e-4)
    torch.testing.assert_close(pooled, want.pooler_output, atol=1e-4, rtol=1e-4)


@torch.inference_mode()
def test_t5_encoder_matches_hf():
    from transformers import T5Config, T5EncoderModel

    from neuronx_distributed_inference_amd.models.flux.text_encoders import (
        T5TextEncoder,
    )

    torch.manual_seed(0)
    cfg = T5Config(vocab_size=120, d_model=32, d_kv=8, d_ff=64,
                   num_layers=2, num_heads=4, relative_attention_num_buckets=8,
                   relative_attention_max_distance=20,
                   feed_forward_proj="gated-gelu", dense_act_fn="gelu_new",
                   is_gated_act=True, tie_word_embeddings=False)
    hf = T5EncoderModel(cfg).eval()
    ours = T5TextEncoder(120, 32, 4, 8, 64, 2, num_buckets=8,
                         max_distance=20).eval()
    sd = T5TextEncoder.convert_hf_to_native_state_dict(hf.state_dict())
    missing, unexpected = ours.load_state_dict(sd, strict=False)
    assert not missing, missing
    ids = torch.randint(0, 120, (2, 12))
    want = hf(input_ids=ids).last_hidden_state
    got = ours(ids)
    torch.testing.assert_close(got, want, atol=1e-4, rtol=1e-4)
