"""Tier-1 unit tests: ops reference implementations, bucketing, masks,
GQA shard math, KV managers, sampling (CPU, no device)."""

import math

import pytest
import torch

from neuronx_distributed_inference_amd.modules import autobucketing as ab
from neuronx_distributed_inference_amd.modules.attention import utils as attn_utils
from neuronx_distributed_inference_amd.modules.attention.gqa import (
    GQA,
    get_shardable_head_counts,
)
from neuronx_distributed_inference_amd.ops import ref


# ---------------- autobucketing -----------------
def test_generate_buckets_powers_of_two():
    assert ab.generate_buckets(128, 1024) == [128, 256, 512, 1024]
    assert ab.generate_buckets(128, 100) == [100]
    assert ab.generate_buckets(128, 1000) == [128, 256, 512, 1000]


def test_get_target_bucket():
    assert ab.get_target_bucket(5, [128, 256]) == 128
    assert ab.get_target_bucket(129, [128, 256]) == 256
    with pytest.raises(ValueError):
        ab.get_target_bucket(300, [128, 256])


# ---------------- masks -----------------
def test_causal_mask():
    m = attn_utils.causal_mask(4, 4)[0, 0]
    assert m.equal(torch.ones(4, 4, dtype=torch.bool).tril())


def test_causal_mask_kv_longer():
    m = attn_utils.causal_mask(2, 6)[0, 0]
    # queries are positions 4,5
    assert m[0].tolist() == [True] * 5 + [False]
    assert m[1].tolist() == [True] * 6


def test_sliding_window_mask():
    m = attn_utils.sliding_window_mask(5, 5, 2)[0, 0]
    for i in range(5):
        for j in range(5):
            assert m[i, j] == (j <= i and j > i - 2)


def test_speculation_mask():
    m = attn_utils.speculation_mask(1, 3, 4)[0, 0]
    assert m.shape == (3, 7)
    assert m[:, :4].all()
    assert m[:, 4:].equal(torch.ones(3, 3, dtype=torch.bool).tril())


# ---------------- GQA shard math -----------------
def test_gqa_shard_strategies():
    assert get_shardable_head_counts(4, 32, 8) == (32, 8, GQA.SHARD_OVER_TP)
    assert get_shardable_head_counts(8, 32, 32) == (32, 32, GQA.MHA)
    heads, kv, strat = get_shardable_head_counts(16, 32, 8)
    assert (heads, kv, strat) == (32, 16, GQA.REPLICATE_TO_TP_DEGREE)
    # non-divisible q heads get padded
    heads, _, _ = get_shardable_head_counts(8, 30, 30)
    assert heads == 32


# ---------------- ops reference numerics -----------------
def test_rms_norm_matches_manual():
    x = torch.randn(2, 5, 16)
    w = torch.randn(16)
    out, _ = ref.rms_norm(x, w, 1e-6)
    expect = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-6) * w
    torch.testing.assert_close(out, expect, atol=1e-5, rtol=1e-5)


def test_rms_norm_fused_residual():
    x = torch.randn(2, 5, 16)
    r = torch.randn(2, 5, 16)
    w = torch.ones(16)
    out, new_r = ref.rms_norm(x, w, 1e-6, residual=r)
    torch.testing.assert_close(new_r, x + r)


def test_rope_matches_hf():
    from transformers.models.llama.modeling_llama import apply_rotary_pos_emb

    B, H, S, D = 2, 4, 6, 16
    q = torch.randn(B, H, S, D)
    k = torch.randn(B, 2, S, D)
    inv = 1.0 / (10000.0 ** (torch.arange(0, D, 2).float() / D))
    pos = torch.arange(S).float()
    freqs = torch.outer(pos, inv)
    emb = torch.cat([freqs, freqs], dim=-1)
    cos, sin = emb.cos(), emb.sin()
    q2, k2 = ref.rope_apply(q, k, cos, sin)
    qh, kh = apply_rotary_pos_emb(q, k, cos.unsqueeze(0), sin.unsqueeze(0))
    torch.testing.assert_close(q2, qh, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(k2, kh, atol=1e-5, rtol=1e-5)


def test_sdpa_gqa_equivalence():
    B, Hq, Hkv, S, D = 1, 4, 2, 8, 16
    q = torch.randn(B, Hq, S, D)
    k = torch.randn(B, Hkv, S, D)
    v = torch.randn(B, Hkv, S, D)
    mask = attn_utils.causal_mask(S, S)
    out = ref.sdpa(q, k, v, mask=mask)
    expect = torch.nn.functional.scaled_dot_product_attention(
        q, k.repeat_interleave(2, 1), v.repeat_interleave(2, 1), is_causal=True
    )
    torch.testing.assert_close(out, expect, atol=1e-5, rtol=1e-5)


def test_kv_cache_update_with_seq_ids():
    kc = torch.zeros(4, 2, 10, 8)
    vc = torch.zeros(4, 2, 10, 8)
    k = torch.randn(2, 2, 3, 8)
    v = torch.randn(2, 2, 3, 8)
    pos = torch.tensor([[0, 1, 2], [4, 5, 6]])
    sid = torch.tensor([3, 1])
    ref.kv_cache_update(kc, vc, k, v, pos, sid)
    torch.testing.assert_close(kc[3, :, 0:3], k[0])
    torch.testing.assert_close(kc[1, :, 4:7], k[1])
    assert kc[0].abs().sum() == 0 and kc[2].abs().sum() == 0


def test_block_kv_cache_update_and_gather():
    nb, bs, H, D = 8, 4, 2, 8
    kc = torch.zeros(nb, bs, H, D)
    vc = torch.zeros(nb, bs, H, D)
    n = 6
    k = torch.randn(n, H, D)
    v = torch.randn(n, H, D)
    # sequence occupies blocks 5 then 2
    slots = torch.tensor([5 * bs + 0, 5 * bs + 1, 5 * bs + 2, 5 * bs + 3, 2 * bs + 0, 2 * bs + 1])
    ref.block_kv_cache_update(kc, vc, k, v, slots)
    table = torch.tensor([[5, 2]])
    kg, vg = ref.paged_gather_kv(kc, vc, table, 6)
    torch.testing.assert_close(kg[0].transpose(0, 1), k)


def test_topk_topp_sample_greedy_and_topp():
    torch.manual_seed(0)
    logits = torch.randn(3, 50)
    out = ref.topk_topp_sample(
        logits,
        top_k=torch.tensor([1, 1, 1]),
        top_p=torch.ones(3),
        temperature=torch.ones(3),
        rand=torch.zeros(3),
    )
    assert torch.equal(out, logits.argmax(-1))
    # top_k=5, rand=0 picks the argmax too (first CDF entry)
    out2 = ref.topk_topp_sample(
        logits, torch.tensor([5, 5, 5]), torch.full((3,), 0.9), torch.ones(3), torch.zeros(3)
    )
    assert torch.equal(out2, logits.argmax(-1))
    # rand near 1 with top_k=2 picks the 2nd-best
    out3 = ref.topk_topp_sample(
        logits, torch.tensor([2, 2, 2]), torch.ones(3), torch.ones(3), torch.full((3,), 0.999)
    )
    second = logits.topk(2, -1).indices[:, 1]
    assert torch.equal(out3, second)


# ---------------- sampler module -----------------
def test_sampler_greedy():
    from neuronx_distributed_inference_amd.config import OnDeviceSamplingConfig
    from neuronx_distributed_inference_amd.modules.generation.sampling import Sampler

    s = Sampler(OnDeviceSamplingConfig(top_k=1), vocab_size=64)
    logits = torch.randn(4, 1, 64)
    out = s(logits)
    assert torch.equal(out, logits[:, -1].argmax(-1))


def test_sampler_multinomial_topk():
    from neuronx_distributed_inference_amd.config import OnDeviceSamplingConfig
    from neuronx_distributed_inference_amd.modules.generation.sampling import (
        Sampler,
        prepare_sampling_params,
    )

    torch.manual_seed(0)
    s = Sampler(OnDeviceSamplingConfig(top_k=4, do_sample=True, deterministic=True), vocab_size=64)
    logits = torch.randn(4, 64)
    params = prepare_sampling_params(4, top_k=4, top_p=0.9, temperature=0.7)
    out = s(logits, params)
    top4 = logits.topk(4, -1).indices
    assert all(out[i] in top4[i] for i in range(4))


@torch.inference_mode()
def test_compact_swa_ring_cache_matches_full():
    """compact_swa_kv: mistral with a tiny sliding window generates the
    same tokens with a ring cache (window rows) as with the full cache —
    incl. prompts and generations well beyond the window."""
    import copy

    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.mistral.modeling_mistral import (
        MistralForCausalLM,
        MistralInferenceConfig,
    )
    from neuronx_distributed_inference_amd.utils.hf_adapter import (
        GenerationConfig,
        HuggingFaceGenerationAdapter,
    )

    def make(compact):
        ec = EngineConfig(batch_size=2, max_batch_size=2, seq_len=64,
                          max_length=64, torch_dtype="float32", tp_degree=1,
                          compact_swa_kv=compact)
        return MistralInferenceConfig(
            neuron_config=ec, vocab_size=128, hidden_size=64,
            intermediate_size=128, num_hidden_layers=2, num_attention_heads=4,
            num_key_value_heads=2, max_position_embeddings=128,
            rms_norm_eps=1e-5, sliding_window=8, tie_word_embeddings=False)

    torch.manual_seed(3)
    full = MistralForCausalLM(make(False)).eval()
    with torch.no_grad():
        for p in full.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    ring = MistralForCausalLM(make(True)).eval()
    sd = {k: v for k, v in full.model.state_dict().items()
          if not k.startswith("kv_mgr.")}
    ring.model.load_state_dict(sd, strict=False)
    assert ring.model.kv_mgr.layer_cache_sizes == [8, 8]
    assert all(ring.model.kv_mgr.ring_layers)

    ids = torch.randint(0, 128, (2, 20))   # prompt longer than the window
    gc = GenerationConfig(max_new_tokens=12)
    want = HuggingFaceGenerationAdapter(full).generate(ids, generation_config=gc)
    got = HuggingFaceGenerationAdapter(ring).generate(ids, generation_config=gc)
    assert torch.equal(got, want), (got.tolist(), want.tolist())


@torch.inference_mode()
def test_moe_dense_matches_gather_path():
    """Dense bmm expert path (graph-safe decode) vs the gather loop."""
    from neuronx_distributed_inference_amd.modules.moe import ExpertMLPs

    torch.manual_seed(0)
    for scale_input in (False, True):
        m = ExpertMLPs(4, 16, 32, tp_degree=1, dtype=torch.float32,
                       scale_input=scale_input)
        for p in m.parameters():
            p.normal_(0, 0.2)
        x = torch.randn(6, 16)
        w = torch.rand(6, 2)
        idx = torch.stack([torch.randperm(4)[:2] for _ in range(6)])
        dense = m._forward_dense(x, w, idx)
        T = x.shape[0]
        out = torch.zeros(T, 16)
        for e in range(4):
            sel = (idx == e)
            rows = sel.any(dim=-1)
            if not rows.any():
                continue
            xe = x[rows]
            ww = (w * sel).sum(dim=-1)[rows].unsqueeze(-1)
            if scale_input:
                xe = xe * ww
            import torch.nn.functional as F
            ge = F.linear(xe, m.w_gate[e])
            ue = F.linear(xe, m.w_up[e])
            he = F.linear(m.act_mul(ge, ue), m.w_down[e])
            out[rows] += he if scale_input else he * ww
        torch.testing.assert_close(dense, out, atol=1e-5, rtol=1e-5)


def test_composite_config_json_roundtrip():
    """Multimodal composite configs (nested text/vision configs) must
    survive save()/load() — nested InferenceConfig objects serialize as
    dicts and are reconstructed by add_derived_config."""
    import tempfile

    from neuronx_distributed_inference_amd.config import (
        EngineConfig,
        InferenceConfig,
    )
    from neuronx_distributed_inference_amd.models.mllama import (
        MllamaInferenceConfig,
    )

    ec = EngineConfig(batch_size=1, max_batch_size=1, seq_len=48, max_length=48,
                      torch_dtype="float32", tp_degree=1)
    tc = dict(vocab_size=120, hidden_size=32, intermediate_size=64,
              num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
              cross_attention_layers=[1, 3], rope_theta=10000.0,
              max_position_embeddings=64, rms_norm_eps=1e-5,
              tie_word_embeddings=False)
    vc = dict(hidden_size=32, intermediate_size=64, num_hidden_layers=3,
              num_global_layers=1, attention_heads=4, image_size=32,
              patch_size=16, max_num_tiles=2,
              supported_aspect_ratios=[[1, 1], [1, 2]],
              intermediate_layers_indices=[1, 2], vision_output_dim=96)
    cfg = MllamaInferenceConfig(neuron_config=ec, text_config=tc,
                                vision_config=vc)
    with tempfile.TemporaryDirectory() as d:
        cfg.save(d)
        back = MllamaInferenceConfig.load(d)
        assert back.text_config.hidden_size == 32
        assert back.text_config.cross_attention_layers == [1, 3]
        assert back.vision_config["max_num_tiles"] == 2
        assert back.hidden_size == 32
        # generic load reconstructs the concrete class by dotted path
        back2 = InferenceConfig.load(d)
        assert type(back2).__name__ == "MllamaInferenceConfig"


def test_qwen2vl_config_json_roundtrip():
    """Composite configs with extra scalar attrs (video_token_id) and a
    nested text InferenceConfig survive save()/load()."""
    import tempfile

    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.qwen2_vl import (
        Qwen2VLInferenceConfig,
    )

    ec = EngineConfig(batch_size=1, max_batch_size=1, seq_len=64, max_length=64,
                      torch_dtype="float32", tp_degree=1)
    tc = dict(vocab_size=160, hidden_size=48, intermediate_size=96,
              num_hidden_layers=2, num_attention_heads=4,
              num_key_value_heads=2, max_position_embeddings=128,
              rms_norm_eps=1e-5, rope_theta=10000.0, tie_word_embeddings=False,
              rope_scaling={"rope_type": "default", "mrope_section": [2, 2, 2]})
    vc = dict(depth=2, embed_dim=32, hidden_size=48, num_heads=4,
              in_channels=3, patch_size=4, temporal_patch_size=2,
              spatial_merge_size=2, mlp_ratio=2, hidden_act="quick_gelu")
    cfg = Qwen2VLInferenceConfig(neuron_config=ec, text_config=tc,
                                 vision_config=vc, image_token_id=159,
                                 video_token_id=156)
    with tempfile.TemporaryDirectory() as d:
        cfg.save(d)
        back = Qwen2VLInferenceConfig.load(d)
        assert back.text_config.hidden_size == 48
        assert back.text_config.rope_scaling["mrope_section"] == [2, 2, 2]
        assert back.video_token_id == 156
        assert back.vision_config["embed_dim"] == 32


def test_async_executor_inflight_cap_and_order():
    """AsyncExecutor (ref modules/async_execution.py): back-pressure at
    max_inflight, results in submission order, drain completes all."""
    import torch

    from neuronx_distributed_inference_amd.modules.async_execution import (
        AsyncExecutor,
    )

    done = []
    ex = AsyncExecutor(max_inflight=2, device=torch.device("cpu"))
    results = [ex.submit(lambda i=i: (done.append(i), i * 10)[1])
               for i in range(5)]
    ex.drain()
    assert [r.result() for r in results] == [0, 10, 20, 30, 40]
    assert done == list(range(5))
    assert all(r.done() for r in results)


def test_moe_grouped_tables():
    import torch

    from neuronx_distributed_inference_amd import ops

    torch.manual_seed(0)
    E, Ttk = 5, 37
    idx = torch.randint(0, E, (Ttk,))
    order, dest, expert_ids, gtab, Mpad = ops.moe_grouped_tables(idx, E)
    assert Mpad % 64 == 0 and dest.max() < Mpad
    # each sorted assignment's destination tile belongs to its expert
    sorted_e = idx[order]
    tiles = (dest // ops.MOE_TILE).long()
    assert (expert_ids[tiles].long() == sorted_e).all()
    # destinations are unique
    assert dest.unique().numel() == Ttk
    # tiles past the padded end are -1
    total_pad = ((torch.bincount(idx, minlength=E) + 63) // 64 * 64).sum()
    assert (expert_ids[int(total_pad) // 64:] == -1).all()


def test_moe_grouped_matches_dense():
    """Grouped expert path (CPU reference kernel) must equal the dense
    bmm path, including gate scaling and input-scaled (Llama4) mode."""
    import torch

    from neuronx_distributed_inference_amd.modules.moe import ExpertMLPs

    torch.manual_seed(1)
    for scale_input in (False, True):
        m = ExpertMLPs(num_experts=4, hidden_size=64, intermediate_size=128,
                       tp_degree=1, dtype=torch.float32,
                       scale_input=scale_input)
        with torch.no_grad():
            for p in m.parameters():
                p.normal_(0, 0.1)
        T, top_k = 9, 2
        x = torch.randn(T, 64) * 0.3
        logits = torch.randn(T, 4)
        w, idx = torch.softmax(logits, -1).topk(top_k, dim=-1)
        want = m._forward_dense(x, w, idx)
        got = m._forward_grouped(x, w, idx)
        torch.testing.assert_close(got, want, atol=1e-4, rtol=1e-4)


def test_moe_grouped_masks_nonlocal_experts():
    """EP: assignments to experts outside the local range contribute
    zero (sentinel segment skipped + masked combine)."""
    import torch

    from neuronx_distributed_inference_amd.modules.moe import ExpertMLPs

    torch.manual_seed(2)
    m = ExpertMLPs(num_experts=2, hidden_size=64, intermediate_size=128,
                   tp_degree=1, dtype=torch.float32)
    with torch.no_grad():
        for p in m.parameters():
            p.normal_(0, 0.1)
    T = 5
    x = torch.randn(T, 64) * 0.3
    # top_k = 2: one local (0/1), one out-of-range (7)
    idx = torch.stack([torch.randint(0, 2, (T,)), torch.full((T,), 7)], dim=1)
    w = torch.full((T, 2), 0.5)
    got = m._forward_grouped(x, w, idx)
    want = m._forward_grouped(x, w[:, :1], idx[:, :1])
    torch.testing.assert_close(got, want, atol=1e-5, rtol=1e-5)


def test_gpt_oss_mxfp4_experts_match_dequantized():
    """GPT-OSS experts with MXFP4-packed weights (grouped path, CPU
    reference dequant) must match the dense path on the SAME dequantized
    values — the packing is exact given identical dequant."""
    import torch

    from neuronx_distributed_inference_amd import ops
    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.gpt_oss.modeling_gpt_oss import (
        GptOssExperts,
        GptOssInferenceConfig,
    )

    torch.manual_seed(0)
    E, H, I = 4, 64, 64
    ec = EngineConfig(batch_size=2, seq_len=32, max_length=32,
                      torch_dtype="float32", tp_degree=1, moe_mxfp4=True)
    cfg = GptOssInferenceConfig(
        neuron_config=ec, vocab_size=64, hidden_size=H, intermediate_size=I,
        num_hidden_layers=1, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=E, num_experts_per_tok=2, max_position_embeddings=64,
        rms_norm_eps=1e-5, rope_theta=10000.0, layer_types=["full_attention"],
        tie_word_embeddings=False)
    mx = GptOssExperts(cfg)
    assert mx.mxfp4
    w_gu = torch.randn(E, 2 * I, H) * 0.2
    w_dn = torch.randn(E, H, I) * 0.2
    with torch.no_grad():
        q, s = ops.mx4_quantize(w_gu)
        mx.gate_up_q.copy_(q)
        mx.gate_up_s.copy_(s)
        q, s = ops.mx4_quantize(w_dn)
        mx.down_q.copy_(q)
        mx.down_s.copy_(s)
        mx.gate_up_proj_bias.normal_(0, 0.05)
        mx.down_proj_bias.normal_(0, 0.05)

    ec2 = EngineConfig(batch_size=2, seq_len=32, max_length=32,
                       torch_dtype="float32", tp_degree=1)
    cfg2 = GptOssInferenceConfig(
        neuron_config=ec2, vocab_size=64, hidden_size=H, intermediate_size=I,
        num_hidden_layers=1, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=E, num_experts_per_tok=2, max_position_embeddings=64,
        rms_norm_eps=1e-5, rope_theta=10000.0, layer_types=["full_attention"],
        tie_word_embeddings=False)
    dense = GptOssExperts(cfg2)
    with torch.no_grad():
        dense.gate_up_proj.copy_(ops.mx4_dequant(mx.gate_up_q, mx.gate_up_s))
        dense.down_proj.copy_(ops.mx4_dequant(mx.down_q, mx.down_s))
        dense.gate_up_proj_bias.copy_(mx.gate_up_proj_bias)
        dense.down_proj_bias.copy_(mx.down_proj_bias)

    T = 7
    x = torch.randn(T, H) * 0.3
    wts = torch.softmax(torch.randn(T, 2), dim=-1)
    idx = torch.randint(0, E, (T, 2))
    got = mx(x, wts, idx)
    want = dense(x, wts, idx)
    torch.testing.assert_close(got, want, atol=1e-4, rtol=1e-4)
    # and the grouped CPU path agrees with the per-expert loop
    got_g = mx._forward_grouped(x, wts, idx)
    torch.testing.assert_close(got_g, want, atol=1e-4, rtol=1e-4)


def test_ring_wrap_positions_drop_superseded():
    """Ring cache: positions whose slot is reclaimed by a NEWER position
    in the same write batch must be dropped (-1) — two racing writes to
    one slot in a single scatter are order-undefined on the GPU."""
    import torch

    from neuronx_distributed_inference_amd.modules.kvcache.kv_cache_manager import (
        KVCacheManager,
    )

    m = KVCacheManager(num_layers=1, batch_size=2, num_kv_heads=1,
                       head_dim=8, max_len=16, dtype=torch.float32,
                       sliding_window=16)
    pos = torch.tensor([[i for i in range(24)],
                        [-1] * 4 + [i for i in range(20)]])
    w = m.wrap_positions(0, pos)
    # row 0: positions 0..7 collide with 16..23 -> dropped
    assert (w[0, :8] == -1).all()
    assert (w[0, 8:] == torch.arange(8, 24) % 16).all()
    # row 1: pads stay -1; positions 0..3 collide with 16..19
    assert (w[1, :4] == -1).all()
    assert (w[1, 4:8] == -1).all()
    assert (w[1, 8:] == torch.arange(4, 20) % 16).all()


def test_bucket_strategies():
    from neuronx_distributed_inference_amd.modules.autobucketing import (
        get_target_bucket,
    )

    bs = [128, 256, 512, 1024]
    assert get_target_bucket(200, bs) == 256
    assert get_target_bucket(200, bs, "second_fit") == 512
    assert get_target_bucket(1000, bs, "second_fit") == 1024
    assert get_target_bucket(5, bs, "max") == 1024


def test_tensor_capture_replacement_configs():
    """TensorCapture/TensorReplacement sub-configs (ref config.py:
    1121-1202): JSON round trip + attach_tensor_capture records module
    outputs + replacements feed goldens."""
    import tempfile

    import torch

    from neuronx_distributed_inference_amd.config import (
        EngineConfig,
        TensorCaptureConfig,
        TensorReplacementConfig,
    )
    from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
        LlamaForCausalLM,
        LlamaInferenceConfig,
    )

    ec = EngineConfig(
        batch_size=1, seq_len=32, max_length=32, torch_dtype="float32",
        tp_degree=1,
        tensor_capture_config=TensorCaptureConfig(
            module_names=["layers.0"], max_steps=4),
    )
    d = ec.to_dict()
    ec2 = EngineConfig.from_dict(d)
    assert ec2.tensor_capture_config.module_names == ["layers.0"]
    assert ec2.tensor_capture_config.max_steps == 4

    cfg = LlamaInferenceConfig(
        neuron_config=ec, vocab_size=64, hidden_size=32, intermediate_size=64,
        num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=1,
        max_position_embeddings=64, rms_norm_eps=1e-5, rope_theta=10000.0,
        tie_word_embeddings=False)
    torch.manual_seed(0)
    m = LlamaForCausalLM(cfg).eval()
    ids = torch.randint(0, 64, (1, 8))
    with m.attach_tensor_capture() as cap:
        m(input_ids=ids)
    assert len(cap.captured["layers.0"]) == 1
    assert cap.captured["layers.0"][0].shape[:2] == (1, 8)

    # replacement: feed a golden input into layer 1 and observe it used
    golden = torch.zeros(1, 8, 32)
    with tempfile.NamedTemporaryFile(suffix=".pt", delete=False) as f:
        torch.save(golden, f.name)
        path = f.name
    m.config.neuron_config.tensor_replacement_config = TensorReplacementConfig(
        replacements={"layers.1": path})
    m.reset()
    with m.attach_tensor_capture() as cap2, m.apply_tensor_replacements():
        m(input_ids=ids)
    # layer 1 received zeros -> its output differs from the unreplaced run
    assert not torch.allclose(cap2.captured["layers.0"][0],
                              golden)


def test_moe_grouped_gemm_fusions_cpu_ref():
    """Fused x-gather staging + fused routing-weighted combine epilogue
    must equal the materialized gather + index_add combine (CPU ref)."""
    import torch

    from neuronx_distributed_inference_amd import ops

    torch.manual_seed(4)
    E, T, top_k, K, N = 4, 50, 2, 64, 128
    idx = torch.randint(0, E, (T * top_k,))
    order, dest, eids, gtab, Mpad = ops.moe_grouped_tables(idx, E, top_k)
    x = torch.randn(T, K)
    w = torch.randn(E, N, K) * 0.1
    gatew = torch.rand(T * top_k)[order]

    # reference: materialized pipeline
    tok = order // top_k
    xs = torch.zeros(Mpad, K)
    xs.index_copy_(0, dest, x.index_select(0, tok))
    y = ops.moe_grouped_gemm(xs, w, eids)
    contrib = y.index_select(0, dest) * gatew.unsqueeze(-1)
    want = torch.zeros(T, N)
    want.index_add_(0, tok, contrib)

    # fused: gather in, combine out
    y2 = ops.moe_grouped_gemm(x, w, eids, x_gather=gtab)
    torch.testing.assert_close(y2.index_select(0, dest),
                               y.index_select(0, dest))
    gw_tab = torch.zeros(Mpad)
    gw_tab[dest] = gatew
    out32 = torch.zeros(T, N)
    got = ops.moe_grouped_gemm(xs, w, eids, combine_out=out32,
                               combine_scale=gw_tab, combine_gather=gtab)
    torch.testing.assert_close(got, want, atol=1e-4, rtol=1e-4)


def test_moe_fp8_grouped_cpu_matches_bf16():
    """ExpertMLPs fp8 mode (e4m3 weights + per-channel scales through the
    grouped path) vs the bf16 dense path on CPU."""
    import torch

    from neuronx_distributed_inference_amd.modules.moe import ExpertMLPs

    torch.manual_seed(6)
    E, H, I, T, top_k = 4, 256, 512, 20, 2
    m = ExpertMLPs(num_experts=E, hidden_size=H, intermediate_size=I,
                   tp_degree=1, dtype=torch.float32)
    with torch.no_grad():
        for p in m.parameters():
            p.normal_(0, 0.05)
    x = torch.randn(T, H) * 0.3
    w, idx = torch.softmax(torch.randn(T, E), -1).topk(top_k, dim=-1)
    want = m._forward_dense(x, w, idx)
    assert m.quantize_fp8()
    assert not any(n in dict(m.named_parameters())
                   for n in ("w_gate", "w_up", "w_down"))
    got = m._forward_grouped(x, w, idx)
    # e4m3 weight round-off: ~2^-3 relative per weight, averaged down by
    # the K-dim fan-in
    err = (got - want).abs().max() / want.abs().max().clamp(min=1e-6)
    assert err < 0.06, float(err)


def test_moe_mx4_grouped_cpu_matches_bf16():
    """ExpertMLPs MXFP4 mode vs the dense path on CPU (4-bit noise is
    ~12% relative on random weights — tolerance reflects that)."""
    import torch

    from neuronx_distributed_inference_amd.modules.moe import ExpertMLPs

    torch.manual_seed(11)
    E, H, I, T, top_k = 4, 256, 512, 16, 2
    m = ExpertMLPs(num_experts=E, hidden_size=H, intermediate_size=I,
                   tp_degree=1, dtype=torch.float32)
    with torch.no_grad():
        for p in m.parameters():
            p.normal_(0, 0.05)
    x = torch.randn(T, H) * 0.3
    w, idx = torch.softmax(torch.randn(T, E), -1).topk(top_k, dim=-1)
    want = m._forward_dense(x, w, idx)
    assert m.quantize_mx4()
    got = m._forward_grouped(x, w, idx)
    err = (got - want).abs().max() / want.abs().max().clamp(min=1e-6)
    assert err < 0.35, float(err)  # 4-bit weights: coarse but bounded


def test_moe_grouped_tile32_cpu_matches_dense(monkeypatch):
    """32-row tile path (fine-grained MoE) vs the dense path on CPU."""
    import torch

    from neuronx_distributed_inference_amd.modules.moe import ExpertMLPs

    monkeypatch.setenv("NXDI_MOE_TILE32", "1")
    torch.manual_seed(14)
    E, H, I, T, top_k = 16, 128, 256, 24, 4   # 6 tokens/expert avg
    m = ExpertMLPs(num_experts=E, hidden_size=H, intermediate_size=I,
                   tp_degree=1, dtype=torch.float32)
    with torch.no_grad():
        for p in m.parameters():
            p.normal_(0, 0.05)
    x = torch.randn(T, H) * 0.3
    w, idx = torch.softmax(torch.randn(T, E), -1).topk(top_k, dim=-1)
    want = m._forward_dense(x, w, idx)
    got = m._forward_grouped(x, w, idx)
    torch.testing.assert_close(got, want, atol=1e-4, rtol=1e-4)
