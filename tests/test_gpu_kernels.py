"""GPU numerics tests: each CDNA4 HIP kernel vs the plain-PyTorch fp32
reference (`ops.ref`) on the same inputs."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.ops import ref


def _dev():
    return torch.device("cuda:0")


def test_extension_loaded():
    assert ops.extension_available(), f"HIP extension missing: {ops._C_IMPORT_ERROR}"


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float16, torch.float32])
@pytest.mark.parametrize("shape", [(2, 5, 256), (1, 1, 4096), (4, 128, 1024)])
def test_rms_norm(dtype, shape):
    torch.manual_seed(0)
    x = torch.randn(shape, dtype=dtype, device=_dev())
    w = torch.randn(shape[-1], dtype=dtype, device=_dev())
    out, _ = ops.rms_norm(x, w, 1e-5)
    want, _ = ref.rms_norm(x.float().cpu(), w.float().cpu(), 1e-5)
    tol = 2e-2 if dtype != torch.float32 else 1e-5
    torch.testing.assert_close(out.float().cpu(), want, atol=tol, rtol=tol)


def test_rms_norm_fused_add():
    torch.manual_seed(0)
    x = torch.randn(2, 7, 512, dtype=torch.bfloat16, device=_dev())
    r = torch.randn(2, 7, 512, dtype=torch.bfloat16, device=_dev())
    w = torch.randn(512, dtype=torch.bfloat16, device=_dev())
    r_gpu = r.clone()
    out, new_r = ops.rms_norm(x, w, 1e-5, residual=r_gpu)
    want, want_r = ref.rms_norm(x.float().cpu(), w.float().cpu(), 1e-5, r.float().cpu())
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)
    torch.testing.assert_close(new_r.float().cpu(), want_r, atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("act", ["silu", "gelu"])
def test_act_mul(act):
    torch.manual_seed(0)
    g = torch.randn(3, 9, 1024, dtype=torch.bfloat16, device=_dev())
    u = torch.randn(3, 9, 1024, dtype=torch.bfloat16, device=_dev())
    fn = ops.silu_mul if act == "silu" else ops.gelu_tanh_mul
    rfn = ref.silu_mul if act == "silu" else ref.gelu_tanh_mul
    out = fn(g, u)
    want = rfn(g.float().cpu(), u.float().cpu())
    torch.testing.assert_close(out.float().cpu(), want, atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("D", [64, 128])
def test_rope(D):
    torch.manual_seed(0)
    B, Hq, Hkv, S = 2, 4, 2, 6
    q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device=_dev())
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=_dev())
    inv = 1.0 / (10000.0 ** (torch.arange(0, D, 2).float() / D))
    pos = torch.arange(S).float()
    emb = torch.cat([torch.outer(pos, inv)] * 2, dim=-1)
    cos = emb.cos().expand(B, S, D).contiguous().to(_dev())
    sin = emb.sin().expand(B, S, D).contiguous().to(_dev())
    q_ref, k_ref = ref.rope_apply(q.float().cpu(), k.float().cpu(), cos.cpu(), sin.cpu())
    q2, k2 = ops.rope_apply(q, k, cos, sin)
    torch.testing.assert_close(q2.float().cpu(), q_ref, atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(k2.float().cpu(), k_ref, atol=2e-2, rtol=2e-2)


def test_kv_cache_update():
    torch.manual_seed(0)
    Bc, Hkv, Smax, D = 4, 2, 64, 128
    kc = torch.zeros(Bc, Hkv, Smax, D, dtype=torch.bfloat16, device=_dev())
    vc = torch.zeros_like(kc)
    B, Sn = 2, 5
    k = torch.randn(B, Hkv, Sn, D, dtype=torch.bfloat16, device=_dev())
    v = torch.randn_like(k)
    pos = torch.tensor([[0, 1, 2, 3, 4], [10, 11, 12, 13, 14]], device=_dev())
    sid = torch.tensor([2, 0], device=_dev())
    ops.kv_cache_update(kc, vc, k, v, pos, sid)
    kc_ref = torch.zeros(Bc, Hkv, Smax, D)
    vc_ref = torch.zeros_like(kc_ref)
    ref.kv_cache_update(kc_ref, vc_ref, k.float().cpu(), v.float().cpu(), pos.cpu(), sid.cpu())
    torch.testing.assert_close(kc.float().cpu(), kc_ref, atol=1e-2, rtol=1e-2)
    torch.testing.assert_close(vc.float().cpu(), vc_ref, atol=1e-2, rtol=1e-2)


def test_block_kv_cache_update():
    torch.manual_seed(0)
    nb, bs, H, D = 8, 16, 2, 128
    kc = torch.zeros(nb, bs, H, D, dtype=torch.bfloat16, device=_dev())
    vc = torch.zeros_like(kc)
    n = 10
    k = torch.randn(n, H, D, dtype=torch.bfloat16, device=_dev())
    v = torch.randn_like(k)
    slots = torch.tensor([5, 6, 7, 32, 33, 34, 35, 100, -1, 101], device=_dev())
    ops.block_kv_cache_update(kc, vc, k, v, slots)
    kc_ref = torch.zeros(nb, bs, H, D)
    vc_ref = torch.zeros_like(kc_ref)
    ref.block_kv_cache_update(kc_ref, vc_ref, k.float().cpu(), v.float().cpu(), slots.cpu())
    torch.testing.assert_close(kc.float().cpu(), kc_ref, atol=1e-2, rtol=1e-2)


@pytest.mark.parametrize("D", [64, 128])
@pytest.mark.parametrize("gq,sq", [(4, 1), (8, 1), (1, 1), (4, 4), (3, 1)])
def test_attention_decode(D, gq, sq):
    torch.manual_seed(0)
    B, Hkv, Smax = 3, 2, 512
    Hq = Hkv * gq
    dev = _dev()
    kc = torch.randn(B + 1, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn_like(kc)
    q = torch.randn(B, Hq, sq, D, dtype=torch.bfloat16, device=dev)
    seq_lens = torch.tensor([17, 300, 512][:B], device=dev)
    seq_ids = torch.tensor([1, 0, 3][:B], device=dev)
    scale = 1.0 / math.sqrt(D)
    out = ops.attention_decode(q, kc, vc, seq_lens, seq_ids=seq_ids, scale=scale)
    want = ops.attention_decode(
        q.float().cpu(), kc.float().cpu(), vc.float().cpu(),
        seq_lens.cpu(), seq_ids=seq_ids.cpu(), scale=scale,
    )
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)


def test_attention_decode_sliding_window():
    torch.manual_seed(0)
    B, Hkv, Smax, D, gq = 2, 2, 256, 128, 4
    dev = _dev()
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn_like(kc)
    q = torch.randn(B, Hkv * gq, 1, D, dtype=torch.bfloat16, device=dev)
    seq_lens = torch.tensor([200, 64], device=dev)
    out = ops.attention_decode(q, kc, vc, seq_lens, sliding_window=128)
    want = ops.attention_decode(
        q.float().cpu(), kc.float().cpu(), vc.float().cpu(), seq_lens.cpu(),
        sliding_window=128,
    )
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)


def test_topk_topp_sample():
    torch.manual_seed(0)
    B, K = 5, 256
    vals = torch.sort(torch.randn(B, K, device=_dev()), dim=-1, descending=True).values
    top_k = torch.tensor([1, 4, 50, 256, 8], device=_dev(), dtype=torch.int32)
    top_p = torch.tensor([1.0, 0.9, 0.5, 0.95, 1.0], device=_dev())
    temp = torch.tensor([1.0, 0.7, 1.3, 1.0, 0.5], device=_dev())
    for rand_val in [0.0, 0.3, 0.77, 0.999]:
        rand = torch.full((B,), rand_val, device=_dev())
        got = ops.topk_topp_sample(vals, top_k, top_p, temp, rand)
        want = ref.topk_topp_sample(
            vals.cpu(), top_k.cpu(), top_p.cpu(), temp.cpu(), rand.cpu()
        )
        assert torch.equal(got.cpu(), want), f"rand={rand_val}: {got.cpu()} vs {want}"


def test_prefill_matches_reference():
    """Device prefill path (flash kernel when present, else masked matmul)
    vs fp32 reference."""
    torch.manual_seed(0)
    B, Hq, Hkv, S, D = 2, 4, 2, 256, 128
    dev = _dev()
    q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    out = ops.attention_prefill(q, k, v, causal=True)
    from neuronx_distributed_inference_amd.modules.attention.utils import causal_mask

    want = ref.sdpa(
        q.float().cpu(), k.float().cpu(), v.float().cpu(),
        mask=causal_mask(S, S), scale=1.0 / math.sqrt(D),
    )
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)


def test_e2e_tiny_decode_gpu_matches_cpu():
    """Tiny llama greedy decode on GPU (bf16 HIP kernels) vs CPU fp32:
    token-level agreement on a short horizon."""
    import neuronx_distributed_inference_amd.models.llama.modeling_llama as ml
    from neuronx_distributed_inference_amd.config import EngineConfig, OnDeviceSamplingConfig
    from neuronx_distributed_inference_amd.utils.hf_adapter import (
        GenerationConfig,
        HuggingFaceGenerationAdapter,
    )

    torch.manual_seed(0)

    def build(dtype, dev):
        ec = EngineConfig(
            batch_size=2, max_batch_size=2, seq_len=64, max_length=64,
            torch_dtype=dtype, tp_degree=1,
            on_device_sampling_config=OnDeviceSamplingConfig(top_k=1),
        )
        cfg = ml.LlamaInferenceConfig(
            neuron_config=ec, vocab_size=128, hidden_size=128, intermediate_size=256,
            num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
            max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
            hidden_act="silu", tie_word_embeddings=False,
        )
        return ml.LlamaForCausalLM(cfg).eval().to(dev)

    cpu = build("float32", "cpu")
    torch.manual_seed(7)
    with torch.no_grad():
        for p in cpu.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    gpu = build("bfloat16", _dev())
    gpu.model.load_state_dict(
        {k: v.to(torch.bfloat16) for k, v in cpu.model.state_dict().items()}, strict=False
    )
    ids = torch.randint(0, 128, (2, 8))
    got = HuggingFaceGenerationAdapter(gpu).generate(
        ids.to(_dev()), generation_config=GenerationConfig(max_new_tokens=8)
    )
    want = HuggingFaceGenerationAdapter(cpu).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=8)
    )
    # bf16 vs fp32: allow small divergence late in the horizon
    agree = (got.cpu() == want).float().mean()
    assert agree > 0.85, f"agreement {agree}: {got.cpu()} vs {want}"


def test_decode_batch_bucket_ladder_gpu():
    """Decode at partial batch replays a SUB-BUCKET graph (smallest rung
    covering B) and must produce the same tokens as the full-batch path
    (ref 2-D batch buckets, autobucketing.py:203-223)."""
    import neuronx_distributed_inference_amd.models.llama.modeling_llama as ml
    from neuronx_distributed_inference_amd.config import (
        EngineConfig,
        OnDeviceSamplingConfig,
    )
    from neuronx_distributed_inference_amd.utils.hf_adapter import (
        GenerationConfig,
        HuggingFaceGenerationAdapter,
    )

    def build(batch_buckets):
        torch.manual_seed(3)
        ec = EngineConfig(
            batch_size=8, max_batch_size=8, seq_len=64, max_length=64,
            torch_dtype="bfloat16", tp_degree=1,
            on_device_sampling_config=OnDeviceSamplingConfig(top_k=1),
            token_generation_batch_buckets=batch_buckets,
            enable_batch_bucketing=batch_buckets is not None,
        )
        cfg = ml.LlamaInferenceConfig(
            neuron_config=ec, vocab_size=128, hidden_size=128,
            intermediate_size=256, num_hidden_layers=2,
            num_attention_heads=4, num_key_value_heads=2,
            max_position_embeddings=128, rms_norm_eps=1e-5,
            rope_theta=10000.0, hidden_act="silu", tie_word_embeddings=False)
        m = ml.LlamaForCausalLM(cfg).eval().to(_dev())
        with torch.no_grad():
            for p in m.parameters():
                if p.dim() >= 2:
                    p.normal_(0, 0.1)
        return m

    ids = torch.randint(0, 128, (3, 8), generator=torch.Generator().manual_seed(5))
    ladder = build([2, 4, 8])
    ladder.warmup()  # creates the submodel wrappers + captures the ladder
    got = HuggingFaceGenerationAdapter(ladder).generate(
        ids.to(_dev()), generation_config=GenerationConfig(max_new_tokens=8))
    # B=3 must have used the 4-rung (not the full batch 8)
    tkg = ladder.token_generation_model
    assert tkg._graphs is not None and (4, 1) in tkg._graphs.runners
    single = build(None)
    single.config.neuron_config.enable_batch_bucketing = False
    single.warmup()
    want = HuggingFaceGenerationAdapter(single).generate(
        ids.to(_dev()), generation_config=GenerationConfig(max_new_tokens=8))
    assert torch.equal(got.cpu(), want.cpu())


@pytest.mark.parametrize("S", [96, 256, 1024])
@pytest.mark.parametrize("D", [64, 128])
def test_flash_prefill_shapes(S, D):
    torch.manual_seed(0)
    B, Hq, Hkv = 2, 4, 2
    dev = _dev()
    q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    out = ops.attention_prefill(q, k, v, causal=True)
    from neuronx_distributed_inference_amd.modules.attention.utils import causal_mask

    want = ref.sdpa(q.float().cpu(), k.float().cpu(), v.float().cpu(),
                    mask=causal_mask(S, S), scale=1.0 / math.sqrt(D))
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("S", [96, 512])
@pytest.mark.parametrize("D", [64, 128])
def test_flash_prefill_cached_matches_direct(S, D):
    """Cache-resident flash (CTE fused path) vs the plain flash kernel:
    K/V scattered into a larger cache with shuffled seq_ids, Smax > S."""
    torch.manual_seed(0)
    B, Hq, Hkv, Bc, Smax = 2, 4, 2, 4, S + 192
    dev = _dev()
    q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    kc = torch.randn(Bc, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn(Bc, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    seq_ids = torch.tensor([3, 1], dtype=torch.int32, device=dev)
    pos = torch.arange(S, dtype=torch.int32, device=dev).expand(B, S).contiguous()
    ops.kv_cache_update(kc, vc, k, v, pos, seq_ids)
    out = ops.attention_prefill_cached(q, kc, vc, seq_ids)
    want = ops.attention_prefill(q, k, v, causal=True)
    torch.testing.assert_close(out.float().cpu(), want.float().cpu(),
                               atol=3e-2, rtol=3e-2)


def test_flash_prefill_cached_ragged_pad_skip():
    """Ragged right-padded batch: pad positions -1 are skipped by the
    scatter; cached-flash output at real rows must match a per-row dense
    reference; stale cache rows beyond each length must not leak in."""
    torch.manual_seed(1)
    B, Hq, Hkv, S, D, Bc, Smax = 2, 4, 2, 160, 128, 3, 256
    dev = _dev()
    lens = [97, 160]
    q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    kc = torch.full((Bc, Hkv, Smax, D), 7.0, dtype=torch.bfloat16, device=dev)
    vc = torch.full((Bc, Hkv, Smax, D), 7.0, dtype=torch.bfloat16, device=dev)
    pos = torch.stack([
        torch.where(torch.arange(S) < L, torch.arange(S), torch.tensor(-1))
        for L in lens
    ]).to(dev, torch.int32)
    seq_ids = torch.tensor([1, 0], dtype=torch.int32, device=dev)
    ops.kv_cache_update(kc, vc, k, v, pos, seq_ids)
    out = ops.attention_prefill_cached(q, kc, vc, seq_ids)
    for b, L in enumerate(lens):
        want = ops.attention_prefill(q[b:b + 1, :, :L], k[b:b + 1, :, :L],
                                     v[b:b + 1, :, :L], causal=True)
        torch.testing.assert_close(out[b:b + 1, :, :L].float().cpu(),
                                   want.float().cpu(), atol=3e-2, rtol=3e-2)


def test_flash_prefill_sliding_window():
    torch.manual_seed(0)
    B, Hq, Hkv, S, D = 1, 2, 1, 512, 128
    dev = _dev()
    q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    out = ops.attention_prefill(q, k, v, causal=True, sliding_window=128)
    from neuronx_distributed_inference_amd.modules.attention.utils import (
        sliding_window_mask,
    )

    want = ref.sdpa(q.float().cpu(), k.float().cpu(), v.float().cpu(),
                    mask=sliding_window_mask(S, S, 128), scale=1.0 / math.sqrt(D))
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)


def test_flash_prefill_online_softmax_spike():
    """Force the online-softmax rescale branch (guide rule 26): one huge K row
    late in the sequence makes the running max jump after many tiles."""
    torch.manual_seed(0)
    B, Hq, Hkv, S, D = 1, 2, 2, 512, 128
    dev = _dev()
    q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev) * 0.1
    v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    k[:, :, 400, :] = q[:, :1, 480, :] * 3.0  # spike aligned with a late query
    out = ops.attention_prefill(q, k, v, causal=True)
    from neuronx_distributed_inference_amd.modules.attention.utils import causal_mask

    want = ref.sdpa(q.float().cpu(), k.float().cpu(), v.float().cpu(),
                    mask=causal_mask(S, S), scale=1.0 / math.sqrt(D))
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)


def test_qkv_rope_update_fused():
    torch.manual_seed(0)
    B, S, Hq, Hkv, D, Bc, Smax = 3, 1, 8, 2, 128, 4, 64
    dev = _dev()
    qkv = torch.randn(B, S, (Hq + 2 * Hkv) * D, dtype=torch.bfloat16, device=dev)
    inv = 1.0 / (10000.0 ** (torch.arange(0, D, 2).float() / D))
    emb = torch.cat([torch.outer(torch.arange(Smax).float(), inv)] * 2, dim=-1)
    pos = torch.tensor([[5], [0], [63]], device=dev)
    cos = emb.cos().to(dev)[pos.long().clamp(min=0)]
    sin = emb.sin().to(dev)[pos.long().clamp(min=0)]
    kc = torch.zeros(Bc, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    vc = torch.zeros_like(kc)
    sid = torch.tensor([1, 0, 3], device=dev)
    q = ops.qkv_rope_update(qkv, cos, sin, kc, vc, pos, sid, Hq, Hkv)

    kc_ref = torch.zeros(Bc, Hkv, Smax, D)
    vc_ref = torch.zeros_like(kc_ref)
    q_ref = ops.qkv_rope_update(qkv.float().cpu(), cos.cpu(), sin.cpu(),
                                kc_ref, vc_ref, pos.cpu(), sid.cpu(), Hq, Hkv)
    torch.testing.assert_close(q.float().cpu(), q_ref, atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(kc.float().cpu(), kc_ref, atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(vc.float().cpu(), vc_ref, atol=2e-2, rtol=2e-2)


def test_silu_mul_fused_gpu():
    torch.manual_seed(0)
    y = torch.randn(4, 3, 512, dtype=torch.bfloat16, device=_dev())
    out = ops.silu_mul_fused(y)
    want = ops.silu_mul_fused(y.float().cpu())
    torch.testing.assert_close(out.float().cpu(), want, atol=2e-2, rtol=2e-2)


def test_attention_decode_paged_gpu():
    torch.manual_seed(0)
    B, Hkv, gq, D = 2, 2, 4, 128
    nb, bs = 32, 16
    dev = _dev()
    kc = torch.randn(nb, bs, Hkv, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn_like(kc)
    q = torch.randn(B, Hkv * gq, 1, D, dtype=torch.bfloat16, device=dev)
    bt = torch.stack([
        torch.arange(0, 16, 2), torch.arange(1, 17, 2),
    ]).to(dev)
    seq_lens = torch.tensor([37, 100], device=dev)
    out = ops.attention_decode_paged(q, kc, vc, seq_lens, bt)
    want = ops.attention_decode_paged(
        q.float().cpu(), kc.float().cpu(), vc.float().cpu(), seq_lens.cpu(), bt.cpu()
    )
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)


def test_attention_decode_fp8_cache():
    """Decode attention reading an OCP-e4m3 KV cache vs the fp32 reference
    on the dequantized values."""
    torch.manual_seed(0)
    B, Hkv, gq, Smax, D = 2, 2, 4, 256, 128
    dev = _dev()
    kc_f = torch.randn(B, Hkv, Smax, D, device=dev) * 2
    vc_f = torch.randn(B, Hkv, Smax, D, device=dev) * 2
    kc = kc_f.to(torch.float8_e4m3fn)
    vc = vc_f.to(torch.float8_e4m3fn)
    q = torch.randn(B, Hkv * gq, 1, D, dtype=torch.bfloat16, device=dev)
    seq_lens = torch.tensor([100, 256], device=dev)
    out = ops.attention_decode(q, kc, vc, seq_lens)
    want = ops.attention_decode(
        q.float().cpu(), kc.cpu().float(), vc.cpu().float(), seq_lens.cpu()
    )
    torch.testing.assert_close(out.float().cpu(), want, atol=5e-2, rtol=5e-2)


def test_kv_update_fp8_cache():
    torch.manual_seed(0)
    Bc, Hkv, Smax, D = 2, 2, 32, 128
    dev = _dev()
    kc = torch.zeros(Bc, Hkv, Smax, D, device=dev).to(torch.float8_e4m3fn)
    vc = torch.zeros_like(kc)
    k = torch.randn(2, Hkv, 3, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn_like(k)
    pos = torch.tensor([[0, 1, 2], [5, 6, 7]], device=dev)
    sid = torch.tensor([0, 1], device=dev)
    ops.kv_cache_update(kc, vc, k, v, pos, sid)
    torch.testing.assert_close(
        kc[0, :, 0:3].cpu().float(), k[0].float().cpu(), atol=0.15, rtol=0.15
    )


def test_attention_decode_sinks_gpu():
    torch.manual_seed(0)
    B, Hkv, gq, Smax, D = 2, 2, 4, 128, 128
    dev = _dev()
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn_like(kc)
    q = torch.randn(B, Hkv * gq, 1, D, dtype=torch.bfloat16, device=dev)
    sinks = torch.randn(Hkv * gq, device=dev)
    seq_lens = torch.tensor([64, 128], device=dev)
    out = ops.attention_decode(q, kc, vc, seq_lens, sinks=sinks)
    want = ops.attention_decode(
        q.float().cpu(), kc.float().cpu(), vc.float().cpu(), seq_lens.cpu(),
        sinks=sinks.cpu(),
    )
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)


def test_flash_prefill_sinks_gpu():
    torch.manual_seed(0)
    B, Hq, Hkv, S, D = 1, 4, 2, 256, 128
    dev = _dev()
    q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    sinks = torch.randn(Hq, device=dev)
    out = ops.attention_prefill(q, k, v, causal=True, sinks=sinks)
    from neuronx_distributed_inference_amd.modules.attention.utils import causal_mask

    want = ref.sdpa(q.float().cpu(), k.float().cpu(), v.float().cpu(),
                    mask=causal_mask(S, S), scale=1.0 / math.sqrt(D),
                    sinks=sinks.cpu())
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("M", [1, 64, 512])
def test_lt_linear_matches_torch(M):
    """Autotuned hipBLASLt linear vs F.linear (same math, library GEMM)."""
    if not ops.lt_linear_enabled():
        pytest.skip("hipblaslt unavailable")
    torch.manual_seed(0)
    dev = _dev()
    N, K = 1024, 512
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.05
    b = torch.randn(N, dtype=torch.bfloat16, device=dev)
    got = ops.lt_linear(x, w, b)
    want = torch.nn.functional.linear(x, w, b)
    torch.testing.assert_close(got.float(), want.float(), atol=2e-2, rtol=2e-2)
    got2 = ops.lt_linear(x.view(1, M, K), w)  # 3D input, no bias
    want2 = torch.nn.functional.linear(x.view(1, M, K), w)
    torch.testing.assert_close(got2.float(), want2.float(), atol=2e-2, rtol=2e-2)


def test_chunked_attention_prefill_kernel():
    """Negative sliding_window = Llama4 chunked attention in the flash
    kernel: must match the chunked-mask fp32 reference."""
    torch.manual_seed(0)
    B, Hq, Hkv, S, D, C = 2, 4, 2, 512, 128, 96
    dev = _dev()
    q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    out = ops.attention_prefill(q, k, v, causal=True, sliding_window=-C)
    from neuronx_distributed_inference_amd.modules.attention.utils import (
        chunked_attention_mask,
    )

    want = ref.sdpa(q.float().cpu(), k.float().cpu(), v.float().cpu(),
                    mask=chunked_attention_mask(S, S, C),
                    scale=1.0 / math.sqrt(D))
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)


def test_chunked_attention_decode_kernel():
    """Chunked decode: queries at positions crossing chunk boundaries must
    only see their own chunk."""
    torch.manual_seed(0)
    B, Hq, Hkv, Smax, D, C = 2, 4, 2, 256, 128, 32
    dev = _dev()
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    q = torch.randn(B, Hq, 1, D, dtype=torch.bfloat16, device=dev)
    for plen in (31, 32, 33, 97, 200):
        lens = torch.tensor([plen, plen - 7], dtype=torch.int32, device=dev)
        out = ops.attention_decode(q, kc, vc, lens, sliding_window=-C)
        want = ops.attention_decode(q.cpu(), kc.float().cpu(), vc.float().cpu(),
                                    lens.cpu(), sliding_window=-C)
        torch.testing.assert_close(out.float().cpu(), want.float(),
                                   atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("M", [16, 100, 128])
def test_skinny_gemm_matches_torch(M):
    """Experimental skinny GEMM numerics vs F.linear (incl. padded lda and
    non-multiple-of-16 M)."""
    from neuronx_distributed_inference_amd.ops import _nxdi_C as C

    torch.manual_seed(0)
    dev = _dev()
    N, K = 512, 1024
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.1
    b = torch.randn(N, dtype=torch.bfloat16, device=dev)
    y = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
    C.skinny_gemm(y, x, w, b)
    want = torch.nn.functional.linear(x, w, b)
    torch.testing.assert_close(y.float(), want.float(), atol=2e-2, rtol=2e-2)
    wp = torch.empty(N, K + 64, dtype=torch.bfloat16, device=dev)[:, :K]
    wp.copy_(w)
    C.skinny_gemm(y, x, wp)
    want = torch.nn.functional.linear(x, w)
    torch.testing.assert_close(y.float(), want.float(), atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("M", [16, 50, 64, 100, 128])
def test_skinny_gemm3_matches_torch(M):
    """v3 skinny GEMM (full-K register accumulation, depth-3 prefetch,
    direct bf16 out) numerics vs F.linear, with and without bias."""
    from neuronx_distributed_inference_amd.ops import _nxdi_C as C

    torch.manual_seed(0)
    dev = _dev()
    N, K = 1536, 1024
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.1
    b = torch.randn(N, dtype=torch.bfloat16, device=dev)
    y = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
    C.skinny_gemm3(y, x, w, b)
    want = torch.nn.functional.linear(x, w, b)
    torch.testing.assert_close(y.float(), want.float(), atol=2e-2, rtol=2e-2)
    C.skinny_gemm3(y, x, w)
    want = torch.nn.functional.linear(x, w)
    torch.testing.assert_close(y.float(), want.float(), atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("M,splitk", [(64, 1), (64, 2), (128, 2), (256, 1),
                                      (256, 2), (256, 4), (100, 2), (8, 1)])
def test_skinny_gemm5_matches_torch(M, splitk):
    """v5 skinny GEMM (all-glds staging, raw-barrier 3-buffer ring,
    counted vmcnt, split-K fold) numerics vs F.linear, with bias, with the
    fp32 per-row scale (rmsnorm fold), and at non-bucket M."""
    from neuronx_distributed_inference_amd import ops

    torch.manual_seed(0)
    dev = _dev()
    N, K = 1536, 1024
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.1
    b = torch.randn(N, dtype=torch.bfloat16, device=dev)
    y = ops.skinny_gemm5(x, w, b, splitk=splitk)
    want = torch.nn.functional.linear(x, w, b)
    torch.testing.assert_close(y.float(), want.float(), atol=2e-2, rtol=2e-2)
    y = ops.skinny_gemm5(x, w, splitk=splitk)
    want = torch.nn.functional.linear(x, w)
    torch.testing.assert_close(y.float(), want.float(), atol=2e-2, rtol=2e-2)
    rs = torch.rand(M, dtype=torch.float32, device=dev) + 0.5
    y = ops.skinny_gemm5(x, w, rs=rs, splitk=splitk)
    want = rs.view(M, 1) * torch.nn.functional.linear(x.float(), w.float())
    torch.testing.assert_close(y.float(), want, atol=3e-2, rtol=3e-2)


def test_skinny_gemm5_decode_shapes():
    """The real decode projection shapes (8B TP=1): qkv N=6144,
    o N=4096, K=4096 — every splitk candidate must agree with F.linear."""
    from neuronx_distributed_inference_amd import ops

    torch.manual_seed(1)
    dev = _dev()
    M = 256
    for N, K in ((6144, 4096), (4096, 4096)):
        x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.05
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.05
        want = torch.nn.functional.linear(x, w)
        for mt, sk in ops._sg5_candidates(M, N, K):
            y = ops.skinny_gemm5(x, w, splitk=sk, mt=mt)
            torch.testing.assert_close(y.float(), want.float(),
                                       atol=2e-2, rtol=2e-2,
                                       msg=f"N={N} K={K} mt={mt} splitk={sk}")


def test_decode_linear_dispatcher():
    """decode_linear autotunes once per shape and replays the winner; the
    result must match F.linear either way."""
    from neuronx_distributed_inference_amd import ops

    torch.manual_seed(2)
    dev = _dev()
    x = torch.randn(2, 64, 512, dtype=torch.bfloat16, device=dev) * 0.1
    w = torch.randn(256, 512, dtype=torch.bfloat16, device=dev) * 0.1
    y = ops.decode_linear(x, w)
    key = (128, 256, 512, False, torch.bfloat16)
    assert key in ops._sg5_state["choice"]
    y2 = ops.decode_linear(x, w)  # replay path
    want = torch.nn.functional.linear(x, w)
    torch.testing.assert_close(y.float(), want.float(), atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(y2, y)


def test_attention_decode_lse_matches_cpu():
    """lse output of the decode kernel vs the fp32 logsumexp reference
    (flash-decoding combine handle), incl. empty (-1) query rows."""
    torch.manual_seed(0)
    B, Hq, Hkv, Smax, D = 2, 4, 2, 96, 64
    dev = _dev()
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    q = torch.randn(B, Hq, 1, D, dtype=torch.bfloat16, device=dev)
    pos = torch.tensor([[57], [-1]], dtype=torch.int32, device=dev)
    out, lse = ops.attention_decode_lse(q, kc, vc, pos)
    want_o, want_l = ops.attention_decode_lse(
        q.float().cpu(), kc.float().cpu(), vc.float().cpu(), pos.cpu())
    torch.testing.assert_close(out.float().cpu(), want_o.float(),
                               atol=3e-2, rtol=3e-2)
    assert torch.isinf(lse[1]).all() and (lse[1] < 0).all()
    torch.testing.assert_close(lse[0].cpu(), want_l[0], atol=2e-2, rtol=2e-2)
    assert (out[1].float() == 0).all()


def test_ring_cache_decode_kernel():
    """Ring (compact sliding-window) decode vs the CPU true-position
    reference, incl. multi-query (speculation) shapes."""
    torch.manual_seed(0)
    B, Hq, Hkv, W, D = 2, 4, 2, 32, 64
    dev = _dev()
    win = W
    kc = torch.randn(B, Hkv, W, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn(B, Hkv, W, D, dtype=torch.bfloat16, device=dev)
    for Sq, plen in ((1, 20), (1, 57), (3, 45)):
        q = torch.randn(B, Hq, Sq, D, dtype=torch.bfloat16, device=dev)
        pos = (torch.tensor([plen, plen - 5]).view(B, 1)
               + torch.arange(Sq).view(1, Sq)).to(dev, torch.int32)
        out = ops.attention_decode(q, kc, vc, pos, sliding_window=win, ring=True)
        want = ops.attention_decode(q.float().cpu(), kc.float().cpu(),
                                    vc.float().cpu(), pos.cpu(),
                                    sliding_window=win, ring=True)
        torch.testing.assert_close(out.float().cpu(), want.float(),
                                   atol=3e-2, rtol=3e-2, msg=f"Sq={Sq} p={plen}")
    # ragged chunk: trailing -1 pads must not zero the valid queries
    # (pmax = max over VALID positions, matching the CPU amax semantics)
    Sq = 4
    q = torch.randn(B, Hq, Sq, D, dtype=torch.bfloat16, device=dev)
    pos = torch.tensor([[45, 46, -1, -1], [38, -1, -1, -1]],
                       dtype=torch.int32, device=dev)
    out = ops.attention_decode(q, kc, vc, pos, sliding_window=win, ring=True)
    want = ops.attention_decode(q.float().cpu(), kc.float().cpu(),
                                vc.float().cpu(), pos.cpu(),
                                sliding_window=win, ring=True)
    torch.testing.assert_close(out.float().cpu(), want.float(),
                               atol=3e-2, rtol=3e-2)
    assert out[0, :, 0].abs().sum() > 0  # valid rows nonzero
    assert (out[0, :, 2:].float() == 0).all()  # pad rows zero


@torch.inference_mode()
def test_gpu_model_variants_match_cpu():
    """GPU e2e for the newer model features: ring SWA cache and Llama4
    chunked/NoPE layers must generate the same tokens as the CPU path."""
    import copy

    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.llama4.modeling_llama4 import (
        Llama4ForCausalLM, Llama4InferenceConfig,
    )
    from neuronx_distributed_inference_amd.models.mistral.modeling_mistral import (
        MistralForCausalLM, MistralInferenceConfig,
    )
    from neuronx_distributed_inference_amd.utils.hf_adapter import (
        GenerationConfig, HuggingFaceGenerationAdapter,
    )

    dev = _dev()
    gc = GenerationConfig(max_new_tokens=8)

    # ring SWA mistral
    ec = EngineConfig(batch_size=2, max_batch_size=2, seq_len=64, max_length=64,
                      torch_dtype="float32", tp_degree=1, compact_swa_kv=True)
    cfg = MistralInferenceConfig(
        neuron_config=ec, vocab_size=128, hidden_size=128, intermediate_size=256,
        num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=1,
        max_position_embeddings=128, rms_norm_eps=1e-5, sliding_window=16,
        tie_word_embeddings=False)
    torch.manual_seed(4)
    cpu = MistralForCausalLM(cfg).eval()
    with torch.no_grad():
        for p in cpu.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    ids = torch.randint(0, 128, (2, 24))
    want = HuggingFaceGenerationAdapter(cpu).generate(ids, generation_config=gc)
    ec2 = copy.deepcopy(ec)
    ec2.torch_dtype = torch.bfloat16
    cfg2 = copy.deepcopy(cfg)
    cfg2.neuron_config = ec2
    gpu = MistralForCausalLM(cfg2).eval()
    sd = {k: v.bfloat16() for k, v in cpu.model.state_dict().items()
          if not k.startswith("kv_mgr.")}
    gpu.model.load_state_dict(sd, strict=False)
    gpu = gpu.to(dev)
    got = HuggingFaceGenerationAdapter(gpu).generate(
        ids.to(dev), generation_config=gc)
    agree = (got.cpu() == want).float().mean()
    assert agree > 0.8, (agree, got.cpu().tolist(), want.tolist())

    # llama4 text: chunked + NoPE + MoE
    ecl = EngineConfig(batch_size=1, max_batch_size=1, seq_len=64, max_length=64,
                       torch_dtype="float32", tp_degree=1)
    cfgl = Llama4InferenceConfig(
        neuron_config=ecl, vocab_size=128, hidden_size=128, intermediate_size=256,
        intermediate_size_mlp=256, num_hidden_layers=4, num_attention_heads=2,
        num_key_value_heads=1, head_dim=64, max_position_embeddings=128,
        num_local_experts=2, num_experts_per_tok=1, interleave_moe_layer_step=2,
        no_rope_layer_interval=2, attention_chunk_size=8, use_qk_norm=True,
        attn_temperature_tuning=True, floor_scale=8, attn_scale=0.1,
        rms_norm_eps=1e-5, tie_word_embeddings=False)
    torch.manual_seed(5)
    cpu = Llama4ForCausalLM(cfgl).eval()
    with torch.no_grad():
        for p in cpu.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    ids = torch.randint(0, 128, (1, 20))
    want = HuggingFaceGenerationAdapter(cpu).generate(ids, generation_config=gc)
    ecg = copy.deepcopy(ecl)
    ecg.torch_dtype = torch.bfloat16
    cfgg = copy.deepcopy(cfgl)
    cfgg.neuron_config = ecg
    gpu = Llama4ForCausalLM(cfgg).eval()
    sd = {k: v.bfloat16() for k, v in cpu.model.state_dict().items()
          if not k.startswith("kv_mgr.")}
    gpu.model.load_state_dict(sd, strict=False)
    gpu = gpu.to(dev)
    got = HuggingFaceGenerationAdapter(gpu).generate(
        ids.to(dev), generation_config=gc)
    agree = (got.cpu() == want).float().mean()
    assert agree > 0.8, (agree, got.cpu().tolist(), want.tolist())


def test_fp8_quant_kernels_match_torch():
    """fp8_quant (amax+cast) and fp8_epilogue vs the torch formulation."""
    from neuronx_distributed_inference_amd.ops import _nxdi_C as C

    torch.manual_seed(0)
    dev = _dev()
    x = torch.randn(64, 512, dtype=torch.bfloat16, device=dev) * 3
    x_q = torch.empty_like(x, dtype=torch.float8_e4m3fn)
    scale = torch.empty(1, 1, dtype=torch.float32, device=dev)
    amax = torch.empty(128, dtype=torch.float32, device=dev)
    C.fp8_quant(x, x_q, scale, amax)
    want_scale = x.float().abs().amax().clamp(min=1e-8) / 448.0
    torch.testing.assert_close(scale.view(()), want_scale, atol=1e-6, rtol=1e-5)
    want_q = (x.float() / want_scale).clamp(-448, 448).to(torch.float8_e4m3fn)
    torch.testing.assert_close(x_q.float(), want_q.float())
    # epilogue
    y32 = torch.randn(64, 128, dtype=torch.float32, device=dev)
    ws = torch.rand(128, dtype=torch.float32, device=dev) + 0.5
    b = torch.randn(128, dtype=torch.bfloat16, device=dev)
    out = torch.empty(64, 128, dtype=torch.bfloat16, device=dev)
    C.fp8_epilogue(out, y32, ws, b)
    want = (y32 * ws + b.float()).to(torch.bfloat16)
    torch.testing.assert_close(out.float(), want.float(), atol=2e-2, rtol=2e-2)


def test_moe_grouped_gemm_kernel():
    """Grouped expert GEMM vs per-expert torch linear on GPU."""
    torch.manual_seed(0)
    dev = _dev()
    E, N, K = 4, 256, 512
    idx = torch.randint(0, E, (100,), device=dev)
    order, dest, expert_ids, gtab, Mpad = ops.moe_grouped_tables(idx, E)
    x_rows = torch.randn(100, K, dtype=torch.bfloat16, device=dev) * 0.1
    w = torch.randn(E, N, K, dtype=torch.bfloat16, device=dev) * 0.1
    x_sorted = torch.zeros(Mpad, K, dtype=torch.bfloat16, device=dev)
    x_sorted.index_copy_(0, dest, x_rows)
    y = ops.moe_grouped_gemm(x_sorted, w, expert_ids)
    got = y.index_select(0, dest)
    want = torch.empty_like(got)
    se = idx[order]
    for e in range(E):
        m = se == e
        want[m] = x_sorted[dest[m]] @ w[e].t()
    torch.testing.assert_close(got.float(), want.float(), atol=2e-2, rtol=2e-2)


def test_moe_grouped_matches_dense_gpu():
    """Full MoE layer: grouped path vs dense bmm path on GPU bf16, plus a
    timing line at decode batch 256 (Mixtral-like shapes, TP-degree 8
    shard sizes to fit the test box quickly)."""
    from neuronx_distributed_inference_amd.modules.moe import ExpertMLPs

    torch.manual_seed(0)
    dev = _dev()
    m = ExpertMLPs(num_experts=8, hidden_size=4096, intermediate_size=1792,
                   tp_degree=1, dtype=torch.bfloat16).to(dev)
    with torch.no_grad():
        for p in m.parameters():
            p.normal_(0, 0.03)
    T, top_k = 256, 2
    x = torch.randn(T, 4096, dtype=torch.bfloat16, device=dev) * 0.3
    w, idx = torch.softmax(torch.randn(T, 8, device=dev), -1).topk(top_k, dim=-1)
    got = m._forward_grouped(x, w, idx)
    want = m._forward_dense(x, w, idx)
    torch.testing.assert_close(got.float(), want.float(), atol=5e-2, rtol=5e-2)

    def t(fn, n=20):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
        e0.record()
        for _ in range(n):
            fn()
        e1.record()
        torch.cuda.synchronize()
        return e0.elapsed_time(e1) / n * 1000

    tg = t(lambda: m._forward_grouped(x, w, idx))
    td = t(lambda: m._forward_dense(x, w, idx))
    print(f"\nMoE T={T} grouped={tg:.0f}us dense={td:.0f}us ratio={tg/td:.2f}")


def test_moe_grouped_mx4_kernel_gpu():
    """MXFP4 grouped kernel (in-LDS dequant) vs the CPU dequant reference
    on the same packed weights, with per-expert bias."""
    torch.manual_seed(0)
    dev = _dev()
    E, N, K = 4, 256, 512
    w = torch.randn(E, N, K) * 0.2
    wq, ws = ops.mx4_quantize(w)
    bias = torch.randn(E, N, dtype=torch.bfloat16) * 0.1
    idx = torch.randint(0, E, (90,))
    order, dest, expert_ids, gtab, Mpad = ops.moe_grouped_tables(idx, E)
    x_rows = torch.randn(90, K, dtype=torch.bfloat16) * 0.1
    x_sorted = torch.zeros(Mpad, K, dtype=torch.bfloat16)
    x_sorted.index_copy_(0, dest, x_rows)
    want = ops.moe_grouped_gemm_mx4(x_sorted, wq, ws, expert_ids, bias=bias)
    got = ops.moe_grouped_gemm_mx4(
        x_sorted.to(dev), wq.to(dev), ws.to(dev), expert_ids.to(dev),
        bias=bias.to(dev))
    torch.testing.assert_close(got.index_select(0, dest.to(dev)).float().cpu(),
                               want.index_select(0, dest).float(),
                               atol=3e-2, rtol=3e-2)


def test_gpt_oss_mxfp4_gpu_matches_dequant():
    """GPT-OSS expert layer on GPU: MXFP4 grouped path vs the bf16 dense
    weights produced by dequantizing the same packing."""
    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.gpt_oss.modeling_gpt_oss import (
        GptOssExperts,
        GptOssInferenceConfig,
    )

    torch.manual_seed(1)
    dev = _dev()
    E, H, I = 4, 256, 256

    def mkcfg(mx):
        ec = EngineConfig(batch_size=2, seq_len=32, max_length=32,
                          torch_dtype="bfloat16", tp_degree=1, moe_mxfp4=mx)
        return GptOssInferenceConfig(
            neuron_config=ec, vocab_size=64, hidden_size=H,
            intermediate_size=I, num_hidden_layers=1, num_attention_heads=4,
            num_key_value_heads=2, num_local_experts=E, num_experts_per_tok=2,
            max_position_embeddings=64, rms_norm_eps=1e-5, rope_theta=10000.0,
            layer_types=["full_attention"], tie_word_embeddings=False)

    mx = GptOssExperts(mkcfg(True))
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
    dense = GptOssExperts(mkcfg(False))
    with torch.no_grad():
        dense.gate_up_proj.copy_(ops.mx4_dequant(mx.gate_up_q, mx.gate_up_s))
        dense.down_proj.copy_(ops.mx4_dequant(mx.down_q, mx.down_s))
        dense.gate_up_proj_bias.copy_(mx.gate_up_proj_bias)
        dense.down_proj_bias.copy_(mx.down_proj_bias)
    mx, dense = mx.to(dev), dense.to(dev)
    T = 64
    x = (torch.randn(T, H) * 0.3).to(dev, torch.bfloat16)
    wts = torch.softmax(torch.randn(T, 2, device=dev), dim=-1)
    idx = torch.randint(0, E, (T, 2), device=dev)
    got = mx(x, wts, idx)
    want = dense(x, wts, idx)
    torch.testing.assert_close(got.float(), want.float(), atol=4e-2, rtol=4e-2)


@pytest.mark.parametrize("gq", [12, 16])
def test_attention_decode_gq_over_8(gq):
    """GQA ratio > 8 (odd shards, e.g. 96q/8kv): split-group dispatch
    must match the CPU reference."""
    torch.manual_seed(0)
    B, Hkv, Smax, D = 2, 2, 256, 128
    dev = _dev()
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn_like(kc)
    q = torch.randn(B, Hkv * gq, 1, D, dtype=torch.bfloat16, device=dev)
    seq_lens = torch.tensor([100, 256], device=dev)
    out = ops.attention_decode(q, kc, vc, seq_lens)
    want = ops.attention_decode(q.float().cpu(), kc.float().cpu(),
                                vc.float().cpu(), seq_lens.cpu())
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)
    # sinks variant exercises the per-group sink slice
    sinks = torch.randn(Hkv * gq, device=dev)
    out = ops.attention_decode(q, kc, vc, seq_lens, sinks=sinks)
    want = ops.attention_decode(q.float().cpu(), kc.float().cpu(),
                                vc.float().cpu(), seq_lens.cpu(),
                                sinks=sinks.cpu())
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)


def test_long_context_decode_32k():
    """32k-context decode numerics (small-shape-only
    GPU tests): split-KV kernel vs fp32 reference at Smax=32768."""
    torch.manual_seed(0)
    B, Hkv, gq, Smax, D = 1, 2, 4, 32768, 128
    dev = _dev()
    kc = (torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev) * 0.3)
    vc = torch.randn_like(kc) * 0.3
    q = torch.randn(B, Hkv * gq, 1, D, dtype=torch.bfloat16, device=dev)
    seq_lens = torch.tensor([32768], device=dev)
    out = ops.attention_decode(q, kc, vc, seq_lens)
    want = ops.attention_decode(q.float().cpu(), kc.float().cpu(),
                                vc.float().cpu(), seq_lens.cpu())
    torch.testing.assert_close(out.float().cpu(), want, atol=3e-2, rtol=3e-2)


def test_long_context_prefill_8k():
    """8k-sequence flash prefill vs the fp32 masked-matmul reference
    computed ON DEVICE (a host fp32 reference at 8k^2 scores is too
    slow; the device ref path uses library GEMMs + explicit mask)."""
    torch.manual_seed(0)
    B, Hq, Hkv, S, D = 1, 2, 1, 8192, 128
    dev = _dev()
    q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device=dev) * 0.3
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev) * 0.3
    v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev) * 0.3
    out = ops.attention_prefill(q, k, v, causal=True)
    from neuronx_distributed_inference_amd.modules.attention.utils import (
        causal_mask,
    )

    want = ref.sdpa(q.float(), k.float(), v.float(),
                    mask=causal_mask(S, S, device=dev),
                    scale=1.0 / math.sqrt(D))
    torch.testing.assert_close(out.float(), want, atol=3e-2, rtol=3e-2)


def test_moe_tables_fused_matches_torch_chain():
    """The fused one-launch tables kernel vs the torch-op chain: same
    expert_ids, same per-expert assignment SETS (within-expert order is
    an arbitrary permutation by design — atomic cursors), consistent
    (order, dest) pairing."""
    torch.manual_seed(3)
    dev = _dev()
    for E, Ttk in ((8, 512), (32, 100), (4, 64), (128, 4096)):
        idx = torch.randint(0, E, (Ttk,), device=dev)
        order_g, dest_g, eids_g, gtab_g, Mpad_g = ops.moe_grouped_tables(idx, E)
        order_c, dest_c, eids_c, gtab_c, Mpad_c = ops.moe_grouped_tables(idx.cpu(), E)
        assert Mpad_g == Mpad_c
        assert torch.equal(eids_g.cpu(), eids_c)
        og, dg = order_g.cpu(), dest_g.cpu()
        # sorted-by-expert invariant + same assignment multiset per expert
        se_g, se_c = idx.cpu()[og], idx.cpu()[order_c]
        assert torch.equal(se_g, se_c)  # non-decreasing & equal counts
        for e in range(E):
            m = se_c == e
            assert set(og[m].tolist()) == set(order_c[m].tolist())
            # dest within an expert: same padded-row SET
            assert set(dg[m].tolist()) == set(dest_c[m].tolist())
        # (order, dest) consistency: round-trip scatter+gather is identity
        x = torch.randn(Ttk, 16, device=dev)
        xs = torch.zeros(Mpad_g, 16, device=dev)
        xs.index_copy_(0, dest_g, x.index_select(0, order_g))
        back = xs.index_select(0, dest_g)
        torch.testing.assert_close(back, x.index_select(0, order_g))


@pytest.mark.parametrize("bn", [64, 128, 192, 256])
def test_moe_grouped_gemm_bn_tiles(bn):
    """Every column-tile width against the per-expert torch reference."""
    torch.manual_seed(1)
    dev = _dev()
    E, N, K = 4, 768, 512
    idx = torch.randint(0, E, (200,), device=dev)
    order, dest, expert_ids, gtab, Mpad = ops.moe_grouped_tables(idx, E)
    x_rows = torch.randn(200, K, dtype=torch.bfloat16, device=dev) * 0.1
    w = torch.randn(E, N, K, dtype=torch.bfloat16, device=dev) * 0.1
    x_sorted = torch.zeros(Mpad, K, dtype=torch.bfloat16, device=dev)
    x_sorted.index_copy_(0, dest, x_rows.index_select(0, order))
    y = ops.moe_grouped_gemm(x_sorted, w, expert_ids, bn=bn)
    got = y.index_select(0, dest)
    se = idx[order]
    want = torch.empty_like(got)
    for e in range(E):
        m = se == e
        want[m] = x_sorted[dest[m]] @ w[e].t()
    torch.testing.assert_close(got.float(), want.float(), atol=2e-2, rtol=2e-2)


def test_moe_grouped_gemm_fusions_gpu():
    """Fused x-gather + fused combine epilogue vs the materialized
    pipeline on GPU."""
    torch.manual_seed(5)
    dev = _dev()
    E, T, top_k, K, N = 8, 200, 2, 512, 768
    idx = torch.randint(0, E, (T * top_k,), device=dev)
    order, dest, eids, gtab, Mpad = ops.moe_grouped_tables(idx, E, top_k)
    x = torch.randn(T, K, dtype=torch.bfloat16, device=dev) * 0.1
    w = torch.randn(E, N, K, dtype=torch.bfloat16, device=dev) * 0.1
    gatew = torch.rand(T * top_k, device=dev)[order]

    tok = order // top_k
    xs = torch.zeros(Mpad, K, dtype=torch.bfloat16, device=dev)
    xs.index_copy_(0, dest, x.index_select(0, tok))
    y = ops.moe_grouped_gemm(xs, w, eids)
    contrib = y.index_select(0, dest) * gatew.to(y.dtype).unsqueeze(-1)
    want = torch.zeros(T, N, dtype=torch.bfloat16, device=dev)
    want.index_add_(0, tok, contrib)

    y2 = ops.moe_grouped_gemm(x, w, eids, x_gather=gtab)
    torch.testing.assert_close(y2.index_select(0, dest).float(),
                               y.index_select(0, dest).float(),
                               atol=2e-2, rtol=2e-2)
    gw_tab = torch.zeros(Mpad, dtype=torch.float32, device=dev)
    gw_tab[dest] = gatew.float()
    out32 = torch.zeros(T, N, dtype=torch.float32, device=dev)
    got = ops.moe_grouped_gemm(xs, w, eids, combine_out=out32,
                               combine_scale=gw_tab, combine_gather=gtab)
    torch.testing.assert_close(got.to(torch.bfloat16).float(), want.float(),
                               atol=5e-2, rtol=5e-2)


def test_moe_grouped_fp8_kernel_gpu():
    """fp8-e4m3 grouped GEMM (all-glds staging, LDS-read dequant,
    epilogue per-channel scale) vs the CPU dequant reference, incl. the
    gather/combine fusions."""
    torch.manual_seed(7)
    dev = _dev()
    E, T, top_k, K, N = 4, 150, 2, 512, 768
    idx = torch.randint(0, E, (T * top_k,), device=dev)
    order, dest, eids, gtab, Mpad = ops.moe_grouped_tables(idx, E, top_k)
    x = torch.randn(T, K, dtype=torch.bfloat16, device=dev) * 0.3
    w = torch.randn(E, N, K) * 0.1
    wq, ws = ops.fp8_moe_quantize(w)
    wq, ws = wq.to(dev), ws.to(dev)

    # CPU dequant reference over the same tables
    y_cpu = ops.moe_grouped_gemm_fp8(
        torch.zeros(Mpad, K, dtype=torch.bfloat16).index_copy_(
            0, dest.cpu(), x.cpu().index_select(0, (order // top_k).cpu())),
        wq.cpu(), ws.cpu(), eids.cpu())

    for bn in (128, 256):
        y = ops.moe_grouped_gemm_fp8(x, wq, ws, eids, bn=bn, x_gather=gtab)
        d = (y.cpu().index_select(0, dest.cpu()).float()
             - y_cpu.index_select(0, dest.cpu()).float())
        rel = d.abs().max() / y_cpu.float().abs().max().clamp(min=1e-6)
        assert rel < 3e-2, (bn, float(rel))

    # fused combine
    gatew = torch.rand(T * top_k, device=dev)[order]
    gw = torch.zeros(Mpad, dtype=torch.float32, device=dev)
    gw[dest] = gatew.float()
    out32 = torch.zeros(T, N, dtype=torch.float32, device=dev)
    got = ops.moe_grouped_gemm_fp8(x, wq, ws, eids, x_gather=gtab,
                                   combine_out=out32, combine_scale=gw,
                                   combine_gather=gtab)
    want = torch.zeros(T, N)
    contrib = y_cpu.index_select(0, dest.cpu()).float() * \
        gatew.cpu().float().unsqueeze(-1)
    want.index_add_(0, (order // top_k).cpu(), contrib)
    rel = (got.cpu() - want).abs().max() / want.abs().max().clamp(min=1e-6)
    assert rel < 3e-2, float(rel)


def test_moe_fp8_module_gpu():
    """ExpertMLPs.quantize_fp8 end-to-end on GPU vs the bf16 dense path."""
    from neuronx_distributed_inference_amd.modules.moe import ExpertMLPs

    torch.manual_seed(8)
    dev = _dev()
    E, H, I, T, top_k = 8, 512, 896, 64, 2
    m = ExpertMLPs(num_experts=E, hidden_size=H, intermediate_size=I,
                   tp_degree=1, dtype=torch.bfloat16)
    with torch.no_grad():
        for p in m.parameters():
            p.normal_(0, 0.05)
    m = m.to(dev)
    x = torch.randn(T, H, dtype=torch.bfloat16, device=dev) * 0.3
    w, idx = torch.softmax(torch.randn(T, E, device=dev), -1).topk(top_k, -1)
    want = m._forward_dense(x, w, idx)
    assert m.quantize_fp8()
    got = m(x, w, idx)
    rel = ((got.float() - want.float()).abs().max()
           / want.float().abs().max().clamp(min=1e-3))
    assert rel < 0.08, float(rel)


@pytest.mark.parametrize("bnbk", [(64, 128), (128, 128)])
def test_moe_grouped_gemm_bk128(bnbk):
    """Double-depth (BK=128) chunk variant vs the per-expert reference."""
    bn, bk = bnbk
    torch.manual_seed(2)
    dev = _dev()
    E, N, K = 4, 768, 640
    idx = torch.randint(0, E, (200,), device=dev)
    order, dest, expert_ids, gtab, Mpad = ops.moe_grouped_tables(idx, E)
    x_rows = torch.randn(200, K, dtype=torch.bfloat16, device=dev) * 0.1
    w = torch.randn(E, N, K, dtype=torch.bfloat16, device=dev) * 0.1
    x_sorted = torch.zeros(Mpad, K, dtype=torch.bfloat16, device=dev)
    x_sorted.index_copy_(0, dest, x_rows.index_select(0, order))
    y = ops.moe_grouped_gemm(x_sorted, w, expert_ids, bn=bn, bk=bk)
    got = y.index_select(0, dest)
    se = idx[order]
    want = torch.empty_like(got)
    for e in range(E):
        m = se == e
        want[m] = x_sorted[dest[m]] @ w[e].t()
    torch.testing.assert_close(got.float(), want.float(), atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("msk", [(64, 1), (128, 2), (256, 4), (100, 1)])
def test_skinny_gemm5_fp8_gpu(msk):
    """fp8-weight skinny decode GEMM vs the dequantized torch reference
    (BK=128 double-depth pipeline, per-channel scale, splitk fold)."""
    M, sk = msk
    torch.manual_seed(9)
    dev = _dev()
    N, K = 512, 1024
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.2
    w = torch.randn(N, K) * 0.1
    scale = w.abs().amax(dim=1).clamp(min=1e-8) / 448.0
    wq = (w / scale.unsqueeze(1)).clamp(-448, 448).to(torch.float8_e4m3fn)
    bias = torch.randn(N, dtype=torch.bfloat16, device=dev) * 0.1
    wq_d = wq.view(torch.uint8).to(dev)
    sc_d = scale.float().to(dev)
    got = ops.skinny_gemm5_fp8(x, wq_d, sc_d, bias=bias, splitk=sk)
    wd = (wq.float() * scale.unsqueeze(1)).to(dev)
    want = x.float() @ wd.t().float() + bias.float()
    torch.testing.assert_close(got.float(), want, atol=5e-2, rtol=5e-2)


def test_fp8_linear_decode_path_gpu():
    """fp8_linear at decode M routes through the skinny fp8 kernel and
    matches the dequantized reference."""
    from neuronx_distributed_inference_amd.parallel.layers import fp8_linear

    torch.manual_seed(10)
    dev = _dev()
    M, N, K = 256, 576, 1152  # K % 128 = 0, N % 64 = 0
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.2
    w = torch.randn(N, K) * 0.1
    scale = w.abs().amax(dim=1, keepdim=True).clamp(min=1e-8) / 448.0
    wq = (w / scale).clamp(-448, 448).to(torch.float8_e4m3fn).to(dev)
    got = fp8_linear(x, wq, scale.float().to(dev), None)
    want = x.float() @ (wq.float().cpu() * scale).t().to(dev).float()
    torch.testing.assert_close(got.float(), want, atol=6e-2, rtol=6e-2)


def test_moe_mx4_module_gpu():
    """ExpertMLPs MXFP4 mode end-to-end on GPU (v3 kernel on K%256
    shapes) vs the bf16 dense path."""
    from neuronx_distributed_inference_amd.modules.moe import ExpertMLPs

    torch.manual_seed(12)
    dev = _dev()
    E, H, I, T, top_k = 8, 512, 1024, 64, 2
    m = ExpertMLPs(num_experts=E, hidden_size=H, intermediate_size=I,
                   tp_degree=1, dtype=torch.bfloat16)
    with torch.no_grad():
        for p in m.parameters():
            p.normal_(0, 0.05)
    m = m.to(dev)
    x = torch.randn(T, H, dtype=torch.bfloat16, device=dev) * 0.3
    w, idx = torch.softmax(torch.randn(T, E, device=dev), -1).topk(top_k, -1)
    want = m._forward_dense(x, w, idx)
    assert m.quantize_mx4()
    got = m(x, w, idx)
    rel = ((got.float() - want.float()).abs().max()
           / want.float().abs().max().clamp(min=1e-3))
    assert rel < 0.35, float(rel)


@pytest.mark.parametrize("mode", ["post", "pre", "sigmoid", "postnorm"])
def test_router_topk_fused_gpu(mode):
    """Fused router vs the torch chain for every activation variant."""
    from neuronx_distributed_inference_amd.modules.moe import MoERouter

    torch.manual_seed(13)
    dev = _dev()
    E, H, T, k = 16, 64, 200, 4
    r = MoERouter(H, E, k,
                  normalize=(mode == "postnorm"),
                  pre_softmax=(mode == "pre"),
                  act=("sigmoid" if mode == "sigmoid" else "softmax"))
    with torch.no_grad():
        r.weight.normal_(0, 0.5)
    x = torch.randn(T, H)
    want_v, want_i = r(x)                 # CPU torch chain
    r = r.to(dev)
    got_v, got_i = r(x.to(dev))
    assert torch.equal(got_i.cpu(), want_i), mode
    torch.testing.assert_close(got_v.cpu(), want_v, atol=1e-5, rtol=1e-5)


def test_moe_grouped_tile32_gpu(monkeypatch):
    """mt=32 grouped kernel (fine-grained tile) vs the dense bmm path on
    GPU at a 128-expert top-8 shape."""
    from neuronx_distributed_inference_amd.modules.moe import ExpertMLPs

    monkeypatch.setenv("NXDI_MOE_TILE32", "1")
    torch.manual_seed(15)
    dev = _dev()
    E, H, I, T, top_k = 32, 256, 384, 48, 8
    m = ExpertMLPs(num_experts=E, hidden_size=H, intermediate_size=I,
                   tp_degree=1, dtype=torch.bfloat16)
    with torch.no_grad():
        for p in m.parameters():
            p.normal_(0, 0.05)
    m = m.to(dev)
    x = torch.randn(T, H, dtype=torch.bfloat16, device=dev) * 0.3
    w, idx = torch.softmax(torch.randn(T, E, device=dev), -1).topk(top_k, -1)
    want = m._forward_dense(x, w, idx)
    got = m._forward_grouped(x, w, idx)
    torch.testing.assert_close(got.float(), want.float(),
                               atol=5e-2, rtol=5e-2)
