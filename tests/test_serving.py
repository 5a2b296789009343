"""Continuous-batching serving engine: mixed prefill/decode interleaving
over KV slots must reproduce per-request reference generations."""

import pytest
import torch

from neuronx_distributed_inference_amd.config import EngineConfig
from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
    LlamaForCausalLM,
    LlamaInferenceConfig,
)
from neuronx_distributed_inference_amd.serving import ServingEngine
from neuronx_distributed_inference_amd.utils.hf_adapter import (
    GenerationConfig,
    HuggingFaceGenerationAdapter,
)


def _model(batch):
    ec = EngineConfig(batch_size=batch, max_batch_size=batch, seq_len=96,
                      max_length=96, torch_dtype="float32", tp_degree=1)
    cfg = LlamaInferenceConfig(
        neuron_config=ec, vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
        tie_word_embeddings=False)
    torch.manual_seed(17)
    m = LlamaForCausalLM(cfg).eval()
    with torch.no_grad():
        for p in m.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    return m


@torch.inference_mode()
def test_serving_streaming_callback():
    m = _model(batch=2)
    seen = []
    eng = ServingEngine(m, on_token=lambda rid, tok: seen.append((rid, tok)))
    rid = eng.add_request(torch.randint(0, 128, (6,)), max_new_tokens=4)
    results = eng.run()
    assert [t for r, t in seen if r == rid] == results[rid]


@torch.inference_mode()
@pytest.mark.parametrize("chunk", [None, 8])
def test_serving_engine_matches_reference(chunk):
    m = _model(batch=3)
    prompts = [
        torch.randint(0, 128, (11,)),
        torch.randint(0, 128, (19,)),
        torch.randint(0, 128, (7,)),
        torch.randint(0, 128, (23,)),  # queued until a slot frees
        torch.randint(0, 128, (13,)),
    ]
    wants = []
    for p in prompts:
        m.reset()
        out = HuggingFaceGenerationAdapter(m).generate(
            p.view(1, -1), generation_config=GenerationConfig(max_new_tokens=10))
        wants.append(out[0, p.shape[0]:].tolist())

    m.reset()
    eng = ServingEngine(m, chunk_size=chunk)
    ids = [eng.add_request(p, max_new_tokens=10) for p in prompts]
    results = eng.run()
    assert set(results) == set(ids)
    for rid, want in zip(ids, wants):
        assert results[rid] == want, (rid, results[rid], want)


@pytest.mark.gpu
@torch.inference_mode()
def test_serving_engine_gpu():
    """Serving engine on GPU (bf16, HIP kernels): chunked-prefill
    interleaving must match per-request generation on the same device."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    # head_dim >= 32 for the HIP decode kernel
    ec = EngineConfig(batch_size=3, max_batch_size=3, seq_len=96,
                      max_length=96, torch_dtype="bfloat16", tp_degree=1)
    cfg = LlamaInferenceConfig(
        neuron_config=ec, vocab_size=128, hidden_size=128, intermediate_size=256,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
        tie_word_embeddings=False)
    torch.manual_seed(17)
    m = LlamaForCausalLM(cfg).eval()
    with torch.no_grad():
        for p in m.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    m = m.to("cuda")
    prompts = [torch.randint(0, 128, (n,)) for n in (11, 19, 7, 23)]
    wants = []
    for p in prompts:
        m.reset()
        out = HuggingFaceGenerationAdapter(m).generate(
            p.view(1, -1).cuda(),
            generation_config=GenerationConfig(max_new_tokens=8))
        wants.append(out[0, p.shape[0]:].cpu().tolist())
    m.reset()
    eng = ServingEngine(m, chunk_size=8)
    ids = [eng.add_request(p, max_new_tokens=8) for p in prompts]
    results = eng.run()
    for rid, want in zip(ids, wants):
        assert results[rid] == want, (rid, results[rid], want)


@torch.inference_mode()
def test_serving_per_request_sampling_params():
    """Requests carry their own (top_k, top_p, temperature); greedy rows
    must still match the reference while a high-temperature row samples."""
    from neuronx_distributed_inference_amd.config import OnDeviceSamplingConfig

    ec = EngineConfig(batch_size=2, max_batch_size=2, seq_len=64, max_length=64,
                      torch_dtype="float32", tp_degree=1,
                      on_device_sampling_config=OnDeviceSamplingConfig(
                          top_k=1, do_sample=True))
    cfg = LlamaInferenceConfig(
        neuron_config=ec, vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
        tie_word_embeddings=False)
    torch.manual_seed(17)
    m = LlamaForCausalLM(cfg).eval()
    with torch.no_grad():
        for p in m.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    prompt = torch.randint(0, 128, (9,))
    m.reset()
    want = HuggingFaceGenerationAdapter(m).generate(
        prompt.view(1, -1), generation_config=GenerationConfig(max_new_tokens=6))
    m.reset()
    eng = ServingEngine(m)
    greedy = eng.add_request(prompt, max_new_tokens=6, top_k=1)
    sampled = eng.add_request(torch.randint(0, 128, (7,)), max_new_tokens=6,
                              top_k=20, temperature=1.5)
    results = eng.run()
    assert results[greedy] == want[0, 9:].tolist()
    assert len(results[sampled]) == 6


@torch.inference_mode()
def test_serving_paged_with_prefix_caching():
    """Paged serving: a later request sharing a prompt prefix reuses the
    cached blocks (tail-only prefill) and still generates exactly the
    per-request reference."""
    ec = EngineConfig(
        batch_size=2, max_batch_size=2, seq_len=64, max_length=64,
        torch_dtype="float32", tp_degree=1,
        is_block_kv_layout=True, pa_block_size=8, pa_num_blocks=40)
    cfg = LlamaInferenceConfig(
        neuron_config=ec, vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
        tie_word_embeddings=False)
    torch.manual_seed(31)
    m = LlamaForCausalLM(cfg).eval()
    with torch.no_grad():
        for p in m.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)

    shared = torch.randint(0, 128, (16,))           # 2 full blocks
    prompts = [
        torch.cat([shared, torch.randint(0, 128, (5,))]),
        torch.randint(0, 128, (11,)),
        torch.cat([shared, torch.randint(0, 128, (3,))]),  # prefix hit
    ]
    # references on a LINEAR-cache twin
    ec2 = EngineConfig(batch_size=2, max_batch_size=2, seq_len=64, max_length=64,
                       torch_dtype="float32", tp_degree=1)
    cfg2 = LlamaInferenceConfig(
        neuron_config=ec2, vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
        tie_word_embeddings=False)
    lin = LlamaForCausalLM(cfg2).eval()
    lin.model.load_state_dict(
        {k: v for k, v in m.model.state_dict().items()
         if not k.startswith("kv_mgr.")}, strict=False)
    wants = []
    for p in prompts:
        lin.reset()
        out = HuggingFaceGenerationAdapter(lin).generate(
            p.view(1, -1), generation_config=GenerationConfig(max_new_tokens=6))
        wants.append(out[0, p.shape[0]:].tolist())

    eng = ServingEngine(m, chunk_size=8)
    assert eng.paged
    ids = [eng.add_request(p, max_new_tokens=6) for p in prompts]
    results = eng.run()
    for rid, want in zip(ids, wants):
        assert results[rid] == want, (rid, results[rid], want)
    # the third request must have hit the 2-block shared prefix
    assert eng.finished[ids[2]].n_cached == 16


@pytest.mark.gpu
@torch.inference_mode()
def test_serving_paged_prefix_caching_gpu():
    """Paged serving + prefix caching on GPU (HIP paged decode kernel)."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    ec = EngineConfig(
        batch_size=2, max_batch_size=2, seq_len=64, max_length=64,
        torch_dtype="bfloat16", tp_degree=1,
        is_block_kv_layout=True, pa_block_size=8, pa_num_blocks=40)
    # head_dim 64 for the paged HIP decode kernel
    cfg = LlamaInferenceConfig(
        neuron_config=ec, vocab_size=128, hidden_size=256, intermediate_size=512,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
        tie_word_embeddings=False)
    torch.manual_seed(31)
    m = LlamaForCausalLM(cfg).eval()
    with torch.no_grad():
        for p in m.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    m = m.to("cuda")
    shared = torch.randint(0, 128, (16,))
    prompts = [
        torch.cat([shared, torch.randint(0, 128, (5,))]),
        torch.cat([shared, torch.randint(0, 128, (3,))]),
    ]
    wants = []
    for p in prompts:
        # per-request reference on fresh blocks, same engine machinery
        e = ServingEngine(m)
        rid = e.add_request(p, max_new_tokens=6)
        wants.append(e.run()[rid])
    # one engine, sequential requests: the second hits the first's prefix
    eng = ServingEngine(m)
    r0 = eng.add_request(prompts[0], max_new_tokens=6)
    eng.run()
    r1 = eng.add_request(prompts[1], max_new_tokens=6)
    eng.run()
    assert eng.finished[r0].output == wants[0]
    assert eng.finished[r1].output == wants[1]
    assert eng.finished[r1].n_cached == 16


@torch.inference_mode()
def test_serving_per_request_lora_adapters():
    """Per-request LoRA: a request on adapter 1 generates differently from
    adapter 0, each matching its single-adapter reference."""
    from neuronx_distributed_inference_amd.modules.lora_serving import (
        LoraServingConfig,
        lora_layers,
        wrap_model_with_lora,
    )

    m = _model(batch=2)
    cfg = LoraServingConfig(max_loras=2, max_lora_rank=4, lora_alpha=8.0)
    state = wrap_model_with_lora(m.model, cfg, max_batch=2)
    torch.manual_seed(7)
    for name, layer in lora_layers(m.model).items():
        a = torch.randn(4, layer.lora_A.shape[2]) * 0.1
        b = torch.randn(layer.lora_B.shape[1], 4) * 0.1
        layer.load_adapter(1, a, b)

    prompt = torch.randint(0, 128, (9,))
    wants = {}
    for ad in (0, 1):
        m.reset()
        eng = ServingEngine(m, lora_state=state)
        rid = eng.add_request(prompt, max_new_tokens=6, adapter_id=ad)
        wants[ad] = eng.run()[rid]
    assert wants[0] != wants[1]

    # both adapters served in one engine, sequentially (per-request rows)
    m.reset()
    eng = ServingEngine(m, lora_state=state)
    r0 = eng.add_request(prompt, max_new_tokens=6, adapter_id=0)
    r1 = eng.add_request(prompt.clone(), max_new_tokens=6, adapter_id=1)
    # different slots -> same decode batch has different adapter rows
    results = eng.run()
    assert results[r0] == wants[0]
    assert results[r1] == wants[1]


@torch.inference_mode()
def test_serving_metrics():
    """metrics(): TTFT/TPOT percentiles + prefix-hit stats over finished
    requests (ref utils/benchmark.py latency report)."""
    m = _model(batch=2)
    eng = ServingEngine(m)
    eng.add_request(torch.randint(0, 128, (4,)), max_new_tokens=4)
    eng.add_request(torch.randint(0, 128, (3,)), max_new_tokens=3)
    eng.run()
    m = eng.metrics()
    assert m["requests"] == 2.0
    assert m["output_tokens"] == 7.0
    assert m["ttft_p50_s"] >= 0 and m["ttft_p99_s"] >= m["ttft_p50_s"] - 1e-9
    assert m["tpot_p50_s"] >= 0
    assert 0.0 <= m["prefix_hit_rate"] <= 1.0


@torch.inference_mode()
def test_serving_bucketed_chunks_match_ragged():
    """Intermediate-chunk padding (bucket_chunks) must not change any
    generated token vs exact ragged chunking."""
    m = _model(batch=2)
    prompts = [torch.randint(0, 128, (n,)) for n in (13, 21)]
    m.reset()
    eng = ServingEngine(m, chunk_size=8, bucket_chunks=False)
    ids = [eng.add_request(p, max_new_tokens=6) for p in prompts]
    ragged = eng.run()
    m.reset()
    eng2 = ServingEngine(m, chunk_size=8, bucket_chunks=True)
    ids2 = [eng2.add_request(p, max_new_tokens=6) for p in prompts]
    bucketed = eng2.run()
    for a, b in zip(ids, ids2):
        assert ragged[a] == bucketed[b], (ragged[a], bucketed[b])


@torch.inference_mode()
def test_serving_rejects_sampling_on_static_greedy():
    """A request asking for sampling while the engine's sampler is
    static-greedy must be rejected loudly, not silently decoded greedy
    (ADVICE r01)."""
    m = _model(batch=2)  # no on_device_sampling_config -> static greedy
    eng = ServingEngine(m)
    with pytest.raises(ValueError, match="static-greedy"):
        eng.add_request(torch.randint(0, 128, (6,)), max_new_tokens=2, top_k=5)
    with pytest.raises(ValueError, match="static-greedy"):
        eng.add_request(torch.randint(0, 128, (6,)), max_new_tokens=2,
                        temperature=0.7)
    # greedy requests still fine
    rid = eng.add_request(torch.randint(0, 128, (6,)), max_new_tokens=2)
    assert len(eng.run()[rid]) == 2


@torch.inference_mode()
def test_serving_mixed_chunking_matches_reference():
    """Mixed chunked prefill (prefill chunk + decode rows in ONE
    multi-token TKG batch — ref chunked-prefill tile buckets,
    autobucketing.py:101-146) must reproduce per-request reference
    generations, and decode steps must not stall behind chunks."""
    m = _model(batch=3)
    prompts = [torch.randint(0, 128, (24,), generator=torch.Generator().manual_seed(i))
               for i in range(3)]
    want = {}
    for i, p in enumerate(prompts):
        ref = HuggingFaceGenerationAdapter(m)
        out = ref.generate(p.view(1, -1),
                           generation_config=GenerationConfig(max_new_tokens=6))
        want[i] = out[0, 24:].tolist()
        m.reset()
    eng = ServingEngine(m, chunk_size=8, mixed_chunking=True)
    rids = {}
    # stagger arrivals so chunks overlap decodes
    rids[0] = eng.add_request(prompts[0], max_new_tokens=6)
    eng.step(); eng.step(); eng.step()  # prefill req 0 fully
    rids[1] = eng.add_request(prompts[1], max_new_tokens=6)
    rids[2] = eng.add_request(prompts[2], max_new_tokens=6)
    res = eng.run()
    for i in range(3):
        assert res[rids[i]] == want[i], (i, res[rids[i]], want[i])
