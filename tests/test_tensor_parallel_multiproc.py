"""Multi-process (gloo, world=2) tensor-parallel correctness tests.

The distributed path must be correct by construction before the driver's
8-GPU RCCL run: these cover the parallel layers, distributed argmax/topk,
checkpoint sharding (incl. GQA KV-head replication) and full tiny-Llama
TP=2 generation vs the TP=1 result.
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

pytestmark = pytest.mark.multiproc

WORLD = 2


def _run(rank, fn_name, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)
    torch.manual_seed(0)
    try:
        result = globals()[fn_name](rank)
        q.put((rank, "ok", result))
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put((rank, "err", traceback.format_exc()))
    finally:
        dist.destroy_process_group()


def spawn(fn_name):
    import random

    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = random.randint(29600, 29999)
    ps = [ctx.Process(target=_run, args=(r, fn_name, port, q)) for r in range(WORLD)]
    for p in ps:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, status, payload = q.get()
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        results[rank] = payload
    for p in ps:
        p.join(timeout=60)
    return results


# ---------------------------------------------------------------------------
def _worker_parallel_linear(rank):
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.parallel.layers import (
        ColumnParallelLinear,
        RowParallelLinear,
    )
    from neuronx_distributed_inference_amd.utils.checkpoint_sharding import (
        shard_state_dict_for_rank,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=WORLD)
    torch.manual_seed(42)
    x = torch.randn(3, 8)
    w_col = torch.randn(16, 8)
    w_row = torch.randn(8, 16)

    col = ColumnParallelLinear(8, 16, dtype=torch.float32, gather_output=True)
    with torch.no_grad():
        col.weight.copy_(w_col[rank * 8:(rank + 1) * 8])
    y = col(x)
    want = x @ w_col.T
    torch.testing.assert_close(y, want, atol=1e-5, rtol=1e-5)

    row = RowParallelLinear(16, 8, dtype=torch.float32, input_is_parallel=True)
    with torch.no_grad():
        row.weight.copy_(w_row[:, rank * 8:(rank + 1) * 8])
    xp = (x @ w_col.T)[:, rank * 8:(rank + 1) * 8]
    z = row(xp)
    torch.testing.assert_close(z, (x @ w_col.T) @ w_row.T, atol=1e-4, rtol=1e-4)
    return True


def test_parallel_linear():
    spawn("_worker_parallel_linear")


# ---------------------------------------------------------------------------
def _worker_distributed_ops(rank):
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.parallel.ops import (
        distributed_argmax,
        distributed_topk,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=WORLD)
    torch.manual_seed(7)
    full = torch.randn(4, 20)
    shard = full[:, rank * 10:(rank + 1) * 10]
    am = distributed_argmax(shard, dim=-1)
    assert torch.equal(am, full.argmax(-1)), (am, full.argmax(-1))
    tv, ti = distributed_topk(shard, 5, dim=-1)
    wv, wi = full.topk(5, dim=-1)
    torch.testing.assert_close(tv, wv)
    assert torch.equal(ti, wi)
    return True


def test_distributed_ops():
    spawn("_worker_distributed_ops")


# ---------------------------------------------------------------------------
def _worker_embedding(rank):
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.parallel.layers import ParallelEmbedding

    ps.initialize_model_parallel(tensor_model_parallel_size=WORLD)
    torch.manual_seed(3)
    w = torch.randn(32, 8)
    emb = ParallelEmbedding(32, 8, dtype=torch.float32)
    with torch.no_grad():
        emb.weight.copy_(w[rank * 16:(rank + 1) * 16])
    ids = torch.randint(0, 32, (2, 5))
    torch.testing.assert_close(emb(ids), w[ids], atol=1e-6, rtol=1e-6)
    return True


def test_parallel_embedding():
    spawn("_worker_embedding")


# ---------------------------------------------------------------------------
def _worker_tp2_generation(rank):
    """Full tiny-Llama TP=2 (gqa replication: 2 kv heads over tp=2) matches
    the single-rank result token for token."""
    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
        LlamaForCausalLM,
        LlamaInferenceConfig,
    )
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.utils.checkpoint_sharding import (
        shard_state_dict_for_rank,
    )
    from neuronx_distributed_inference_amd.utils.hf_adapter import (
        GenerationConfig,
        HuggingFaceGenerationAdapter,
    )

    def make_cfg(tp):
        ec = EngineConfig(batch_size=2, max_batch_size=2, seq_len=64, max_length=64,
                          torch_dtype="float32", tp_degree=tp)
        return LlamaInferenceConfig(
            neuron_config=ec, vocab_size=128, hidden_size=64, intermediate_size=128,
            num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
            max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
            hidden_act="silu", tie_word_embeddings=False)

    # build the full model on every rank with identical weights (seeded)
    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(11)
    full = LlamaForCausalLM(make_cfg(1)).eval()
    with torch.no_grad():
        for p in full.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    ids = torch.randint(0, 128, (2, 8))
    want = HuggingFaceGenerationAdapter(full).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=8))

    # now the TP=2 sharded model
    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=WORLD)
    tp_model = LlamaForCausalLM(make_cfg(WORLD)).eval()
    full_sd = {k: v for k, v in full.model.state_dict().items()
               if not k.startswith("kv_mgr.")}
    shard = shard_state_dict_for_rank(full_sd, tp_model.model, rank, WORLD)
    tp_model.model.load_state_dict(shard, strict=False)
    got = HuggingFaceGenerationAdapter(tp_model).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=8))
    assert torch.equal(got, want), (got.tolist(), want.tolist())
    return True


def test_tp2_generation_matches_tp1():
    spawn("_worker_tp2_generation")


# ---------------------------------------------------------------------------
def _worker_sp_generation(rank):
    """Sequence-parallel prefill (TP=2) must match the non-SP TP=2 result."""
    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
        LlamaForCausalLM,
        LlamaInferenceConfig,
    )
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.utils.checkpoint_sharding import (
        shard_state_dict_for_rank,
    )
    from neuronx_distributed_inference_amd.utils.hf_adapter import (
        GenerationConfig,
        HuggingFaceGenerationAdapter,
    )

    def make_cfg(tp, sp):
        ec = EngineConfig(batch_size=2, max_batch_size=2, seq_len=64, max_length=64,
                          torch_dtype="float32", tp_degree=tp,
                          sequence_parallel_enabled=sp)
        return LlamaInferenceConfig(
            neuron_config=ec, vocab_size=128, hidden_size=64, intermediate_size=128,
            num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
            max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
            hidden_act="silu", tie_word_embeddings=False)

    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(21)
    full = LlamaForCausalLM(make_cfg(1, False)).eval()
    with torch.no_grad():
        for p in full.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    ids = torch.randint(0, 128, (2, 8))  # 8 divisible by tp=2 -> SP active
    want = HuggingFaceGenerationAdapter(full).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=6))

    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=WORLD)
    sp_model = LlamaForCausalLM(make_cfg(WORLD, True)).eval()
    full_sd = {k: v for k, v in full.model.state_dict().items()
               if not k.startswith("kv_mgr.")}
    shard = shard_state_dict_for_rank(full_sd, sp_model.model, rank, WORLD)
    sp_model.model.load_state_dict(shard, strict=False)
    got = HuggingFaceGenerationAdapter(sp_model).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=6))
    assert torch.equal(got, want), (got.tolist(), want.tolist())
    return True


def test_sequence_parallel_generation():
    spawn("_worker_sp_generation")


# ---------------------------------------------------------------------------
def _worker_ep_mixtral(rank):
    """Mixtral with EP=2 (experts split across ranks, moe_tp=1) matches the
    single-rank result."""
    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.mixtral.modeling_mixtral import (
        MixtralForCausalLM,
        MixtralInferenceConfig,
    )
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.utils.checkpoint_sharding import (
        shard_state_dict_for_rank,
    )
    from neuronx_distributed_inference_amd.utils.hf_adapter import (
        GenerationConfig,
        HuggingFaceGenerationAdapter,
    )

    def make_cfg(tp, ep):
        ec = EngineConfig(batch_size=2, max_batch_size=2, seq_len=48, max_length=48,
                          torch_dtype="float32", tp_degree=tp, ep_degree=ep)
        return MixtralInferenceConfig(
            neuron_config=ec, vocab_size=128, hidden_size=64, intermediate_size=64,
            num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
            num_local_experts=4, num_experts_per_tok=2,
            max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
            hidden_act="silu", tie_word_embeddings=False)

    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(31)
    full = MixtralForCausalLM(make_cfg(1, 1)).eval()
    with torch.no_grad():
        for p in full.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    ids = torch.randint(0, 128, (2, 8))
    want = HuggingFaceGenerationAdapter(full).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=6))

    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=WORLD,
                                 expert_model_parallel_size=WORLD)
    ep_model = MixtralForCausalLM(make_cfg(WORLD, WORLD)).eval()
    full_sd = {k: v for k, v in full.model.state_dict().items()
               if not k.startswith("kv_mgr.")}
    shard = shard_state_dict_for_rank(full_sd, ep_model.model, rank, WORLD)
    ep_model.model.load_state_dict(shard, strict=False)
    got = HuggingFaceGenerationAdapter(ep_model).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=6))
    assert torch.equal(got, want), (got.tolist(), want.tolist())
    return True


def test_expert_parallel_mixtral():
    spawn("_worker_ep_mixtral")


# ---------------------------------------------------------------------------
def _worker_dp_sampler(rank):
    """DataParallelSampler (batch<->vocab all-to-all) matches the plain
    distributed sampler's greedy result."""
    from neuronx_distributed_inference_amd.config import OnDeviceSamplingConfig
    from neuronx_distributed_inference_amd.modules.generation.sampling import (
        DataParallelSampler,
        Sampler,
    )
    from neuronx_distributed_inference_amd.parallel import state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=WORLD)
    torch.manual_seed(17)
    V, B = 40, 4
    full = torch.randn(B, V)
    shard = full[:, rank * (V // WORLD):(rank + 1) * (V // WORLD)]
    cfg = OnDeviceSamplingConfig(top_k=1)
    want = Sampler(cfg, V)(shard)
    got = DataParallelSampler(cfg, V, dp_degree=WORLD)(shard)
    assert torch.equal(got, full.argmax(-1)), (got, full.argmax(-1))
    assert torch.equal(want, got)
    return True


def test_data_parallel_sampler():
    spawn("_worker_dp_sampler")


# ---------------------------------------------------------------------------
def _worker_cp_generation(rank):
    """Context-parallel prefill (tp=2, cp=2 -> prefill_tp=1) matches the
    single-rank result."""
    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
        LlamaForCausalLM,
        LlamaInferenceConfig,
    )
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.utils.checkpoint_sharding import (
        shard_state_dict_for_rank,
    )
    from neuronx_distributed_inference_amd.utils.hf_adapter import (
        GenerationConfig,
        HuggingFaceGenerationAdapter,
    )

    def make_cfg(tp, cp):
        ec = EngineConfig(batch_size=2, max_batch_size=2, seq_len=64, max_length=64,
                          torch_dtype="float32", tp_degree=tp, cp_degree=cp)
        return LlamaInferenceConfig(
            neuron_config=ec, vocab_size=128, hidden_size=64, intermediate_size=128,
            num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
            max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
            hidden_act="silu", tie_word_embeddings=False)

    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(41)
    full = LlamaForCausalLM(make_cfg(1, 1)).eval()
    with torch.no_grad():
        for p in full.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    ids = torch.randint(0, 128, (2, 8))  # S divisible by cp=2
    want = HuggingFaceGenerationAdapter(full).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=6))

    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=WORLD,
                                 context_parallel_size=WORLD)
    cp_model = LlamaForCausalLM(make_cfg(WORLD, WORLD)).eval()
    full_sd = {k: v for k, v in full.model.state_dict().items()
               if not k.startswith("kv_mgr.")}
    # the CP model has extra cte_* params; feed duplicated full weights
    for k in list(full_sd):
        if ".self_attn.qkv_proj." in k:
            full_sd[k.replace(".qkv_proj.", ".cte_qkv_proj.")] = full_sd[k]
        elif ".self_attn.o_proj.o_proj." in k:
            full_sd[k.replace(".o_proj.o_proj.", ".cte_o_proj.o_proj.")] = full_sd[k]
    shard = shard_state_dict_for_rank(full_sd, cp_model.model, rank, WORLD)
    missing, unexpected = cp_model.model.load_state_dict(shard, strict=False)
    assert not [m for m in missing if not m.startswith("kv_mgr")], missing
    got = HuggingFaceGenerationAdapter(cp_model).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=6))
    assert torch.equal(got, want), (got.tolist(), want.tolist())
    return True


def test_context_parallel_generation():
    spawn("_worker_cp_generation")


# ---------------------------------------------------------------------------
def _worker_attention_dp(rank):
    """Attention-DP decode (tp=2, dp=2 -> decode_tp=1; batch sharded, KV
    cache B/dp per rank) matches the single-rank result."""
    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
        LlamaForCausalLM,
        LlamaInferenceConfig,
    )
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.utils.checkpoint_sharding import (
        shard_state_dict_for_rank,
    )
    from neuronx_distributed_inference_amd.utils.hf_adapter import (
        GenerationConfig,
        HuggingFaceGenerationAdapter,
    )

    def make_cfg(tp, dp):
        ec = EngineConfig(batch_size=2, max_batch_size=2, seq_len=64, max_length=64,
                          torch_dtype="float32", tp_degree=tp,
                          attention_dp_degree=dp)
        return LlamaInferenceConfig(
            neuron_config=ec, vocab_size=128, hidden_size=64, intermediate_size=128,
            num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
            max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
            hidden_act="silu", tie_word_embeddings=False)

    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(51)
    full = LlamaForCausalLM(make_cfg(1, 1)).eval()
    with torch.no_grad():
        for p in full.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    ids = torch.randint(0, 128, (2, 8))
    want = HuggingFaceGenerationAdapter(full).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=6))

    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=WORLD,
                                 attention_dp_size=WORLD)
    dp_model = LlamaForCausalLM(make_cfg(WORLD, WORLD)).eval()
    assert dp_model.model.kv_mgr.k_caches[0].shape[0] == 1  # B/dp rows
    full_sd = {k: v for k, v in full.model.state_dict().items()
               if not k.startswith("kv_mgr.")}
    for k in list(full_sd):
        if ".self_attn.qkv_proj." in k:
            full_sd[k.replace(".qkv_proj.", ".tkg_qkv_proj.")] = full_sd[k]
        elif ".self_attn.o_proj.o_proj." in k:
            full_sd[k.replace(".o_proj.o_proj.", ".tkg_o_proj.o_proj.")] = full_sd[k]
    shard = shard_state_dict_for_rank(full_sd, dp_model.model, rank, WORLD)
    missing, unexpected = dp_model.model.load_state_dict(shard, strict=False)
    assert not [m for m in missing if not m.startswith("kv_mgr")], missing
    got = HuggingFaceGenerationAdapter(dp_model).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=6))
    assert torch.equal(got, want), (got.tolist(), want.tolist())
    return True


def test_attention_dp_generation():
    spawn("_worker_attention_dp")


# ---------------------------------------------------------------------------
def _worker_flash_decoding(rank):
    """Flash decoding op-level: strided S-sharded cache + LSE combine must
    match single-rank full-cache decode."""
    import torch

    from neuronx_distributed_inference_amd import ops
    from neuronx_distributed_inference_amd.modules.attention import flash_decoding as fd
    from neuronx_distributed_inference_amd.parallel import state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=2, kv_shard_size=2)
    torch.manual_seed(7)
    B, Hq, Hkv, Smax, D, Sq = 2, 4, 1, 64, 32, 1
    fdg = 2
    # identical full K/V on both ranks (replicated KV head)
    k_full = torch.randn(B, Hkv, Smax, D)
    v_full = torch.randn(B, Hkv, Smax, D)
    # local strided shard caches
    kc = torch.zeros(B, Hkv, Smax // fdg, D)
    vc = torch.zeros(B, Hkv, Smax // fdg, D)
    pos_all = torch.arange(Smax).view(1, Smax).expand(B, Smax).contiguous()
    wpos = fd.shard_write_positions(pos_all, fdg, rank)
    seq_ids = torch.arange(B, dtype=torch.int32)
    ops.kv_cache_update(kc, vc, k_full, v_full, wpos, seq_ids)

    # ragged current positions per row
    positions = torch.tensor([[37], [52]])
    # local q heads per rank: rank owns heads [rank*2, rank*2+2)
    torch.manual_seed(11)
    q_full = torch.randn(B, Hq, Sq, D)
    q_local = q_full[:, rank * 2:(rank + 1) * 2].contiguous()
    out = fd.flash_decoding_attention(
        q_local, kc, vc, positions, seq_ids, scale=D ** -0.5)

    want_full = ops.attention_decode(
        q_full.float(), k_full.float(), v_full.float(), positions,
        seq_ids=seq_ids, scale=D ** -0.5)
    want = want_full[:, rank * 2:(rank + 1) * 2]
    torch.testing.assert_close(out.float(), want.float(), atol=2e-3, rtol=2e-3)
    ps.destroy_model_parallel()
    return True


def test_flash_decoding_sharded_kv():
    spawn("_worker_flash_decoding")


# ---------------------------------------------------------------------------
def _worker_flash_decoding_e2e(rank):
    """Full tiny-Llama generation with flash decoding (TP=2, KV head
    replicated, decode cache S-sharded) matches the single-rank result."""
    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
        LlamaForCausalLM,
        LlamaInferenceConfig,
    )
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.utils.checkpoint_sharding import (
        shard_state_dict_for_rank,
    )
    from neuronx_distributed_inference_amd.utils.hf_adapter import (
        GenerationConfig,
        HuggingFaceGenerationAdapter,
    )

    def make_cfg(tp, flash):
        ec = EngineConfig(batch_size=2, max_batch_size=2, seq_len=64, max_length=64,
                          torch_dtype="float32", tp_degree=tp,
                          flash_decoding_enabled=flash)
        return LlamaInferenceConfig(
            neuron_config=ec, vocab_size=128, hidden_size=64, intermediate_size=128,
            num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=1,
            max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
            hidden_act="silu", tie_word_embeddings=False)

    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(23)
    full = LlamaForCausalLM(make_cfg(1, False)).eval()
    with torch.no_grad():
        for p in full.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    ids = torch.randint(0, 128, (2, 9))
    want = HuggingFaceGenerationAdapter(full).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=8))

    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=WORLD,
                                 kv_shard_size=WORLD)
    tp_model = LlamaForCausalLM(make_cfg(WORLD, True)).eval()
    # S-sharded cache is half length
    assert tp_model.model.kv_mgr.max_len == 32
    full_sd = {k: v for k, v in full.model.state_dict().items()
               if not k.startswith("kv_mgr.")}
    shard = shard_state_dict_for_rank(full_sd, tp_model.model, rank, WORLD)
    tp_model.model.load_state_dict(shard, strict=False)
    got = HuggingFaceGenerationAdapter(tp_model).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=8))
    assert torch.equal(got, want), (got.tolist(), want.tolist())
    return True


def test_flash_decoding_e2e_generation():
    spawn("_worker_flash_decoding_e2e")


# ---------------------------------------------------------------------------
def _worker_cp_flash_decoding_e2e(rank):
    """COMBINED seam test: context-parallel prefill
    (tp=2, cp=2) writing into an S-sharded flash-decoding cache
    (kv_shard=2) — the CP query reorder and the cross-rank LSE-combined
    decode must compose to the single-rank result."""
    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
        LlamaForCausalLM,
        LlamaInferenceConfig,
    )
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.utils.checkpoint_sharding import (
        shard_state_dict_for_rank,
    )
    from neuronx_distributed_inference_amd.utils.hf_adapter import (
        GenerationConfig,
        HuggingFaceGenerationAdapter,
    )

    def make_cfg(tp, cp, flash):
        ec = EngineConfig(batch_size=2, max_batch_size=2, seq_len=64,
                          max_length=64, torch_dtype="float32", tp_degree=tp,
                          cp_degree=cp, flash_decoding_enabled=flash)
        return LlamaInferenceConfig(
            neuron_config=ec, vocab_size=128, hidden_size=64,
            intermediate_size=128, num_hidden_layers=2,
            num_attention_heads=4, num_key_value_heads=1,
            max_position_embeddings=128, rms_norm_eps=1e-5,
            rope_theta=10000.0, hidden_act="silu", tie_word_embeddings=False)

    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(29)
    full = LlamaForCausalLM(make_cfg(1, 1, False)).eval()
    with torch.no_grad():
        for p in full.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    ids = torch.randint(0, 128, (2, 8))  # S divisible by cp
    want = HuggingFaceGenerationAdapter(full).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=8))

    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=WORLD,
                                 context_parallel_size=WORLD,
                                 kv_shard_size=WORLD)
    model = LlamaForCausalLM(make_cfg(WORLD, WORLD, True)).eval()
    assert model.model.kv_mgr.max_len == 32  # S-sharded cache
    full_sd = {k: v for k, v in full.model.state_dict().items()
               if not k.startswith("kv_mgr.")}
    for k in list(full_sd):
        if ".self_attn.qkv_proj." in k:
            full_sd[k.replace(".qkv_proj.", ".cte_qkv_proj.")] = full_sd[k]
        elif ".self_attn.o_proj.o_proj." in k:
            full_sd[k.replace(".o_proj.o_proj.", ".cte_o_proj.o_proj.")] = full_sd[k]
    shard = shard_state_dict_for_rank(full_sd, model.model, rank, WORLD)
    model.model.load_state_dict(shard, strict=False)
    got = HuggingFaceGenerationAdapter(model).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=8))
    assert torch.equal(got, want), (got.tolist(), want.tolist())
    return True


def test_cp_plus_flash_decoding_e2e():
    spawn("_worker_cp_flash_decoding_e2e")


def _worker_draft_tp_speculation(rank):
    """Two-model speculation with the TARGET at TP=2 and the DRAFT at
    TP=1 via the draft sub-communicator (ref speculative-draft group):
    output must equal the target-only TP=2 greedy stream."""
    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
        LlamaForCausalLM,
        LlamaInferenceConfig,
    )
    from neuronx_distributed_inference_amd.modules.speculation import (
        DraftTPWrapper,
        SpeculativeGenerationAdapter,
    )
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.utils.checkpoint_sharding import (
        shard_state_dict_for_rank,
    )
    from neuronx_distributed_inference_amd.utils.hf_adapter import (
        GenerationConfig,
        HuggingFaceGenerationAdapter,
    )

    def make_cfg(tp, spec_len=0, layers=2, hidden=64):
        from neuronx_distributed_inference_amd.config import (
            OnDeviceSamplingConfig,
        )

        ec = EngineConfig(batch_size=2, max_batch_size=2, seq_len=96,
                          max_length=96, torch_dtype="float32", tp_degree=tp,
                          speculation_length=spec_len,
                          on_device_sampling_config=OnDeviceSamplingConfig(
                              top_k=1))
        return LlamaInferenceConfig(
            neuron_config=ec, vocab_size=128, hidden_size=hidden,
            intermediate_size=hidden * 2, num_hidden_layers=layers,
            num_attention_heads=4, num_key_value_heads=2,
            max_position_embeddings=128, rms_norm_eps=1e-5,
            rope_theta=10000.0, hidden_act="silu", tie_word_embeddings=False)

    # reference: target-only greedy at TP=1
    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(21)
    full = LlamaForCausalLM(make_cfg(1)).eval()
    with torch.no_grad():
        for p in full.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.1)
    ids = torch.randint(0, 128, (2, 8))
    want = HuggingFaceGenerationAdapter(full).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=12))

    # target TP=2 + draft TP=1 (replicated across the two subgroups)
    ps.destroy_model_parallel()
    ps.initialize_model_parallel(tensor_model_parallel_size=WORLD,
                                 draft_tp_size=1)
    target = LlamaForCausalLM(make_cfg(WORLD, spec_len=4)).eval()
    full_sd = {k: v for k, v in full.model.state_dict().items()
               if not k.startswith("kv_mgr.")}
    shard = shard_state_dict_for_rank(full_sd, target.model, rank, WORLD)
    target.model.load_state_dict(shard, strict=False)

    torch.manual_seed(33)  # identical draft weights on every rank
    with ps.draft_tp_context():
        draft_lm = LlamaForCausalLM(make_cfg(1, layers=1, hidden=32)).eval()
        with torch.no_grad():
            for p in draft_lm.parameters():
                if p.dim() >= 2:
                    p.normal_(0, 0.1)
    draft = DraftTPWrapper(draft_lm)

    got = SpeculativeGenerationAdapter(target, draft, 4).generate(
        ids, generation_config=GenerationConfig(max_new_tokens=12))
    for i in range(2):
        row = got[i]
        n = row.shape[0] - int((row != 0).flip(0).float().argmax())
        n = min(n, want.shape[1])
        assert n >= ids.shape[1] + 8
        assert torch.equal(row[:n], want[i, :n]), (row.tolist(),
                                                   want[i].tolist())
    # sanity: the draft group really is size 1 inside the context
    with ps.draft_tp_context():
        assert ps.get_tensor_model_parallel_size() == 1
    assert ps.get_tensor_model_parallel_size() == WORLD
    return True


def test_draft_tp_speculation():
    spawn("_worker_draft_tp_speculation")
