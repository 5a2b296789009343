"""End-to-end `inference_demo` CLI test on CPU — BASELINE config 1:
tiny Llama (2-layer) greedy decode through the full compile -> load ->
accuracy-check -> generate -> benchmark pipeline."""

import json
import os

import pytest
import torch


@pytest.fixture
def tiny_hf_checkpoint(tmp_path):
    from transformers import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(0)
    cfg = LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=256, rms_norm_eps=1e-5, tie_word_embeddings=False,
    )
    m = LlamaForCausalLM(cfg)
    path = tmp_path / "ckpt"
    m.save_pretrained(str(path), safe_serialization=True)
    return str(path)


def test_inference_demo_cpu_pipeline(tiny_hf_checkpoint, tmp_path, capsys):
    from neuronx_distributed_inference_amd.inference_demo import main

    compiled = str(tmp_path / "compiled")
    prompt_ids = json.dumps([[5, 9, 13, 21, 7], [3, 3, 8]])
    rc = main([
        "--model-type", "llama", "run",
        "--model-path", tiny_hf_checkpoint,
        "--compiled-model-path", compiled,
        "--prompt-ids", prompt_ids,
        "--max-new-tokens", "8",
        "--on-cpu", "--torch-dtype", "float32",
        "--batch-size", "2", "--max-context-length", "32", "--seq-len", "64",
        "--check-accuracy-mode", "token-matching",
        "--benchmark", "--num-benchmark-runs", "2",
    ])
    assert rc == 0
    out = capsys.readouterr().out
    assert "accuracy check passed" in out
    assert os.path.exists(os.path.join(compiled, "engine_config.json"))
    assert os.path.exists(os.path.join(compiled, "weights",
                                       "tp0_sharded_checkpoint.safetensors"))
    assert os.path.exists(os.path.join(compiled, "benchmark_report.json"))
    report = json.load(open(os.path.join(compiled, "benchmark_report.json")))
    assert report["e2e_model"]["throughput"] > 0
    assert "latency_ms_p50" in report["token_generation_model"]


def test_inference_demo_on_device_sampling_flags(tiny_hf_checkpoint, tmp_path):
    from neuronx_distributed_inference_amd.inference_demo import main

    compiled = str(tmp_path / "compiled2")
    rc = main([
        "--model-type", "llama", "run",
        "--model-path", tiny_hf_checkpoint,
        "--compiled-model-path", compiled,
        "--prompt-ids", json.dumps([[1, 2, 3, 4]]),
        "--max-new-tokens", "4",
        "--on-cpu", "--on-device-sampling", "--top-k", "1",
        "--batch-size", "1", "--max-context-length", "16", "--seq-len", "32",
    ])
    assert rc == 0


@pytest.mark.parametrize("extra", [
    [],                                             # EAGLE-1 chain
    ["--enable-token-tree", "--token-tree-branching", "2,2"],
    ["--enable-eagle3"],
])
def test_inference_demo_eagle_flags(tiny_hf_checkpoint, tmp_path, extra):
    """CLI wiring of the EAGLE family (chain / tree / EAGLE-3): draft =
    same tiny checkpoint; fc is random-init (no fc key in the ckpt), so
    only the pipeline contract is checked, not acceptance quality."""
    from neuronx_distributed_inference_amd.inference_demo import main

    compiled = str(tmp_path / ("compiled_eagle" + str(len(extra))))
    rc = main([
        "--model-type", "llama", "run",
        "--model-path", tiny_hf_checkpoint,
        "--compiled-model-path", compiled,
        "--prompt-ids", json.dumps([[5, 9, 13, 21, 7]]),
        "--max-new-tokens", "6",
        "--on-cpu", "--torch-dtype", "float32",
        "--batch-size", "1", "--max-context-length", "32", "--seq-len", "64",
        "--draft-model-path", tiny_hf_checkpoint,
        "--speculation-length", "2",
        "--enable-eagle-speculation",
    ] + extra)
    assert rc == 0


def test_eagle_forces_on_device_greedy_sampling():
    """EAGLE verification consumes on-device tokens: the config builder
    must arm greedy on-device sampling when --enable-eagle-speculation is
    set without --on-device-sampling."""
    from neuronx_distributed_inference_amd.inference_demo import (
        create_engine_config,
        parse_args,
    )

    args = parse_args([
        "--model-type", "llama", "run",
        "--model-path", "x", "--compiled-model-path", "y",
        "--enable-eagle-speculation", "--speculation-length", "3",
    ])
    ec = create_engine_config(args, "llama")
    assert ec.on_device_sampling_config is not None
    assert ec.on_device_sampling_config.top_k == 1
    assert ec.enable_eagle_speculation


def test_inference_demo_quantize_moe_flag(tmp_path):
    """--quantize-moe runtime-quantizes expert stacks (tiny mixtral)."""
    from transformers import MixtralConfig, MixtralForCausalLM

    from neuronx_distributed_inference_amd.inference_demo import main

    torch.manual_seed(0)
    cfg = MixtralConfig(
        vocab_size=128, hidden_size=64, intermediate_size=512,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=4, num_experts_per_tok=2,
        max_position_embeddings=256, tie_word_embeddings=False)
    m = MixtralForCausalLM(cfg)
    path = tmp_path / "mx"
    m.save_pretrained(str(path), safe_serialization=True)
    rc = main([
        "--model-type", "mixtral", "run",
        "--model-path", str(path),
        "--compiled-model-path", str(tmp_path / "cmx"),
        "--prompt-ids", json.dumps([[1, 2, 3, 4]]),
        "--max-new-tokens", "4",
        "--on-cpu", "--torch-dtype", "float32",
        "--batch-size", "1", "--max-context-length", "16", "--seq-len", "32",
        "--quantize-moe", "fp8",
    ])
    assert rc == 0
