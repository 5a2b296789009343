import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (ROCm)")
    config.addinivalue_line("markers", "multiproc: spawns a multi-process gloo world")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def tiny_llama_config():
    """Shared tiny Llama geometry for CPU tests."""
    from neuronx_distributed_inference_amd.config import EngineConfig
    from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
        LlamaInferenceConfig,
    )

    ec = EngineConfig(
        batch_size=2, max_batch_size=2, seq_len=64, max_length=64,
        torch_dtype="float32", tp_degree=1,
    )
    return LlamaInferenceConfig(
        neuron_config=ec,
        vocab_size=128,
        hidden_size=64,
        intermediate_size=128,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        max_position_embeddings=256,
        rms_norm_eps=1e-5,
        rope_theta=10000.0,
        hidden_act="silu",
        tie_word_embeddings=False,
    )
