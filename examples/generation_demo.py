"""Minimal programmatic generation example (ref examples/generation_demo.py).

Compile once, load per process, generate. For the full pipeline with
accuracy checks and benchmarking use `inference_demo` (see README).
"""

import torch

from neuronx_distributed_inference_amd.config import (
    EngineConfig,
    OnDeviceSamplingConfig,
)
from neuronx_distributed_inference_amd.models.application_base import (
    InferenceApplication,
)
from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
    LlamaForCausalLM,
    LlamaInferenceConfig,
)
from neuronx_distributed_inference_amd.parallel import state as ps
from neuronx_distributed_inference_amd.utils.hf_adapter import (
    GenerationConfig,
    HuggingFaceGenerationAdapter,
)

MODEL_PATH = "/ckpt/llama-3.1-8b"          # HF checkpoint dir
COMPILED_PATH = "/compiled/llama-3.1-8b"   # sharded-weights output dir


def main():
    ps.init_distributed_from_env()
    ps.initialize_model_parallel(tensor_model_parallel_size=1)

    ec = EngineConfig(
        batch_size=4, max_batch_size=4, seq_len=4096, max_length=4096,
        max_context_length=2048, torch_dtype="bfloat16", tp_degree=1,
        on_device_sampling_config=OnDeviceSamplingConfig(top_k=1),
        enable_hip_graphs=True, fused_qkv=True, fused_gate_up=True,
    )
    config = LlamaInferenceConfig.from_hf_config(MODEL_PATH, neuron_config=ec)

    app = InferenceApplication(LlamaForCausalLM, config, model_path=MODEL_PATH)
    app.compile(COMPILED_PATH)             # serialize config + weight shards
    model = app.load(COMPILED_PATH)        # load shard, warmup, capture graphs

    from transformers import AutoTokenizer

    tok = AutoTokenizer.from_pretrained(COMPILED_PATH)
    ids = tok(["I believe the meaning of life is"], return_tensors="pt")
    adapter = HuggingFaceGenerationAdapter(model)
    out = adapter.generate(
        ids.input_ids.to(next(model.parameters()).device),
        generation_config=GenerationConfig(max_new_tokens=64),
    )
    print(tok.batch_decode(out, skip_special_tokens=True)[0])


if __name__ == "__main__":
    main()
