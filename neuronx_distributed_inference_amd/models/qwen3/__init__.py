from neuronx_distributed_inference_amd.models.qwen3.modeling_qwen3 import (  # noqa: F401
    Qwen3ForCausalLM,
    Qwen3InferenceConfig,
)
