from neuronx_distributed_inference_amd.models.qwen2.modeling_qwen2 import (  # noqa: F401
    Qwen2ForCausalLM,
    Qwen2InferenceConfig,
)
