from neuronx_distributed_inference_amd.models.gemma3.modeling_gemma3 import (  # noqa: F401
    Gemma3ForCausalLM,
    Gemma3InferenceConfig,
)
