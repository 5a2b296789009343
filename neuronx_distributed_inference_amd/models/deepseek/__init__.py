from neuronx_distributed_inference_amd.models.deepseek.modeling_deepseek import (  # noqa: F401
    DeepseekV3ForCausalLM,
    DeepseekV3InferenceConfig,
)
