from neuronx_distributed_inference_amd.models.mistral.modeling_mistral import (  # noqa: F401
    MistralForCausalLM,
    MistralInferenceConfig,
)
