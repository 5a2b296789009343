from neuronx_distributed_inference_amd.models.mixtral.modeling_mixtral import (  # noqa: F401
    MixtralForCausalLM,
    MixtralInferenceConfig,
)
