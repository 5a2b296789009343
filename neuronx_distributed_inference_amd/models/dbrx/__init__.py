from neuronx_distributed_inference_amd.models.dbrx.modeling_dbrx import (  # noqa: F401
    DbrxForCausalLM,
    DbrxInferenceConfig,
)
