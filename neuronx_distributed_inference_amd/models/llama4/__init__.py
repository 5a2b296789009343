from neuronx_distributed_inference_amd.models.llama4.modeling_llama4 import (  # noqa: F401
    Llama4ForCausalLM,
    Llama4InferenceConfig,
)
