from neuronx_distributed_inference_amd.models.llama.modeling_llama import (  # noqa: F401
    LlamaInferenceConfig,
    LlamaForCausalLM,
    LlamaModel,
)
