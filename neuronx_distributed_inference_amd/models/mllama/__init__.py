from neuronx_distributed_inference_amd.models.mllama.modeling_mllama import (  # noqa: F401
    MllamaForConditionalGeneration,
    MllamaInferenceConfig,
)
