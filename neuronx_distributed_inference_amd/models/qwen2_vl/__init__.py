from neuronx_distributed_inference_amd.models.qwen2_vl.modeling_qwen2_vl import (  # noqa: F401
    Qwen2VLForConditionalGeneration,
    Qwen2VLInferenceConfig,
)
