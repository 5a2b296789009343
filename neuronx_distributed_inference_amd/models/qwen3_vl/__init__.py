from neuronx_distributed_inference_amd.models.qwen3_vl.modeling_qwen3_vl import (  # noqa: F401
    Qwen3VLForConditionalGeneration,
    Qwen3VLInferenceConfig,
)
