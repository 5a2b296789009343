from neuronx_distributed_inference_amd.models.gpt_oss.modeling_gpt_oss import (  # noqa: F401
    GptOssForCausalLM,
    GptOssInferenceConfig,
)
