from neuronx_distributed_inference_amd.models.qwen3_moe.modeling_qwen3_moe import (  # noqa: F401
    Qwen3MoeForCausalLM,
    Qwen3MoeInferenceConfig,
)
