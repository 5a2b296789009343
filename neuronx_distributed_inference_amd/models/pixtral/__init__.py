from neuronx_distributed_inference_amd.models.pixtral.modeling_pixtral import (  # noqa: F401
    PixtralForConditionalGeneration,
    PixtralInferenceConfig,
)
