from neuronx_distributed_inference_amd.models.whisper.modeling_whisper import (  # noqa: F401
    WhisperForConditionalGeneration,
    WhisperInferenceConfig,
)
