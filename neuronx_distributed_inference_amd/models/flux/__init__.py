from neuronx_distributed_inference_amd.models.flux.modeling_flux import (  # noqa: F401
    FluxInferenceConfig,
    FluxPipeline,
    FluxTransformer2DModel,
    FlowMatchEulerScheduler,
)
