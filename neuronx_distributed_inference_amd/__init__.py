"""MI355X-native distributed LLM inference engine.

A from-scratch framework with the capability surface of AWS's NxD Inference
(aws-neuron/neuronx-distributed-inference): submodel decomposition
(context-encoding / token-generation / speculation), (batch, seq) bucketing,
KV-cache managers, on-device sampling, TP/SP/CP/DP/EP parallelism over RCCL,
an ``inference_demo`` CLI and a compiled-model directory layout.

The compute path is PyTorch-ROCm + hand-written HIP/CDNA4 kernels (gfx950
MFMA, LDS-staged tiles) + RCCL collectives over xGMI.  The reference's
trace -> neuronx-cc AOT pipeline is replaced by per-bucket hipGraph captures.
"""

__version__ = "0.1.0"

from neuronx_distributed_inference_amd.config import (  # noqa: F401
    EngineConfig,
    InferenceConfig,
    MoEEngineConfig,
    OnDeviceSamplingConfig,
    FusedSpecConfig,
    to_torch_dtype,
)
