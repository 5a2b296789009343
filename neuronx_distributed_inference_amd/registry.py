"""Model-hub registry (ref `utils/constants.py:52-69` MODEL_TYPES).

Maps HF `model_type` / CLI model-type strings to (config class, causal-LM
class). Lazy imports keep CLI startup fast.
"""

from __future__ import annotations

import importlib
from typing import Dict, Tuple, Type

_REGISTRY: Dict[str, Tuple[str, str, str]] = {
    # model_type: (module, config_cls, model_cls)
    "llama": (
        "neuronx_distributed_inference_amd.models.llama.modeling_llama",
        "LlamaInferenceConfig",
        "LlamaForCausalLM",
    ),
    "mistral": (
        "neuronx_distributed_inference_amd.models.mistral.modeling_mistral",
        "MistralInferenceConfig",
        "MistralForCausalLM",
    ),
    "qwen2": (
        "neuronx_distributed_inference_amd.models.qwen2.modeling_qwen2",
        "Qwen2InferenceConfig",
        "Qwen2ForCausalLM",
    ),
    "qwen3": (
        "neuronx_distributed_inference_amd.models.qwen3.modeling_qwen3",
        "Qwen3InferenceConfig",
        "Qwen3ForCausalLM",
    ),
    "mixtral": (
        "neuronx_distributed_inference_amd.models.mixtral.modeling_mixtral",
        "MixtralInferenceConfig",
        "MixtralForCausalLM",
    ),
    "dbrx": (
        "neuronx_distributed_inference_amd.models.dbrx.modeling_dbrx",
        "DbrxInferenceConfig",
        "DbrxForCausalLM",
    ),
    "gpt_oss": (
        "neuronx_distributed_inference_amd.models.gpt_oss.modeling_gpt_oss",
        "GptOssInferenceConfig",
        "GptOssForCausalLM",
    ),
    "deepseek_v3": (
        "neuronx_distributed_inference_amd.models.deepseek.modeling_deepseek",
        "DeepseekV3InferenceConfig",
        "DeepseekV3ForCausalLM",
    ),
    "gemma3": (
        "neuronx_distributed_inference_amd.models.gemma3.modeling_gemma3",
        "Gemma3InferenceConfig",
        "Gemma3ForCausalLM",
    ),
    "llama4": (
        "neuronx_distributed_inference_amd.models.llama4.modeling_llama4",
        "Llama4InferenceConfig",
        "Llama4ForCausalLM",
    ),
    "qwen3_moe": (
        "neuronx_distributed_inference_amd.models.qwen3_moe.modeling_qwen3_moe",
        "Qwen3MoeInferenceConfig",
        "Qwen3MoeForCausalLM",
    ),
    # conditional-generation families (multimodal / speech): resolved for
    # programmatic use; the CLI's causal-lm run path does not drive these
    "whisper": (
        "neuronx_distributed_inference_amd.models.whisper.modeling_whisper",
        "WhisperInferenceConfig",
        "WhisperForConditionalGeneration",
    ),
    "mllama": (
        "neuronx_distributed_inference_amd.models.mllama.modeling_mllama",
        "MllamaInferenceConfig",
        "MllamaForConditionalGeneration",
    ),
    "pixtral": (
        "neuronx_distributed_inference_amd.models.pixtral.modeling_pixtral",
        "PixtralInferenceConfig",
        "PixtralForConditionalGeneration",
    ),
    "qwen2_vl": (
        "neuronx_distributed_inference_amd.models.qwen2_vl.modeling_qwen2_vl",
        "Qwen2VLInferenceConfig",
        "Qwen2VLForConditionalGeneration",
    ),
    "qwen3_vl": (
        "neuronx_distributed_inference_amd.models.qwen3_vl.modeling_qwen3_vl",
        "Qwen3VLInferenceConfig",
        "Qwen3VLForConditionalGeneration",
    ),
    "flux": (
        "neuronx_distributed_inference_amd.models.flux.modeling_flux",
        "FluxInferenceConfig",
        "FluxTransformer2DModel",
    ),
    "llama4_vision": (
        "neuronx_distributed_inference_amd.models.llama4.modeling_llama4",
        "Llama4VisionInferenceConfig",
        "Llama4ForConditionalGeneration",
    ),
}


def register_model(model_type: str, module: str, config_cls: str, model_cls: str):
    _REGISTRY[model_type] = (module, config_cls, model_cls)


def available_model_types():
    return sorted(_REGISTRY)


def get_model_classes(model_type: str) -> Tuple[Type, Type]:
    if model_type not in _REGISTRY:
        raise KeyError(
            f"unknown model type {model_type!r}; available: {available_model_types()}"
        )
    mod_name, cfg_name, mdl_name = _REGISTRY[model_type]
    mod = importlib.import_module(mod_name)
    return getattr(mod, cfg_name), getattr(mod, mdl_name)
