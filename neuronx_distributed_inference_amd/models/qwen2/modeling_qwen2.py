"""Qwen2/Qwen2.5: Llama architecture with QKV projection biases
(ref `models/qwen2/`)."""

from __future__ import annotations

from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
    LlamaForCausalLM,
    LlamaInferenceConfig,
    LlamaModel,
)


class Qwen2InferenceConfig(LlamaInferenceConfig):
    def add_derived_config(self):
        super().add_derived_config()
        # Qwen2 has biases on q/k/v projections only
        if not hasattr(self, "attention_bias"):
            self.attention_bias = True


class Qwen2Model(LlamaModel):
    pass


class Qwen2ForCausalLM(LlamaForCausalLM):
    _model_cls = Qwen2Model
    _config_cls = Qwen2InferenceConfig
