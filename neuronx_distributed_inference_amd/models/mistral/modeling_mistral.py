"""Mistral: Llama architecture + (optional) sliding-window attention
(ref `models/mistral/`, a thin delta off the llama base)."""

from __future__ import annotations

from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
    LlamaForCausalLM,
    LlamaInferenceConfig,
    LlamaModel,
)


class MistralInferenceConfig(LlamaInferenceConfig):
    def add_derived_config(self):
        super().add_derived_config()
        if not hasattr(self, "sliding_window"):
            self.sliding_window = None


class MistralModel(LlamaModel):
    pass


class MistralForCausalLM(LlamaForCausalLM):
    _model_cls = MistralModel
    _config_cls = MistralInferenceConfig
