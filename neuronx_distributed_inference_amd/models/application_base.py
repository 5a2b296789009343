"""Application layer: compile / load / warmup and the compiled-model
directory layout.

Analog of the reference's `NeuronApplicationBase`
(`models/application_base.py:68-822`). On MI355X, "compile" is cheap —
there is no neuronx-cc: it serializes the config, pre-shards weights per TP
rank and copies tokenizer files; graph capture (the analog of the NEFF) is
re-done at load()+warmup() time in seconds. Directory layout mirrors the
reference (application_base.py:47,292-316,389-399):

    <compiled_path>/engine_config.json           (accepts neuron_config.json)
    <compiled_path>/weights/tp{r}_sharded_checkpoint.safetensors
    <compiled_path>/tokenizer files (if present beside the checkpoint)
"""

from __future__ import annotations

import logging
import os
import shutil
from typing import Optional, Type

import torch

from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.models.model_base import CausalLMBase
from neuronx_distributed_inference_amd.parallel import state as ps

logger = logging.getLogger("nxdi_amd")

TOKENIZER_FILES = (
    "tokenizer.json", "tokenizer_config.json", "special_tokens_map.json",
    "tokenizer.model", "vocab.json", "merges.txt",
)


class InferenceApplication:
    """Owns one CausalLM and its compiled artifact."""

    def __init__(self, model_cls: Type[CausalLMBase], config: InferenceConfig,
                 model_path: Optional[str] = None):
        self.model_cls = model_cls
        self.config = config
        self.model_path = model_path
        self.causal_lm: Optional[CausalLMBase] = None

    # ------------------------------------------------------------------
    def compile(self, compiled_path: str):
        """Serialize config + per-rank weight shards (ref compile :292-316)."""
        os.makedirs(compiled_path, exist_ok=True)
        self.config.save(compiled_path)
        tp = self.config.neuron_config.tp_degree

        from neuronx_distributed_inference_amd.modules.checkpoint import load_state_dict
        from neuronx_distributed_inference_amd.utils.checkpoint_sharding import (
            save_sharded_checkpoints,
        )

        hf_sd = load_state_dict(self.model_path)
        native_sd = self.model_cls.convert_hf_to_native_state_dict(hf_sd, self.config)
        # a skeleton on the meta device provides the partition metadata
        with torch.device("meta"):
            skeleton = self.model_cls(self.config)
        save_sharded_checkpoints(native_sd, skeleton.model, compiled_path, tp)
        del skeleton

        for name in TOKENIZER_FILES:
            src = os.path.join(self.model_path, name)
            if os.path.exists(src):
                shutil.copy(src, os.path.join(compiled_path, name))
        logger.info("compiled -> %s (tp=%d)", compiled_path, tp)

    # ------------------------------------------------------------------
    def load(self, compiled_path: str, device: Optional[torch.device] = None,
             skip_warmup: bool = False) -> CausalLMBase:
        """Instantiate this rank's model from the compiled dir, move to the
        device, capture decode graphs (ref load :318-347 + warmup :349-373)."""
        self.config = type(self.config).load(compiled_path)
        ec = self.config.neuron_config
        if device is None:
            device = torch.device("cuda", torch.cuda.current_device()) \
                if torch.cuda.is_available() else torch.device("cpu")

        rank = ps.get_tensor_model_parallel_rank()
        from neuronx_distributed_inference_amd.utils.checkpoint_sharding import (
            load_sharded_checkpoint,
        )

        model = self.model_cls(self.config)
        sd = load_sharded_checkpoint(compiled_path, rank)
        missing, unexpected = model.model.load_state_dict(sd, strict=False)
        real_missing = [m for m in missing if not m.startswith("kv_mgr.")]
        if real_missing:
            raise RuntimeError(f"missing weights: {real_missing[:8]}")
        model = model.to(device).eval()
        if ec.quantized:
            from neuronx_distributed_inference_amd.parallel.layers import (
                quantize_model_fp8,
            )

            n = quantize_model_fp8(model.model)
            logger.info("quantized %d linears to fp8", n)
        model.enable_submodels()
        if not skip_warmup and not ec.skip_warmup:
            model.warmup()
        self.causal_lm = model
        return model
