"""Model base: the device graph body and the host-side dispatcher.

Analogs of the reference's `models/model_base.py`:

- :class:`DecoderModel` ~ ``NeuronBaseModel`` (:70-1597): embeddings ->
  decoder layers (attention + MLP with the HIP kernels + RCCL collectives)
  -> norm -> last-token gather -> lm_head -> on-device sampler. One module
  instance serves every phase; context-encoding vs token-generation is
  selected from the input shape exactly like the reference
  (`_is_context_encoding` :181, `_is_for_speculation` :184). This is the
  module that hipGraph capture records per (batch, seq) bucket.

- :class:`CausalLMBase` ~ ``NeuronBaseForCausalLM`` (:3024-3960): host-side
  phase dispatch by ``position_ids.min() == 0`` (:3546), submodel wrappers,
  ``kv_cache_populated`` / ``reset()`` bookkeeping (:3926).
"""

from __future__ import annotations

import logging
from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd.config import InferenceConfig
from neuronx_distributed_inference_amd.modules.attention.attention_base import (
    PHASE_CTE,
    PHASE_TKG,
)
from neuronx_distributed_inference_amd.modules.generation.sampling import (
    Sampler,
    prepare_sampling_params,
)
from neuronx_distributed_inference_amd.modules.kvcache.kv_cache_manager import KVCacheManager
from neuronx_distributed_inference_amd.modules.norm import RMSNorm
from neuronx_distributed_inference_amd.modules.rope import RotaryEmbedding
from neuronx_distributed_inference_amd.parallel import mappings
from neuronx_distributed_inference_amd.parallel import state as ps
from neuronx_distributed_inference_amd.parallel.layers import (
    ColumnParallelLinear,
    ParallelEmbedding,
)

logger = logging.getLogger("nxdi_amd")


@dataclass
class ModelOutput:
    """Host-visible output of one forward (HF CausalLMOutputWithPast-shaped,
    ref model_base.py:3797-3862)."""

    tokens: Optional[torch.Tensor] = None   # (B,) or (B, spec) sampled ids
    logits: Optional[torch.Tensor] = None   # (B, S, V) when output_logits
    hidden_states: Optional[torch.Tensor] = None


class DecoderModel(nn.Module):
    """The device graph body. Subclasses implement `build_layers` and
    (optionally) `build_rotary`."""

    def __init__(self, config: InferenceConfig):
        super().__init__()
        ec = config.neuron_config
        self.config = config
        self.ec = ec
        self.vocab_size = config.vocab_size
        self.hidden_size = config.hidden_size
        self.num_layers = config.num_hidden_layers
        self.dtype = ec.torch_dtype
        # EAGLE hidden capture must be armed BEFORE hipGraph capture so the
        # CTE graph records the (static) full_hidden write; the demo sets
        # enable_eagle_speculation on the config, adapters may also toggle
        # the attribute directly in eager runs
        self.eagle_capture_hidden = bool(
            getattr(ec, "enable_eagle_speculation", False))
        if getattr(ec, "eagle3_capture_layers", None):
            self.eagle3_capture_layers = list(ec.eagle3_capture_layers)

        self.embed_tokens = ParallelEmbedding(
            config.vocab_size, config.hidden_size, dtype=self.dtype,
            tp_degree=ec.tp_degree,
        )
        self.rotary = self.build_rotary(config)
        self.layers = self.build_layers(config)
        self.norm = self.build_norm(config)
        self.lm_head = ColumnParallelLinear(
            config.hidden_size, config.vocab_size, bias=False, dtype=self.dtype,
            tp_degree=ec.tp_degree,
        )
        head_dim = getattr(config, "head_dim", None) or config.hidden_size // config.num_attention_heads
        n_kv_heads = getattr(config, "num_key_value_heads", config.num_attention_heads)
        # FP8 KV-cache quantization (ref kv_cache_manager.py:636-692): cache
        # stored OCP e4m3, quantized on write and dequantized in the decode
        # kernel — halves KV HBM traffic and doubles cache capacity
        cache_dtype = torch.float8_e4m3fn if ec.kv_cache_quant else ec.torch_dtype
        # attention-DP decode runs at reduced TP (tp/dp), so each rank's
        # cache holds correspondingly MORE kv heads (and B/dp rows)
        cache_tp = ec.tp_degree // ec.attention_dp_degree \
            if ec.attention_dp_degree > 1 else ec.tp_degree
        if ec.is_block_kv_layout:
            from neuronx_distributed_inference_amd.modules.kvcache.block_kv_cache_manager import (
                BlockKVCacheManager,
            )

            num_blocks = ec.pa_num_blocks or (
                (ec.max_length + ec.pa_block_size - 1) // ec.pa_block_size
                * ec.kv_cache_batch_size
            )
            self.kv_mgr = BlockKVCacheManager(
                num_layers=self.num_layers, num_kv_heads=n_kv_heads,
                head_dim=head_dim, num_blocks=num_blocks,
                block_size=ec.pa_block_size, dtype=cache_dtype,
                tp_degree=cache_tp,
            )
        else:
            # flash decoding: the decode cache is S-sharded over the
            # kv-shard group (strided rows) — each rank holds 1/fd of S
            fd = ps.get_kv_shard_degree() if getattr(
                ec, "flash_decoding_enabled", False) else 1
            layer_sizes = None
            if ec.compact_swa_kv and fd == 1:
                sw = getattr(config, "sliding_window", None) or 0
                lt = getattr(config, "layer_types", None)
                if sw and sw < ec.max_length:
                    if lt is not None:
                        layer_sizes = [
                            sw if t == "sliding_attention" else ec.max_length
                            for t in lt
                        ]
                    else:
                        layer_sizes = [sw] * self.num_layers
            self.kv_mgr = KVCacheManager(
                num_layers=self.num_layers,
                num_kv_heads=n_kv_heads,
                head_dim=head_dim,
                max_len=(ec.max_length + fd - 1) // fd,
                batch_size=ec.kv_cache_batch_size,
                dtype=cache_dtype,
                padding_rows=ec.kv_cache_padding_size,
                tp_degree=cache_tp,
                layer_cache_sizes=layer_sizes,
            )
        self.sampler = (
            Sampler(ec.on_device_sampling_config, config.vocab_size)
            if ec.on_device_sampling_config
            else None
        )

    # -- subclass hooks -------------------------------------------------
    def build_rotary(self, config: InferenceConfig) -> RotaryEmbedding:
        head_dim = getattr(config, "head_dim", None) or config.hidden_size // config.num_attention_heads
        return RotaryEmbedding(
            head_dim,
            max_position_embeddings=getattr(config, "max_position_embeddings", 4096),
            base=getattr(config, "rope_theta", 10000.0),
            scaling=getattr(config, "rope_scaling", None),
        )

    def build_layers(self, config: InferenceConfig) -> nn.ModuleList:
        raise NotImplementedError

    def build_norm(self, config: InferenceConfig) -> nn.Module:
        return RMSNorm(
            config.hidden_size, eps=getattr(config, "rms_norm_eps", 1e-6),
            dtype=self.dtype,
        )

    # -- phase detection (ref model_base.py:181-186) --------------------
    # Shape-only so no host sync happens inside a captured decode step;
    # speculation passes `phase` explicitly.
    @staticmethod
    def _phase(input_ids: torch.Tensor, position_ids: torch.Tensor) -> str:
        return PHASE_CTE if input_ids.shape[-1] > 1 else PHASE_TKG

    # -- the graph body -------------------------------------------------
    def forward(
        self,
        input_ids: torch.Tensor,
        attention_mask: Optional[torch.Tensor],
        position_ids: torch.Tensor,
        seq_ids: Optional[torch.Tensor] = None,
        sampling_params: Optional[torch.Tensor] = None,
        phase: Optional[str] = None,
        block_table: Optional[torch.Tensor] = None,
    ):
        phase = phase or self._phase(input_ids, position_ids)
        B, S = input_ids.shape
        if block_table is not None and getattr(self.kv_mgr, "is_paged", False):
            from neuronx_distributed_inference_amd.modules.kvcache.block_kv_cache_manager import (
                generate_slot_mapping,
            )

            slots = generate_slot_mapping(block_table, position_ids, self.kv_mgr.block_size)
            self.kv_mgr.set_step_context(block_table, slots)
        from neuronx_distributed_inference_amd.parallel import state as _ps

        tp = _ps.get_tensor_model_parallel_size()
        # SP: prefill activations sharded along S (ref §2.8 SP; disabled for
        # TKG exactly like the reference, model_base.py:3093,3135)
        sp_active = (
            self.ec.sequence_parallel_enabled
            and phase == PHASE_CTE and tp > 1 and S % tp == 0 and S >= tp
        )
        h = self.embed_tokens(input_ids, sequence_parallel=sp_active) \
            if sp_active else self.embed_tokens(input_ids)
        # multimodal splice (llava/pixtral pattern): projected vision
        # features replace the image-placeholder token embeddings. The mask
        # derives from input_ids, so bucket padding is handled for free.
        mm = getattr(self, "multimodal_ctx", None)
        if mm and phase == PHASE_CTE and not sp_active:
            # one splice per modality: [(placeholder_token_id, features), ...]
            # (single image/videos: qwen2-vl passes both in one prefill)
            splices = mm.get("splices")
            if splices is None and mm.get("features") is not None:
                splices = [(mm["token_id"], mm["features"])]
            for tok, feats in splices or []:
                mask = (input_ids == tok).unsqueeze(-1)
                h = h.masked_scatter(mask, feats.to(h.dtype).reshape(-1, h.shape[-1]))
        # models with non-1D rope (qwen2-vl m-rope) install a custom hook
        cos_sin_fn = getattr(self, "_cos_sin_fn", None) or self.rotary.get_cos_sin
        cos, sin = cos_sin_fn(position_ids, device=h.device)
        # int32 once here: the KV/attention kernels take int32 and a per-layer
        # .to() would launch 4 conversion kernels per layer per step
        position_ids = position_ids.to(torch.int32)
        if seq_ids is not None:
            seq_ids = seq_ids.to(torch.int32)
        residual = None
        post_fns = getattr(self, "_layer_post_fns", None)
        for li, layer in enumerate(self.layers):
            if sp_active:
                h, residual = layer(
                    h, residual, cos, sin, self.kv_mgr, position_ids, phase,
                    attention_mask, seq_ids, sequence_parallel=True,
                )
            else:
                h, residual = layer(
                    h, residual, cos, sin, self.kv_mgr, position_ids, phase,
                    attention_mask, seq_ids,
                )
            # per-layer post hook (qwen3-vl deepstack feature injection);
            # adding to h is equivalent to adding to the logical hidden
            # (h + deferred residual)
            if post_fns and li in post_fns:
                h = post_fns[li](h, input_ids, phase)
            # EAGLE-3 conditions the draft on SEVERAL target layers'
            # hiddens (ref model_base.py:1402-1437: 3 layers concatenated)
            cap = getattr(self, "eagle3_capture_layers", None)
            if cap and li in cap:
                if li == cap[0]:
                    self.eagle3_hiddens = {}
                self.eagle3_hiddens[li] = (h + residual) if residual is not None else h
        if residual is not None:
            h, _ = self.norm(h, residual)
        else:
            h = self.norm(h)
        is_speculation = phase == PHASE_TKG and S > 1
        h_last = None
        if sp_active and phase == PHASE_CTE and S > 1:
            # SP exit: move ONE token per row instead of all-gathering the
            # (B, S, H) activations (ref seq_parallel_logits_slice.py:9-70
            # — index math + masking + reduce). Each rank contributes its
            # own shard's last-token rows; the sum-reduce assembles them.
            if attention_mask is not None:
                last = attention_mask.sum(dim=-1).long() - 1
            else:
                last = torch.full((B,), S - 1, dtype=torch.long, device=h.device)
            Sl = h.shape[1]                      # local shard length
            rank = mappings.ps.get_tensor_model_parallel_rank()
            owner = last // Sl
            local = (last % Sl).view(B, 1, 1).expand(B, 1, h.shape[-1])
            mine = (owner == rank).view(B, 1, 1)
            contrib = torch.where(mine, h.gather(1, local), torch.zeros_like(h[:, :1]))
            h_last = mappings.reduce_from_tensor_parallel_region(contrib.contiguous())
        elif sp_active:
            # non-CTE SP (disabled for TKG like the reference) — full exit
            h = mappings.gather_along_dim(h, dim=1)

        if h_last is not None:
            pass
        elif phase == PHASE_CTE and S > 1:
            if attention_mask is not None:
                last = attention_mask.sum(dim=-1).long() - 1
            else:
                last = torch.full((B,), S - 1, dtype=torch.long, device=h.device)
            h_last = h[torch.arange(B, device=h.device), last].unsqueeze(1)
        elif is_speculation:
            h_last = h  # verify pass needs every position's logits
        else:
            h_last = h[:, -1:, :]

        # kept for medusa/eagle heads (graph-safe: a static tensor each capture)
        self.last_hidden = h_last
        # EAGLE draft prefill needs the FULL prompt hidden states (ref
        # EAGLE CTE, model_base.py:1931-2093); opt-in to avoid keeping
        # (B, S, H) alive on every CTE
        if getattr(self, "eagle_capture_hidden", False):
            self.full_hidden = h
        logits_shard = self.lm_head(h_last)  # (B, 1|S, V/tp)

        # rejection-sampling speculation reads the per-position top-Kg
        # distribution of every draft/verify pass (ref draft_probs /
        # target_probs flow, model_base.py:1678-1759); stored as an
        # attribute so FusedSpeculationModel can read it inside the same
        # graph capture (same pattern as last_hidden)
        if getattr(self, "spec_capture_topk", 0):
            from neuronx_distributed_inference_amd.parallel.ops import (
                distributed_topk,
            )

            self.spec_topk = distributed_topk(
                logits_shard, int(self.spec_capture_topk), dim=-1
            )

        tokens = None
        logits = None
        if self.sampler is not None:
            if is_speculation:
                # greedy per-position target tokens for the accept test
                from neuronx_distributed_inference_amd.parallel.ops import (
                    distributed_argmax,
                )

                tokens = distributed_argmax(logits_shard, dim=-1)  # (B, S)
            else:
                tokens = self.sampler(logits_shard, sampling_params)
        if self.sampler is None or self.ec.output_logits:
            logits = mappings.gather_from_tensor_parallel_region(logits_shard, dim=-1)
        return tokens, logits

    def reset_cache(self):
        self.kv_mgr.reset()


class CausalLMBase(nn.Module):
    """Host-side application model: owns the DecoderModel, dispatches
    prefill vs decode, tracks KV state. Submodel graph capture and bucketing
    are layered on by `ModelWrapper`/`ApplicationBase`."""

    _model_cls = DecoderModel

    def __init__(self, config: InferenceConfig, model: Optional[DecoderModel] = None):
        super().__init__()
        self.config = config
        self.ec = config.neuron_config
        self.model = model or self._model_cls(config)
        self.kv_cache_populated = False
        self.context_encoding_model = None   # ModelWrapper (ref :3107)
        self.token_generation_model = None   # ModelWrapper (ref :3130)
        self.speculation_model = None

    def enable_submodels(self):
        """Create the per-phase submodel wrappers (ref enable_context_encoding
        / enable_token_generation, model_base.py:3107-3190)."""
        from neuronx_distributed_inference_amd.models.model_wrapper import (
            CONTEXT_ENCODING_MODEL_TAG,
            TOKEN_GENERATION_MODEL_TAG,
            ModelWrapper,
        )

        self.context_encoding_model = ModelWrapper(
            self.config, self.model, tag=CONTEXT_ENCODING_MODEL_TAG,
            batch_size=self.ec.ctx_batch_size,
        )
        self.token_generation_model = ModelWrapper(
            self.config, self.model, tag=TOKEN_GENERATION_MODEL_TAG,
            batch_size=self.ec.tkg_batch_size, n_active_tokens=1,
        )
        if self.ec.enable_speculation:
            from neuronx_distributed_inference_amd.models.model_wrapper import (
                SPECULATION_MODEL_TAG,
            )

            self.speculation_model = ModelWrapper(
                self.config, self.model, tag=SPECULATION_MODEL_TAG,
                batch_size=self.ec.tkg_batch_size,
                n_active_tokens=self.ec.speculation_length + 1,
            )

    def warmup(self):
        """Run/capture every submodel bucket once (ref warmup :349-373).
        On GPU this is where hipGraph capture happens."""
        if self.context_encoding_model is None:
            self.enable_submodels()
        self.token_generation_model.capture()
        if self.speculation_model is not None:
            self.speculation_model.capture()

    # ------------------------------------------------------------------
    @classmethod
    def from_pretrained(cls, model_path: str, config: InferenceConfig) -> "CausalLMBase":
        """Instantiate with HF checkpoint weights (full, unsharded — used by
        the CPU path and as the sharding source)."""
        obj = cls(config)
        from neuronx_distributed_inference_amd.modules.checkpoint import load_state_dict

        hf_sd = load_state_dict(model_path)
        sd = cls.convert_hf_to_native_state_dict(hf_sd, config)
        from neuronx_distributed_inference_amd.utils.checkpoint_sharding import (
            shard_state_dict_for_rank,
        )
        from neuronx_distributed_inference_amd.parallel import state as ps

        tp = config.neuron_config.tp_degree
        if tp > 1:
            sd = shard_state_dict_for_rank(
                sd, obj.model, ps.get_tensor_model_parallel_rank(), tp
            )
        missing, unexpected = obj.model.load_state_dict(sd, strict=False)
        real_missing = [m for m in missing if not m.startswith("kv_mgr.")]
        if real_missing:
            raise RuntimeError(f"missing weights: {real_missing[:10]}")
        if unexpected:
            logger.warning("unexpected checkpoint keys: %s", unexpected[:10])
        return obj

    @staticmethod
    def convert_hf_to_native_state_dict(sd: Dict[str, torch.Tensor], config) -> Dict[str, torch.Tensor]:
        """Subclasses rewrite HF key names to native module names
        (ref convert_hf_to_neuron_state_dict, modeling_llama.py:1208-1276)."""
        return sd

    # ------------------------------------------------------------------
    def _is_prefill(self, position_ids: torch.Tensor) -> bool:
        # ref model_base.py:3546
        return bool(position_ids.min() == 0) and position_ids.shape[-1] > 1

    def infer_attention_mask(self, position_ids: torch.Tensor) -> torch.Tensor:
        """Rebuild a padding mask from position ids (ref :3485-3504)."""
        B, S = position_ids.shape
        if S == 1:
            seq_len = int(position_ids.max().item()) + 1
            ar = torch.arange(seq_len, device=position_ids.device).view(1, -1)
            return (ar <= position_ids).to(torch.int32)
        return (position_ids >= 0).to(torch.int32) | (position_ids == 0).to(torch.int32)

    def forward(
        self,
        input_ids: torch.Tensor,
        attention_mask: Optional[torch.Tensor] = None,
        position_ids: Optional[torch.Tensor] = None,
        seq_ids: Optional[torch.Tensor] = None,
        sampling_params: Optional[torch.Tensor] = None,
        block_table: Optional[torch.Tensor] = None,
        phase: Optional[str] = None,
    ) -> ModelOutput:
        B, S = input_ids.shape
        if position_ids is None:
            if attention_mask is not None:
                # padded entries get position -1 so the KV scatter skips them
                # (a clamped position would overwrite a real cache row)
                cum = attention_mask.long().cumsum(-1) - 1
                position_ids = torch.where(
                    attention_mask.bool(), cum, torch.full_like(cum, -1)
                )[:, -S:]
            else:
                position_ids = torch.arange(S, device=input_ids.device).view(1, S).expand(B, S)
        if sampling_params is None and self.ec.on_device_sampling_config:
            # memoized: building this tensor does host->device copies, which
            # would otherwise run every decode step
            cached = getattr(self, "_default_sampling_params", None)
            if cached is None or cached.shape[0] != B or cached.device != input_ids.device:
                c = self.ec.on_device_sampling_config
                cached = prepare_sampling_params(
                    B, c.top_k if c.do_sample else 1, c.top_p, c.temperature,
                    device=input_ids.device,
                )
                self._default_sampling_params = cached
            sampling_params = cached
        # drop an all-ones padding mask on the host so the device path never
        # syncs per layer
        if attention_mask is not None and S > 1 and bool(attention_mask.all()):
            attention_mask = None
        if seq_ids is None:
            seq_ids = torch.arange(B, device=input_ids.device)
        # route through the submodel wrappers (bucketing + hipGraph replay)
        # when enabled, else call the module directly
        # windowed context encoding (ref model_base.py:878-933): prefill in
        # window-sized chunks to cap the attention working set; each chunk
        # runs the multi-token prefix path (causal within the chunk, full
        # prior from the cache), which is also the chunked-prefill compute.
        # attention_mask=None (the common dense/unpadded batch, incl. one
        # dropped above) is fine: the last-token gather falls back to S-1.
        wce = getattr(self.ec, "windowed_context_encoding_size", 0)
        if S > 1 and not self.kv_cache_populated and wce and S > wce:
            return self._windowed_context_encoding(
                input_ids, attention_mask, position_ids, seq_ids, sampling_params, wce
            )

        is_spec_step = (S > 1 and self.kv_cache_populated) or \
            (phase == "token_generation" and S > 1)
        if block_table is not None:
            # paged requests bypass the submodel wrappers (per-request block
            # tables change shape semantics; serving drives this path)
            tokens, logits = self.model(
                input_ids, attention_mask, position_ids, seq_ids,
                sampling_params, phase=phase, block_table=block_table,
            )
            self.kv_cache_populated = True
            return ModelOutput(tokens=tokens, logits=logits)
        if is_spec_step and self.speculation_model is not None:
            tokens, logits = self.speculation_model(
                input_ids, attention_mask, position_ids, seq_ids, sampling_params
            )
        elif S > 1 and not is_spec_step and self.context_encoding_model is not None:
            tokens, logits = self.context_encoding_model(
                input_ids, attention_mask, position_ids, seq_ids, sampling_params
            )
        elif S == 1 and self.token_generation_model is not None:
            tokens, logits = self.token_generation_model(
                input_ids, attention_mask, position_ids, seq_ids, sampling_params
            )
        else:
            from neuronx_distributed_inference_amd.modules.attention.attention_base import (
                PHASE_TKG,
            )

            tokens, logits = self.model(
                input_ids, attention_mask, position_ids, seq_ids, sampling_params,
                phase=PHASE_TKG if is_spec_step else None,
            )
        self.kv_cache_populated = True
        return ModelOutput(tokens=tokens, logits=logits)

    def _windowed_context_encoding(
        self, input_ids, attention_mask, position_ids, seq_ids, sampling_params,
        window: int,
    ) -> "ModelOutput":
        """Prefill in window-sized chunks (ref windowed CTE,
        model_base.py:878-933). Every chunk runs the multi-token prefix path
        (greedy per-position tokens); the per-row next token is gathered at
        each row's true last index."""
        from neuronx_distributed_inference_amd.modules.attention.attention_base import (
            PHASE_TKG,
        )

        B, S = input_ids.shape
        chunks = []
        start = 0
        while start < S:
            end = min(start + window, S)
            t, l = self.model(
                input_ids[:, start:end], None, position_ids[:, start:end],
                seq_ids, sampling_params, phase=PHASE_TKG,
            )
            if t is None:  # no on-device sampler configured: greedy off logits
                t = l.argmax(dim=-1)
            chunks.append(t.view(B, end - start))
            start = end
        all_tok = torch.cat(chunks, dim=1)              # (B, S) greedy per pos
        if attention_mask is not None:
            last = (attention_mask.sum(dim=-1).long() - 1).clamp(min=0)
        else:  # dense/unpadded batch: every row's last token is at S-1
            last = torch.full((B,), S - 1, dtype=torch.long, device=all_tok.device)
        tokens = all_tok.gather(1, last.view(B, 1)).view(B)
        self.kv_cache_populated = True
        return ModelOutput(tokens=tokens, logits=None)

    def attach_tensor_capture(self):
        """Context manager capturing the configured modules' outputs per
        step (ref TensorCaptureConfig config.py:1121-1169 + collection
        model_base.py:1026-1149). Uses the engine config's
        tensor_capture_config; defaults to every decoder layer."""
        from neuronx_distributed_inference_amd.utils.tensor_capture import (
            TensorCapture,
        )

        cfg = getattr(self.ec, "tensor_capture_config", None)
        names = (cfg.module_names if cfg and cfg.module_names else
                 [n for n, _ in self.model.named_modules()
                  if n.startswith("layers.") and n.count(".") == 1])
        return TensorCapture(self.model, names,
                             max_steps=cfg.max_steps if cfg else 16)

    def apply_tensor_replacements(self):
        """Context manager substituting golden tensors into the
        configured modules (ref TensorReplacementConfig
        config.py:1172-1202): isolates which submodule diverges."""
        from neuronx_distributed_inference_amd.utils.tensor_capture import (
            TensorReplacement,
        )

        cfg = self.ec.tensor_replacement_config
        repl = {name: torch.load(path) for name, path in cfg.replacements.items()}
        return TensorReplacement(self.model, repl)

    def reset(self):
        """Clear KV state (ref model_base.py:3926)."""
        self.kv_cache_populated = False
        self.model.reset_cache()
