"""Configuration system.

Mirrors the capability surface of the reference config layer
(`neuronx_distributed_inference/models/config.py`: NeuronConfig at :92-643,
InferenceConfig at :850-1038, sub-configs at :1045-1202) with an
MI355X-native engine underneath: instead of neuronx-cc compiler flags the
engine knobs here control hipGraph capture, HIP kernel selection and RCCL
process-group layout.

Serialization contract (compiled-model directory layout parity):
``<compiled_path>/engine_config.json`` holds the full InferenceConfig
(engine config nested under ``"neuron_config"`` for reference-layout
compatibility), and weights live in
``<compiled_path>/weights/tp{rank}_sharded_checkpoint.safetensors``.
"""

from __future__ import annotations

import json
import logging
import os
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import torch

logger = logging.getLogger("nxdi_amd")

CONFIG_FILE = "engine_config.json"
# Reference-compat alias: the reference writes `neuron_config.json`
# (`models/config.py:23`); load() accepts either name.
LEGACY_CONFIG_FILE = "neuron_config.json"

_DTYPE_MAP = {
    "bfloat16": torch.bfloat16,
    "float16": torch.float16,
    "float32": torch.float32,
    "float8_e4m3": torch.float8_e4m3fn,  # OCP e4m3fn (gfx950 FP8; NOT fnuz)
}


def to_torch_dtype(name) -> torch.dtype:
    if isinstance(name, torch.dtype):
        return name
    key = str(name).replace("torch.", "")
    if key not in _DTYPE_MAP:
        raise ValueError(f"unsupported dtype {name!r}; one of {list(_DTYPE_MAP)}")
    return _DTYPE_MAP[key]


def dtype_name(dtype: torch.dtype) -> str:
    for k, v in _DTYPE_MAP.items():
        if v == dtype:
            return k
    return str(dtype).replace("torch.", "")


@dataclass
class OnDeviceSamplingConfig:
    """On-device sampling knobs (ref: config.py:1064-1077)."""

    top_k: int = 1
    top_p: float = 1.0
    temperature: float = 1.0
    dynamic: bool = False          # per-request sampling params tensor
    deterministic: bool = False    # seeded RNG for token-matching tests
    global_topk: int = 256         # candidate set size before top-p
    do_sample: bool = False

    def to_dict(self):
        return dict(self.__dict__)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in d.items() if k in cls.__dataclass_fields__})


@dataclass
class ChunkedPrefillConfig:
    """vLLM-style chunked prefill (ref: config.py:1078-1095)."""

    max_num_seqs: int = 8
    tkg_model_enabled: bool = True

    def to_dict(self):
        return dict(self.__dict__)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in d.items() if k in cls.__dataclass_fields__})


@dataclass
class FusedSpecConfig:
    """Fused speculative decoding: draft + target verified in one captured
    graph (ref: FusedSpecNeuronConfig config.py:1045-1063)."""

    draft_model_path: str = ""
    draft_config: Optional["InferenceConfig"] = None
    speculation_length: int = 5
    # True rejection sampling in the fused round (accept d_j iff rand <
    # p(d_j)/q(d_j), resample from the residual; ref model_base.py:1678-
    # 1759) so speculation composes with top-k/top-p sampling. False =
    # exact greedy token match.
    rejection_sampling: bool = False

    def to_dict(self):
        return {
            "draft_model_path": self.draft_model_path,
            "draft_config": self.draft_config.to_dict() if self.draft_config else None,
            "speculation_length": self.speculation_length,
            "rejection_sampling": self.rejection_sampling,
        }

    @classmethod
    def from_dict(cls, d):
        dc = d.get("draft_config")
        return cls(
            draft_model_path=d.get("draft_model_path", ""),
            draft_config=InferenceConfig.from_dict(dc) if dc else None,
            speculation_length=d.get("speculation_length", 5),
            rejection_sampling=d.get("rejection_sampling", False),
        )


@dataclass
class TensorCaptureConfig:
    """Capture chosen modules' outputs during forward for divergence
    debugging (ref TensorCaptureConfig, config.py:1121-1169; collection
    model_base.py:1026-1149). Wired by CausalLMBase.attach_tensor_capture
    and the CLI's --capture-inputs-on-divergence flow."""

    module_names: List[str] = field(default_factory=list)
    max_steps: int = 16

    def to_dict(self):
        return dict(self.__dict__)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in d.items() if k in cls.__dataclass_fields__})


@dataclass
class TensorReplacementConfig:
    """Feed golden tensors into chosen modules to isolate the diverging
    submodule (ref TensorReplacementConfig, config.py:1172-1202;
    utils/tensor_replacement/registry.py). `replacements` maps module
    name -> path of a torch-saved tensor fed as the module's input."""

    replacements: Dict[str, str] = field(default_factory=dict)

    def to_dict(self):
        return dict(self.__dict__)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in d.items() if k in cls.__dataclass_fields__})


class EngineConfig:
    """The single engine config (analog of the reference's NeuronConfig god
    object, config.py:92-643). Unknown kwargs warn rather than raise, like
    the reference (config.py:639-640)."""

    def __init__(self, **kwargs):
        # --- batch / sequence geometry (ref config.py:130-180) ---
        def _opt(name, default):
            v = kwargs.pop(name, None)
            return default if v is None else v

        self.batch_size: int = _opt("batch_size", 1)
        self.max_batch_size: int = _opt("max_batch_size", self.batch_size)
        self.ctx_batch_size: int = _opt("ctx_batch_size", self.max_batch_size)
        self.tkg_batch_size: int = _opt("tkg_batch_size", self.max_batch_size)
        self.seq_len: int = kwargs.pop("seq_len", 2048)
        self.max_context_length: int = kwargs.pop("max_context_length", self.seq_len)
        self.max_length: int = kwargs.pop("max_length", self.seq_len)
        self.n_active_tokens: int = kwargs.pop("n_active_tokens", 1)
        self.padding_side: str = kwargs.pop("padding_side", "right")
        self.is_continuous_batching: bool = kwargs.pop("is_continuous_batching", False)

        # --- dtypes ---
        self.torch_dtype: torch.dtype = to_torch_dtype(kwargs.pop("torch_dtype", "bfloat16"))
        self.rpl_reduce_dtype: torch.dtype = to_torch_dtype(
            kwargs.pop("rpl_reduce_dtype", dtype_name(self.torch_dtype))
        )
        self.attention_dtype: torch.dtype = to_torch_dtype(
            kwargs.pop("attention_dtype", dtype_name(self.torch_dtype))
        )
        self.cast_type: str = kwargs.pop("cast_type", "config")

        # --- bucketing (ref config.py:185-213) ---
        self.enable_bucketing: bool = kwargs.pop("enable_bucketing", True)
        self.context_encoding_buckets: Optional[List[int]] = kwargs.pop(
            "context_encoding_buckets", None
        )
        self.token_generation_buckets: Optional[List[int]] = kwargs.pop(
            "token_generation_buckets", None
        )
        # decode BATCH sub-buckets: serving at partial occupancy replays
        # the smallest captured graph covering the active batch (ref 2-D
        # batch x seq buckets, autobucketing.py:203-223). None = auto
        # ladder; [] / enable_batch_bucketing=False = single full-batch
        # graph (the r01 behavior).
        self.token_generation_batch_buckets: Optional[List[int]] = kwargs.pop(
            "token_generation_batch_buckets", None
        )
        self.enable_batch_bucketing: bool = kwargs.pop("enable_batch_bucketing", True)
        # bucket selection strategy (ref pad_inputs strategies
        # model_wrapper.py:730-829): first_fit | second_fit | max
        self.bucket_strategy: str = kwargs.pop("bucket_strategy", "first_fit")
        self.bucket_n_active_tokens: bool = kwargs.pop("bucket_n_active_tokens", False)

        # --- parallelism degrees (ref config.py:360-390) ---
        self.tp_degree: int = kwargs.pop("tp_degree", 1)
        self.pp_degree: int = kwargs.pop("pp_degree", 1)
        self.ep_degree: int = kwargs.pop("ep_degree", 1)
        self.cp_degree: int = kwargs.pop("cp_degree", 1)
        self.mlp_cp_degree: int = kwargs.pop("mlp_cp_degree", 1)
        self.attention_dp_degree: int = kwargs.pop("attention_dp_degree", 1)
        self.sampling_dp_degree: int = kwargs.pop("sampling_dp_degree", 1)
        self.moe_tp_degree: Optional[int] = kwargs.pop("moe_tp_degree", None)
        self.moe_ep_degree: Optional[int] = kwargs.pop("moe_ep_degree", None)
        self.start_rank_id: int = kwargs.pop("start_rank_id", 0)
        self.local_ranks_size: Optional[int] = kwargs.pop("local_ranks_size", None)
        self.world_size: Optional[int] = kwargs.pop("world_size", None)
        self.vocab_parallel: bool = kwargs.pop("vocab_parallel", False)
        self.sequence_parallel_enabled: bool = kwargs.pop("sequence_parallel_enabled", False)

        # --- KV cache (ref kvcache/, config.py:276-317,395-415) ---
        self.kv_cache_batch_size: int = kwargs.pop(
            "kv_cache_batch_size", self.max_batch_size
        )
        self.kv_cache_padding_size: int = kwargs.pop("kv_cache_padding_size", 0)
        # compact sliding-window KV: windowed layers allocate only `window`
        # rows (ring buffer) instead of max_length (ref per-layer KV sizes)
        self.compact_swa_kv: bool = kwargs.pop("compact_swa_kv", False)
        self.is_block_kv_layout: bool = kwargs.pop("is_block_kv_layout", False)
        self.pa_num_blocks: Optional[int] = kwargs.pop("pa_num_blocks", None)
        self.pa_block_size: int = kwargs.pop("pa_block_size", 128)
        self.is_prefix_caching: bool = kwargs.pop("is_prefix_caching", False)
        self.is_chunked_prefill: bool = kwargs.pop("is_chunked_prefill", False)
        self.chunked_prefill_config: Optional[ChunkedPrefillConfig] = kwargs.pop(
            "chunked_prefill_config", None
        )
        if isinstance(self.chunked_prefill_config, dict):
            self.chunked_prefill_config = ChunkedPrefillConfig.from_dict(
                self.chunked_prefill_config
            )
        self.kv_cache_quant: bool = kwargs.pop("kv_cache_quant", False)
        tc = kwargs.pop("tensor_capture_config", None)
        if isinstance(tc, dict):
            tc = TensorCaptureConfig.from_dict(tc)
        self.tensor_capture_config: Optional[TensorCaptureConfig] = tc
        tr = kwargs.pop("tensor_replacement_config", None)
        if isinstance(tr, dict):
            tr = TensorReplacementConfig.from_dict(tr)
        self.tensor_replacement_config: Optional[TensorReplacementConfig] = tr
        # GPT-OSS MXFP4 MoE weights kept packed (e2m1 + e8m0), dequantized
        # in-kernel (ref models/gpt_oss/mx_layout_transform.py)
        self.moe_mxfp4: bool = kwargs.pop("moe_mxfp4", False)
        self.flash_decoding_enabled: bool = kwargs.pop("flash_decoding_enabled", False)
        self.num_cores_per_group: int = kwargs.pop("num_cores_per_group", 1)

        # --- sampling (ref modules/generation) ---
        odsc = kwargs.pop("on_device_sampling_config", None)
        if isinstance(odsc, dict):
            odsc = OnDeviceSamplingConfig.from_dict(odsc)
        self.on_device_sampling_config: Optional[OnDeviceSamplingConfig] = odsc
        self.output_logits: bool = kwargs.pop("output_logits", False)

        # --- speculation (ref config.py:242-274) ---
        self.speculation_length: int = kwargs.pop("speculation_length", 0)
        self.spec_batch_size: int = kwargs.pop("spec_batch_size", self.batch_size)
        self.enable_fused_speculation: bool = kwargs.pop("enable_fused_speculation", False)
        self.enable_eagle_speculation: bool = kwargs.pop("enable_eagle_speculation", False)
        self.medusa_speculation_length: int = kwargs.pop("medusa_speculation_length", 0)
        self.num_medusa_heads: int = kwargs.pop("num_medusa_heads", 0)
        self.is_medusa: bool = kwargs.pop("is_medusa", False)
        self.token_tree_config = kwargs.pop("token_tree_config", None)
        # EAGLE-3: target layer indices whose hiddens the draft consumes;
        # set before load() so hipGraph capture records the hidden writes
        self.eagle3_capture_layers = kwargs.pop("eagle3_capture_layers", None)

        # --- quantization (ref config.py:215-241) ---
        self.quantized: bool = kwargs.pop("quantized", False)
        self.quantization_dtype: str = kwargs.pop("quantization_dtype", "f8e4m3")
        self.quantization_type: str = kwargs.pop("quantization_type", "per_tensor_symmetric")
        self.quantized_mlp_kernel_enabled: bool = kwargs.pop(
            "quantized_mlp_kernel_enabled", False
        )

        # --- MI355X engine knobs (replace the ref's ~30 NKI kernel flags and
        # neuronx-cc compiler-flag assembly, model_wrapper.py:85-167) ---
        self.enable_hip_graphs: bool = kwargs.pop("enable_hip_graphs", True)
        self.attn_kernel_enabled: bool = kwargs.pop("attn_kernel_enabled", True)
        self.attn_tkg_kernel_enabled: bool = kwargs.pop("attn_tkg_kernel_enabled", True)
        self.rmsnorm_kernel_enabled: bool = kwargs.pop("rmsnorm_kernel_enabled", True)
        self.rope_kernel_enabled: bool = kwargs.pop("rope_kernel_enabled", True)
        self.mlp_kernel_enabled: bool = kwargs.pop("mlp_kernel_enabled", True)
        self.sampling_kernel_enabled: bool = kwargs.pop("sampling_kernel_enabled", True)
        self.kv_update_kernel_enabled: bool = kwargs.pop("kv_update_kernel_enabled", True)
        self.windowed_context_encoding_size: int = kwargs.pop(
            "windowed_context_encoding_size", 0
        )
        self.logical_nc_config: int = kwargs.pop("logical_nc_config", 1)  # accepted, unused
        self.async_mode: bool = kwargs.pop("async_mode", False)
        self.on_cpu: bool = kwargs.pop("on_cpu", False)
        self.skip_warmup: bool = kwargs.pop("skip_warmup", False)
        self.weights_to_skip_layout_optimization = kwargs.pop(
            "weights_to_skip_layout_optimization", []
        )
        self.compiler_args = kwargs.pop("compiler_args", None)  # accepted for CLI parity

        # --- attention structure ---
        self.fused_qkv: bool = kwargs.pop("fused_qkv", False)
        self.fused_gate_up: bool = kwargs.pop("fused_gate_up", False)
        self.qkv_kernel_enabled: bool = kwargs.pop("qkv_kernel_enabled", False)
        self.sliding_window: Optional[int] = kwargs.pop("sliding_window", None)

        # --- LoRA (ref modules/lora_serving) ---
        self.lora_config = kwargs.pop("lora_config", None)

        if kwargs:
            logger.warning("EngineConfig ignoring unknown kwargs: %s", sorted(kwargs))

        self._derive()

    # ------------------------------------------------------------------
    def _derive(self):
        if self.world_size is None:
            self.world_size = self.tp_degree * self.pp_degree * self.ep_degree
        if self.local_ranks_size is None:
            self.local_ranks_size = self.world_size
        if self.moe_tp_degree is None:
            self.moe_tp_degree = self.tp_degree // self.ep_degree if self.ep_degree else self.tp_degree
        if self.moe_ep_degree is None:
            self.moe_ep_degree = self.ep_degree
        if self.is_chunked_prefill and self.chunked_prefill_config is None:
            self.chunked_prefill_config = ChunkedPrefillConfig()
        # attention-DP shards the decode KV cache by batch (ref config.py:513-520)
        if self.attention_dp_degree > 1:
            if self.tkg_batch_size % self.attention_dp_degree != 0:
                raise ValueError(
                    f"tkg_batch_size {self.tkg_batch_size} not divisible by "
                    f"attention_dp_degree {self.attention_dp_degree}"
                )
            self.kv_cache_batch_size = self.tkg_batch_size // self.attention_dp_degree

    # ------------------------------------------------------------------
    @property
    def is_paged_attention(self) -> bool:
        return self.is_block_kv_layout

    @property
    def enable_speculation(self) -> bool:
        return self.speculation_length > 0

    def to_dict(self) -> Dict[str, Any]:
        out = {}
        for k, v in self.__dict__.items():
            if k.startswith("_"):
                continue
            if isinstance(v, torch.dtype):
                v = dtype_name(v)
            elif hasattr(v, "to_dict"):
                v = v.to_dict()
            out[k] = v
        return out

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "EngineConfig":
        return cls(**dict(d))

    def copy(self, **overrides) -> "EngineConfig":
        d = self.to_dict()
        d.update(overrides)
        return EngineConfig.from_dict(d)


class MoEEngineConfig(EngineConfig):
    """MoE additions (ref MoENeuronConfig config.py:798-846)."""

    def __init__(self, **kwargs):
        self.capacity_factor: Optional[float] = kwargs.pop("capacity_factor", None)
        self.glu_mlp: bool = kwargs.pop("glu_mlp", True)
        self.normalize_top_k_affinities: bool = kwargs.pop(
            "normalize_top_k_affinities", True
        )
        self.router_dtype: str = kwargs.pop("router_dtype", "float32")
        self.ep_dispatch_cc_option: str = kwargs.pop("ep_dispatch_cc_option", "all_to_all")
        super().__init__(**kwargs)


class InferenceConfig:
    """Model config + engine config bundle (ref InferenceConfig,
    config.py:850-1038). HF-style attributes live directly on this object;
    ``attribute_map`` aliases model-specific names (ref config.py:850-886)."""

    # subclasses may alias attrs, e.g. {"n_embd": "hidden_size"}
    attribute_map: Dict[str, str] = {}

    def __init__(self, neuron_config: Optional[EngineConfig] = None, metadata=None, **kwargs):
        self.neuron_config = neuron_config or EngineConfig()
        self.metadata = metadata or {}
        fused = kwargs.pop("fused_spec_config", None)
        if isinstance(fused, dict):
            fused = FusedSpecConfig.from_dict(fused)
        self.fused_spec_config: Optional[FusedSpecConfig] = fused
        for k, v in kwargs.items():
            setattr(self, k, v)
        self.add_derived_config()
        self.validate_config()

    # `engine_config` is the MI355X-native name; `neuron_config` kept for
    # reference-API parity.
    @property
    def engine_config(self) -> EngineConfig:
        return self.neuron_config

    def __getattr__(self, name):
        amap = object.__getattribute__(self, "__class__").attribute_map
        if name in amap:
            return getattr(self, amap[name])
        raise AttributeError(f"{type(self).__name__} has no attribute {name!r}")

    def __setattr__(self, name, value):
        amap = type(self).attribute_map
        if name in amap:
            name = amap[name]
        object.__setattr__(self, name, value)

    # ------------------------------------------------------------------
    def add_derived_config(self):
        """Hook for subclasses (ref LlamaInferenceConfig.add_derived_config,
        modeling_llama.py:262)."""

    def get_required_attributes(self) -> List[str]:
        return []

    def validate_config(self):
        missing = [a for a in self.get_required_attributes() if not hasattr(self, a)]
        if missing:
            raise ValueError(f"Config missing required attributes: {missing}")

    # ------------------------------------------------------------------
    def to_dict(self) -> Dict[str, Any]:
        out = {}
        for k, v in self.__dict__.items():
            if k == "neuron_config":
                out[k] = v.to_dict()
            elif k == "fused_spec_config":
                out[k] = v.to_dict() if v else None
            elif isinstance(v, torch.dtype):
                out[k] = dtype_name(v)
            elif isinstance(v, InferenceConfig):
                # nested config (multimodal text_config etc.): serialize as
                # a dict; the shared engine config is rebuilt on load
                nested = v.to_dict()
                nested.pop("neuron_config", None)
                nested.pop("_config_class", None)
                out[k] = nested
            else:
                try:
                    json.dumps(v)
                    out[k] = v
                except TypeError:
                    continue
        out["_config_class"] = f"{type(self).__module__}.{type(self).__name__}"
        return out

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "InferenceConfig":
        d = dict(d)
        d.pop("_config_class", None)
        ec = d.pop("neuron_config", None)
        moe = any(k in (ec or {}) for k in ("capacity_factor", "glu_mlp")) and (
            "num_local_experts" in d or "num_experts" in d or "ffn_config" in d
        )
        engine = (MoEEngineConfig if moe else EngineConfig).from_dict(ec) if ec else EngineConfig()
        return cls(neuron_config=engine, **d)

    def save(self, path: str):
        os.makedirs(path, exist_ok=True)
        with open(os.path.join(path, CONFIG_FILE), "w") as f:
            json.dump(self.to_dict(), f, indent=2, default=str)

    @classmethod
    def load(cls, path: str) -> "InferenceConfig":
        for name in (CONFIG_FILE, LEGACY_CONFIG_FILE):
            p = os.path.join(path, name)
            if os.path.exists(p):
                with open(p) as f:
                    d = json.load(f)
                klass = cls
                dotted = d.get("_config_class")
                if dotted and cls is InferenceConfig:
                    # reconstruct the concrete config class by dotted path
                    # (ref config.py:956-1038)
                    try:
                        import importlib

                        mod, _, nm = dotted.rpartition(".")
                        klass = getattr(importlib.import_module(mod), nm)
                    except Exception:
                        klass = cls
                return klass.from_dict(d)
        raise FileNotFoundError(f"no {CONFIG_FILE} or {LEGACY_CONFIG_FILE} under {path}")

    # HF-config ingestion -------------------------------------------------
    @classmethod
    def from_hf_config(
        cls, model_path: str, neuron_config: Optional[EngineConfig] = None, **overrides
    ) -> "InferenceConfig":
        """Build from a HuggingFace `config.json` directory."""
        cfg_path = os.path.join(model_path, "config.json")
        with open(cfg_path) as f:
            hf = json.load(f)
        hf.pop("torch_dtype", None)
        hf.update(overrides)
        return cls(neuron_config=neuron_config, **hf)
