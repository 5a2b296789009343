"""Multi-LoRA serving.

Analog of the reference's `modules/lora_serving/` (1897 LoC:
`wrap_model_with_lora` walks the model and swaps target linears for
MultiLora{Linear,ColumnParallelLinear,RowParallelLinear} holding stacked
(max_loras, r, ...) weights selected per request by `adapter_ids`,
lora_model.py:29-201, lora_layer.py:10-239; dynamic adapters keep a CPU
cache and swap shards into live weights, lora_model.py:263-682).

TP math:
- column-parallel target: LoRA A replicated, B sharded along out (the delta
  is born sharded; zero extra comm).
- row-parallel target: A sharded along in (partial r-activations), B
  replicated; the partial delta is added to the partial GEMM output BEFORE
  the TP all-reduce, which sums it correctly.

`adapter_ids` lives in a persistent device buffer (`LoraState`) filled per
request so captured decode graphs read the current assignment at replay.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from neuronx_distributed_inference_amd.parallel import mappings
from neuronx_distributed_inference_amd.parallel.layers import (
    ColumnParallelLinear,
    ParallelEmbedding,
    RowParallelLinear,
)

DEFAULT_TARGETS = ("q_proj", "k_proj", "v_proj", "o_proj", "qkv",
                   "gate_proj", "up_proj", "down_proj", "gate_up_proj")


@dataclass
class LoraServingConfig:
    max_loras: int = 4
    max_lora_rank: int = 16
    lora_alpha: float = 16.0
    target_modules: tuple = DEFAULT_TARGETS
    lora_ckpt_paths: Dict[str, str] = field(default_factory=dict)

    def to_dict(self):
        return {
            "max_loras": self.max_loras, "max_lora_rank": self.max_lora_rank,
            "lora_alpha": self.lora_alpha,
            "target_modules": list(self.target_modules),
            "lora_ckpt_paths": dict(self.lora_ckpt_paths),
        }

    @classmethod
    def from_dict(cls, d):
        d = dict(d)
        d["target_modules"] = tuple(d.get("target_modules", DEFAULT_TARGETS))
        return cls(**{k: v for k, v in d.items() if k in cls.__dataclass_fields__})


class LoraState:
    """Per-model adapter assignment; buffer content is read inside captured
    graphs so replays see the latest ids."""

    def __init__(self, max_batch: int, device=None):
        self.adapter_ids = torch.zeros(max_batch, dtype=torch.long, device=device)

    def set(self, ids: torch.Tensor):
        n = ids.shape[0]
        self.adapter_ids[:n].copy_(ids.to(self.adapter_ids.device))

    def to_device(self, device):
        self.adapter_ids = self.adapter_ids.to(device)
        return self


class MultiLoraLinear(nn.Module):
    """Wraps a (parallel) linear with stacked per-adapter low-rank deltas."""

    def __init__(self, base: nn.Module, cfg: LoraServingConfig, state: LoraState,
                 shard_mode: str):
        super().__init__()
        self.base = base
        self.cfg = cfg
        self.state = state
        self.shard_mode = shard_mode  # "column" | "row"
        r = cfg.max_lora_rank
        dtype = base.weight.dtype
        out_local, in_local = base.weight.shape
        if shard_mode == "column":
            # A: (L, r, in_full) replicated; B: (L, out_local, r) sharded
            self.lora_A = nn.Parameter(
                torch.zeros(cfg.max_loras, r, in_local, dtype=dtype), requires_grad=False
            )
            self.lora_B = nn.Parameter(
                torch.zeros(cfg.max_loras, out_local, r, dtype=dtype), requires_grad=False
            )
        else:
            # A: (L, r, in_local) sharded along in; B: (L, out_full, r) replicated
            self.lora_A = nn.Parameter(
                torch.zeros(cfg.max_loras, r, in_local, dtype=dtype), requires_grad=False
            )
            self.lora_B = nn.Parameter(
                torch.zeros(cfg.max_loras, out_local, r, dtype=dtype), requires_grad=False
            )
        self.scaling = cfg.lora_alpha / max(r, 1)

    def _delta(self, x: torch.Tensor) -> torch.Tensor:
        ids = self.state.adapter_ids[: x.shape[0]]
        A = self.lora_A[ids]          # (B, r, in)
        Bm = self.lora_B[ids]         # (B, out, r)
        h = torch.einsum("bsi,bri->bsr", x.float(), A.float())
        d = torch.einsum("bsr,bor->bso", h, Bm.float())
        return (d * self.scaling).to(x.dtype)

    def forward(self, x: torch.Tensor, sequence_parallel: bool = False) -> torch.Tensor:
        if self.shard_mode == "column":
            if sequence_parallel:
                x = mappings.gather_along_dim(x, dim=1, group=self.base.group)
            y = F.linear(x, self.base.weight, self.base.bias)
            y = y + self._delta(x)
            if self.base.gather_output:
                y = mappings.gather_from_tensor_parallel_region(y, dim=-1, group=self.base.group)
            return y
        # row-parallel: add the partial delta before the reduce
        y = F.linear(x, self.base.weight)
        y = y + self._delta(x)
        if self.base.reduce_output:
            if sequence_parallel:
                y = mappings.reduce_scatter_along_dim(y, dim=1, group=self.base.group)
            else:
                y = mappings.reduce_from_tensor_parallel_region(y, group=self.base.group)
        if self.base.bias is not None:
            y = y + self.base.bias
        return y

    # ------------------------------------------------------------------
    def load_adapter(self, slot: int, a: torch.Tensor, b: torch.Tensor,
                     tp_rank: int = 0, tp_degree: int = 1):
        """Install full (r, in_full)/(out_full, r) A/B into `slot`, slicing
        for this rank per shard_mode. Ranks < max_lora_rank are zero-padded."""
        r = a.shape[0]
        with torch.no_grad():
            self.lora_A[slot].zero_()
            self.lora_B[slot].zero_()
            if self.shard_mode == "column":
                out_local = self.lora_B.shape[1]
                b_sh = b[tp_rank * out_local:(tp_rank + 1) * out_local]
                self.lora_A[slot, :r].copy_(a.to(self.lora_A.dtype))
                self.lora_B[slot, :, :r].copy_(b_sh.to(self.lora_B.dtype))
            else:
                in_local = self.lora_A.shape[2]
                a_sh = a[:, tp_rank * in_local:(tp_rank + 1) * in_local]
                self.lora_A[slot, :r].copy_(a_sh.to(self.lora_A.dtype))
                self.lora_B[slot, :, :r].copy_(b.to(self.lora_B.dtype))




class MultiLoraEmbedding(nn.Module):
    """Embedding with stacked per-adapter low-rank deltas (ref
    MultiLoraEmbedding, lora_layer.py:10-239): delta = (one_hot(ids) @ A^T)
    @ B^T computed as A-row gather + B projection. Wraps ParallelEmbedding
    (vocab-sharded base; LoRA factors replicated — adapter deltas are
    dense over the hidden dim, so the delta adds AFTER the base's
    all-reduce)."""

    def __init__(self, base: nn.Module, cfg: LoraServingConfig, state: LoraState):
        super().__init__()
        self.base = base
        self.cfg = cfg
        self.state = state
        r = cfg.max_lora_rank
        vocab, hidden = base.num_embeddings, base.embedding_dim
        dtype = base.weight.dtype
        # A: (L, vocab, r) gathered by token id; B: (L, hidden, r)
        self.lora_A = nn.Parameter(
            torch.zeros(cfg.max_loras, vocab, r, dtype=dtype), requires_grad=False)
        self.lora_B = nn.Parameter(
            torch.zeros(cfg.max_loras, hidden, r, dtype=dtype), requires_grad=False)
        self.scaling = cfg.lora_alpha / max(r, 1)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        y = self.base(input_ids)
        ids = self.state.adapter_ids[: input_ids.shape[0]]
        A = self.lora_A[ids]                       # (B, V, r)
        h = A.gather(1, input_ids.unsqueeze(-1).expand(
            -1, -1, A.shape[-1]))                  # (B, S, r)
        Bm = self.lora_B[ids]                      # (B, H, r)
        d = torch.einsum("bsr,bhr->bsh", h.float(), Bm.float())
        return y + (d * self.scaling).to(y.dtype)

    def load_adapter(self, slot: int, a: torch.Tensor, b: torch.Tensor,
                     tp_rank: int = 0, tp_degree: int = 1):
        """a: (r, vocab) (HF embedding lora_A is (r, num_embeddings)),
        b: (hidden, r)."""
        r = a.shape[0]
        with torch.no_grad():
            self.lora_A[slot].zero_()
            self.lora_B[slot].zero_()
            self.lora_A[slot, :, :r].copy_(a.t().to(self.lora_A.dtype))
            self.lora_B[slot, :, :r].copy_(b.to(self.lora_B.dtype))


class MultiLoraConv2d(nn.Module):
    """Conv2d with stacked per-adapter low-rank deltas (vision towers;
    ref MultiLoraConv2d, lora_layer.py:10-239). delta(x) = conv(x, B) of
    conv(x, A) where A: (r, Cin, kh, kw) with the base stride/padding and
    B: (Cout, r, 1, 1) — grouped per request via a batched loop over the
    distinct adapters in the batch (vision prefill is once per request)."""

    def __init__(self, base: nn.Conv2d, cfg: LoraServingConfig, state: LoraState):
        super().__init__()
        self.base = base
        self.cfg = cfg
        self.state = state
        r = cfg.max_lora_rank
        dtype = base.weight.dtype
        self.lora_A = nn.Parameter(
            torch.zeros(cfg.max_loras, r, base.in_channels,
                        *base.kernel_size, dtype=dtype), requires_grad=False)
        self.lora_B = nn.Parameter(
            torch.zeros(cfg.max_loras, base.out_channels, r, 1, 1, dtype=dtype),
            requires_grad=False)
        self.scaling = cfg.lora_alpha / max(r, 1)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = self.base(x)
        ids = self.state.adapter_ids[: x.shape[0]]
        for slot in ids.unique().tolist():
            rows = ids == slot
            h = F.conv2d(x[rows], self.lora_A[slot], stride=self.base.stride,
                         padding=self.base.padding, dilation=self.base.dilation)
            d = F.conv2d(h, self.lora_B[slot])
            y[rows] = y[rows] + (d * self.scaling).to(y.dtype)
        return y

    def load_adapter(self, slot: int, a: torch.Tensor, b: torch.Tensor,
                     tp_rank: int = 0, tp_degree: int = 1):
        r = a.shape[0]
        with torch.no_grad():
            self.lora_A[slot].zero_()
            self.lora_B[slot].zero_()
            self.lora_A[slot, :r].copy_(a.to(self.lora_A.dtype))
            self.lora_B[slot, :, :r].copy_(b.view(*b.shape[:2], 1, 1)
                                           .to(self.lora_B.dtype))


def wrap_model_with_lora(model: nn.Module, cfg: LoraServingConfig,
                         max_batch: int) -> LoraState:
    """Swap every target (parallel) linear for a MultiLoraLinear
    (ref wrap_model_with_lora, lora_model.py:29-201)."""
    state = LoraState(max_batch, device=next(model.parameters()).device)
    for name, module in list(model.named_modules()):
        for child_name, child in list(module.named_children()):
            if child_name not in cfg.target_modules:
                continue
            if isinstance(child, ColumnParallelLinear):
                setattr(module, child_name, MultiLoraLinear(child, cfg, state, "column"))
            elif isinstance(child, RowParallelLinear):
                setattr(module, child_name, MultiLoraLinear(child, cfg, state, "row"))
            elif isinstance(child, ParallelEmbedding):
                setattr(module, child_name, MultiLoraEmbedding(child, cfg, state))
            elif isinstance(child, nn.Conv2d):
                setattr(module, child_name, MultiLoraConv2d(child, cfg, state))
    return state


def lora_layers(model: nn.Module) -> Dict[str, nn.Module]:
    return {n: m for n, m in model.named_modules()
            if isinstance(m, (MultiLoraLinear, MultiLoraEmbedding, MultiLoraConv2d))}


class AdapterCache:
    """Dynamic multi-LoRA: CPU-resident adapters with LRU slot assignment;
    requested adapters are swapped into live device slots between requests
    (ref dynamic multi-LoRA, lora_model.py:263-682,
    application_base.py:401-419)."""

    def __init__(self, model: nn.Module, cfg: LoraServingConfig, state: LoraState,
                 tp_rank: int = 0, tp_degree: int = 1):
        self.layers = lora_layers(model)
        self.cfg = cfg
        self.state = state
        self.tp_rank = tp_rank
        self.tp_degree = tp_degree
        self.cpu_store: Dict[str, Dict[str, tuple]] = {}
        # slot 0 is the permanent zero adapter ("no LoRA")
        self.slot_of: Dict[str, int] = {}
        self.lru: List[str] = []
        self.free_slots = list(range(1, cfg.max_loras))

    def register_adapter(self, name: str, weights: Dict[str, tuple]):
        """weights: {layer_name: (A (r,in_full), B (out_full,r))} on CPU."""
        self.cpu_store[name] = weights

    def load_adapter_dir(self, name: str, path: str):
        """Read a PEFT-style safetensors adapter: keys
        `...<target>.lora_A.weight` / `...<target>.lora_B.weight`."""
        from safetensors.torch import load_file
        import glob as _glob
        import os as _os

        files = _glob.glob(_os.path.join(path, "*.safetensors"))
        sd = {}
        for f in files:
            sd.update(load_file(f))
        weights: Dict[str, tuple] = {}
        for k, v in sd.items():
            if ".lora_A." in k:
                base = k.split(".lora_A.")[0]
                bkey = k.replace(".lora_A.", ".lora_B.")
                if bkey in sd:
                    weights[base] = (v, sd[bkey])
        self.register_adapter(name, weights)

    def _match_layer(self, adapter_layer_name: str):
        for lname, layer in self.layers.items():
            if adapter_layer_name.endswith(lname) or lname.endswith(adapter_layer_name) \
                    or adapter_layer_name.split(".")[-1] == lname.split(".")[-2]:
                yield lname, layer

    def ensure_loaded(self, names: List[str]) -> Dict[str, int]:
        """Make every named adapter resident; returns name -> slot. Evicts
        least-recently-used residents when slots run out."""
        out = {}
        for name in names:
            if name in self.slot_of:
                self.lru.remove(name)
                self.lru.append(name)
                out[name] = self.slot_of[name]
                continue
            if name not in self.cpu_store:
                raise KeyError(f"unknown adapter {name!r}")
            if self.free_slots:
                slot = self.free_slots.pop(0)
            else:
                evict = self.lru.pop(0)
                slot = self.slot_of.pop(evict)
            self._install(name, slot)
            self.slot_of[name] = slot
            self.lru.append(name)
            out[name] = slot
        return out

    def _install(self, name: str, slot: int):
        weights = self.cpu_store[name]
        for lname, layer in self.layers.items():
            hit = None
            for aname, ab in weights.items():
                if lname.endswith(aname) or aname.split(".")[-1] == lname.split(".")[-2] \
                        or aname.split(".")[-1] == lname.split(".")[-1]:
                    hit = ab
                    break
            with torch.no_grad():
                if hit is None:
                    layer.lora_A[slot].zero_()
                    layer.lora_B[slot].zero_()
                else:
                    a, b = hit
                    layer.load_adapter(slot, a, b, self.tp_rank, self.tp_degree)

    def set_request_adapters(self, names: List[str]):
        """Map a batch's adapter names ('' = none) to slots and install the
        per-row adapter_ids buffer."""
        uniq = [n for n in dict.fromkeys(names) if n]
        mapping = self.ensure_loaded(uniq)
        ids = torch.tensor([mapping.get(n, 0) for n in names], dtype=torch.long)
        self.state.set(ids)
        return ids
