"""Tensor-parallel layers.

MI355X-native analogs of NxD's ColumnParallelLinear / RowParallelLinear /
ParallelEmbedding (used throughout the reference, e.g.
`models/llama/modeling_llama.py:354-387,1112-1133`). GEMMs go through
torch F.linear (rocBLAS/hipBLASLt on ROCm); the collective seams are
`parallel.mappings` RCCL calls.

Sharding contract: every parallel parameter carries ``partition_dim`` (0 for
column-parallel output shards, 1 for row-parallel input shards) and
``partition_stride``; `utils.checkpoint.shard_state_dict` slices the full
checkpoint per rank from this metadata — the analog of the reference's
preshard hooks + `weights/tp{rank}_sharded_checkpoint.safetensors` layout
(`models/application_base.py:389-399`).
"""

from __future__ import annotations

from typing import Optional

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from neuronx_distributed_inference_amd.parallel import mappings, state as ps


def mark_parallel(param: torch.Tensor, partition_dim: int, stride: int = 1, num_partitions: Optional[int] = None):
    param.tensor_model_parallel = True
    param.partition_dim = partition_dim
    param.partition_stride = stride
    if num_partitions is not None:
        param.num_partitions = num_partitions
    return param


def divide(a: int, b: int) -> int:
    if a % b != 0:
        raise ValueError(f"{a} not divisible by tp degree {b}")
    return a // b


# ---------------------------------------------------------------------------
# FP8 (OCP e4m3) weight quantization — the reference's quantized-checkpoint
# path (ref application_base.py:746-799, quantized_mlp_kernel flags). Weights
# stored fp8 with a per-output-row scale; on GPU the GEMM runs through
# torch._scaled_mm (hipBLASLt fp8 MFMA) with dynamic per-tensor activation
# quantization; on CPU it dequantizes (the oracle path).

FP8_MAX = 448.0


def quantize_linear_fp8(mod: nn.Module):
    """In-place: weight -> fp8 + per-row scale; forward switches to fp8."""
    w = mod.weight.data
    scale = w.float().abs().amax(dim=1, keepdim=True).clamp(min=1e-8) / FP8_MAX
    q = (w.float() / scale).clamp(-FP8_MAX, FP8_MAX).to(torch.float8_e4m3fn)
    mod.weight = nn.Parameter(q, requires_grad=False)
    mod.register_buffer("weight_scale", scale.float())
    mod.fp8 = True


_FP8_ONES = {}


def _fp8_one(device):
    t = _FP8_ONES.get(device)
    if t is None:
        t = torch.ones((1, 1), device=device, dtype=torch.float32)
        _FP8_ONES[device] = t
    return t


_FP8_CHOICE = {}


def _fp8_choice(x, w_fp8, w_scale, bias, M, N, K) -> int:
    """0 = hipBLASLt scaled-mm; >0 = sk5f8 with that splitk. Timed once
    per shape (never inside a graph capture); the library keeps the
    shape unless sk5f8 wins by >15% (warm-cache rankings transfer with
    error — see the sg5 steady-state sweep)."""
    key = (M, N, K)
    c = _FP8_CHOICE.get(key)
    if c is not None:
        return c
    if torch.cuda.is_current_stream_capturing():
        return 0
    import neuronx_distributed_inference_amd.ops as _ops

    x2 = x.reshape(-1, K).contiguous()
    mb = (M + 127) // 128
    cands = [sk for sk in (1, 2, 4, 8)
             if K % (128 * sk) == 0 and (N // 64) * sk * mb <= 2048]

    # rotating COLD weight clones: warm-L2 timings flatter whichever
    # kernel re-reads weights (measured: sk5f8 "won" 70B shapes warm but
    # lost 10% in the real layer walk)
    nclones = 6
    clones = [w_fp8.clone() for _ in range(nclones)]

    def _time(fn, n=12):
        for _ in range(2):
            fn(clones[0])
        torch.cuda.synchronize()
        a = torch.cuda.Event(True)
        b = torch.cuda.Event(True)
        a.record()
        for i in range(n):
            fn(clones[(i + 1) % nclones])
        b.record()
        torch.cuda.synchronize()
        return a.elapsed_time(b)

    os.environ["NXDI_DISABLE_SK5F8"] = "1"
    try:
        t_lib = _time(lambda w: fp8_linear(x, w, w_scale, bias))
    finally:
        os.environ.pop("NXDI_DISABLE_SK5F8", None)
    best = 0
    t_best = t_lib
    for sk in cands:
        try:
            t = _time(lambda w, s=sk: _ops.skinny_gemm5_fp8(
                x2, w.view(torch.uint8), w_scale, bias=bias, splitk=s))
        except Exception:  # noqa: BLE001
            continue
        if t < t_best and t < 0.85 * t_lib:
            best, t_best = sk, t
    del clones
    _FP8_CHOICE[key] = best
    return best


def fp8_linear(x: torch.Tensor, w_fp8: torch.Tensor, w_scale: torch.Tensor, bias):
    """y = x @ dequant(w)^T (+bias). GPU: torch._scaled_mm with dynamic
    per-tensor activation scale; CPU: dequantized matmul."""
    if x.is_cuda:
        import neuronx_distributed_inference_amd.ops as _ops

        # decode-size batches: the hand-written fp8-weight skinny GEMM
        # (sk5_fp8_kernel: double-depth all-glds pipeline, bf16 x — no
        # activation-quant kernels, half the weight stream) vs hipBLASLt
        # fp8 scaled-mm, decided by a per-shape TIMED autotune. The
        # library stays unless sk5f8 clearly wins (M > 128 runs two
        # row-blocks whose W re-read usually cancels the fp8 saving;
        # M <= 128 — e.g. 70B single-GPU decode — is sk5f8 territory).
        M = x.numel() // x.shape[-1]
        N, K = w_fp8.shape
        # M >= 64: at very small M the isolated timing flatters sk5f8
        # (cold or warm) but the real layer walk prefers the library's
        # fp8 MFMA path — measured 70B b16: 729 lib vs 662 tok/s sk5f8
        if (64 <= M <= 256 and N % 64 == 0 and K % 128 == 0
                and x.dtype in (torch.bfloat16, torch.float16)
                and os.environ.get("NXDI_DISABLE_SK5F8") != "1"):
            choice = _fp8_choice(x, w_fp8, w_scale, bias, M, N, K)
            if choice > 0:
                y = _ops.skinny_gemm5_fp8(
                    x.reshape(-1, K).contiguous(),
                    w_fp8.view(torch.uint8), w_scale, bias=bias,
                    splitk=choice)
                return y.view(*x.shape[:-1], N)

        # hipBLASLt scaled-mm takes per-tensor scales on this stack; the
        # per-row weight scale rides in the HIP epilogue kernel. Dynamic
        # activation quant runs as two HIP kernels (amax atomic + cast) —
        # no fp32 materialization of x, no host sync (K6-analog,
        # ops/csrc/fp8_quant.hip).
        c = _ops._require_ext("fp8_quant")
        shape = x.shape[:-1] + (w_fp8.shape[0],)
        x2 = x.reshape(-1, x.shape[-1]).contiguous()
        x_q = torch.empty_like(x2, dtype=torch.float8_e4m3fn)
        scale = torch.empty(1, 1, dtype=torch.float32, device=x.device)
        amax = torch.empty(128, dtype=torch.float32, device=x.device)
        c.fp8_quant(x2, x_q, scale, amax)
        one = _fp8_one(x.device)
        y32 = torch._scaled_mm(
            x_q, w_fp8.t(), scale_a=scale, scale_b=one,
            bias=None, out_dtype=torch.float32,
        )
        out = torch.empty(shape, dtype=x.dtype, device=x.device)
        c.fp8_epilogue(out, y32, w_scale.reshape(-1).contiguous(),
                       None if bias is None else bias)
        return out
    w = w_fp8.float() * w_scale
    y = F.linear(x.float(), w)
    if bias is not None:
        y = y + bias.float()
    return y.to(x.dtype)


def quantize_model_fp8(model: nn.Module, skip_names=("lm_head",)):
    """Quantize every Column/RowParallelLinear except `skip_names`
    substrings (ref convert()/quantize flow, model_wrapper.py:1455-1627)."""
    n = 0
    for name, mod in model.named_modules():
        if isinstance(mod, (ColumnParallelLinear, RowParallelLinear)):
            if any(sk in name for sk in skip_names):
                continue
            mod.quantize_fp8()
            n += 1
        elif hasattr(mod, "quantize_fp8") and type(mod).__name__ == "ExpertMLPs":
            # stacked MoE expert weights: fp8 grouped kernel path
            if mod.quantize_fp8():
                n += 1
    return n


def _linear(x: torch.Tensor, weight: torch.Tensor,
            bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Library GEMM dispatch: autotuned hipBLASLt on GPU (bf16/f16),
    torch F.linear (rocBLAS) otherwise. See ops.lt_linear."""
    if x.is_cuda and x.dtype in (torch.bfloat16, torch.float16):
        import neuronx_distributed_inference_amd.ops as _ops

        if _ops.lt_linear_enabled():
            # decode-size batches route through the shape-autotuned
            # skinny_gemm5-vs-hipBLASLt dispatcher (ops.decode_linear)
            if x.numel() // x.shape[-1] <= 256:
                return _ops.decode_linear(x, weight, bias)
            return _ops.lt_linear(x, weight, bias)
    return F.linear(x, weight, bias)


class ColumnParallelLinear(nn.Module):
    """Y = X W^T with W sharded along the output dim across the TP group."""

    def __init__(
        self,
        input_size: int,
        output_size: int,
        bias: bool = False,
        gather_output: bool = False,
        dtype: torch.dtype = torch.float32,
        tp_degree: Optional[int] = None,
        group=None,
        stride: int = 1,
        sequence_parallel: bool = False,
    ):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        self.gather_output = gather_output
        # SP: input arrives sharded along the sequence dim (dim 1 of B,S,H);
        # all-gather it before the GEMM (ref C3, gather_from_sequence_parallel
        # region before QKV/MLP GEMMs, modeling_llama.py:677-679)
        self.sequence_parallel = sequence_parallel
        self._group = group
        self.tp_degree = tp_degree or (
            ps.get_tensor_model_parallel_size() if group is None else 1
        )
        if group is not None and not callable(group):
            import torch.distributed as dist

            self.tp_degree = dist.get_world_size(group=group) if dist.is_initialized() else 1
        self.output_size_per_partition = divide(output_size, self.tp_degree)
        self.weight = nn.Parameter(
            torch.empty(self.output_size_per_partition, input_size, dtype=dtype)
        )
        mark_parallel(self.weight, partition_dim=0, stride=stride, num_partitions=self.tp_degree)
        if bias:
            self.bias = nn.Parameter(torch.empty(self.output_size_per_partition, dtype=dtype))
            mark_parallel(self.bias, partition_dim=0, stride=stride, num_partitions=self.tp_degree)
        else:
            self.register_parameter("bias", None)

    @property
    def group(self):
        if callable(self._group):
            return self._group()
        return self._group if self._group is not None else ps.get_tensor_model_parallel_group()

    def quantize_fp8(self):
        quantize_linear_fp8(self)

    def forward(self, x: torch.Tensor, sequence_parallel: Optional[bool] = None) -> torch.Tensor:
        sp = self.sequence_parallel if sequence_parallel is None else sequence_parallel
        if sp:
            x = mappings.gather_along_dim(x, dim=1, group=self.group)
        if getattr(self, "fp8", False):
            y = fp8_linear(x, self.weight, self.weight_scale, self.bias)
        else:
            y = _linear(x, self.weight, self.bias)
        if self.gather_output:
            y = mappings.gather_from_tensor_parallel_region(y, dim=-1, group=self.group)
        return y


class RowParallelLinear(nn.Module):
    """Y = X W^T with W sharded along the input dim; output all-reduced
    (collective C1 — the decode-latency-critical all-reduce)."""

    def __init__(
        self,
        input_size: int,
        output_size: int,
        bias: bool = False,
        input_is_parallel: bool = True,
        dtype: torch.dtype = torch.float32,
        tp_degree: Optional[int] = None,
        group=None,
        reduce_output: bool = True,
        reduce_dtype: Optional[torch.dtype] = None,
    ):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        self.input_is_parallel = input_is_parallel
        self.reduce_output = reduce_output
        self.reduce_dtype = reduce_dtype
        self._group = group
        self.tp_degree = tp_degree or (
            ps.get_tensor_model_parallel_size() if group is None else 1
        )
        if group is not None and not callable(group):
            import torch.distributed as dist

            self.tp_degree = dist.get_world_size(group=group) if dist.is_initialized() else 1
        self.input_size_per_partition = divide(input_size, self.tp_degree)
        self.weight = nn.Parameter(
            torch.empty(output_size, self.input_size_per_partition, dtype=dtype)
        )
        mark_parallel(self.weight, partition_dim=1, num_partitions=self.tp_degree)
        if bias:
            # bias is NOT sharded; applied once after the reduce
            self.bias = nn.Parameter(torch.empty(output_size, dtype=dtype))
        else:
            self.register_parameter("bias", None)

    @property
    def group(self):
        if callable(self._group):
            return self._group()
        return self._group if self._group is not None else ps.get_tensor_model_parallel_group()

    def quantize_fp8(self):
        quantize_linear_fp8(self)

    def forward(self, x: torch.Tensor, sequence_parallel: bool = False) -> torch.Tensor:
        if not self.input_is_parallel:
            x = mappings.scatter_to_tensor_parallel_region(x, dim=-1, group=self.group)
        if getattr(self, "fp8", False):
            y = fp8_linear(x, self.weight, self.weight_scale, None)
        else:
            y = _linear(x, self.weight)
        if self.reduce_output:
            if sequence_parallel:
                # SP exit: reduce-scatter along S instead of all-reduce
                # (ref C2, attention_base.py:1350)
                y = mappings.reduce_scatter_along_dim(y, dim=1, group=self.group)
            else:
                y = mappings.reduce_from_tensor_parallel_region(y, group=self.group)
        if self.bias is not None:
            y = y + self.bias
        return y


class ParallelEmbedding(nn.Module):
    """Vocab-sharded embedding: each rank owns rows
    [rank*V/tp, (rank+1)*V/tp); out-of-shard ids hit a zero row and the
    partial sums are all-reduced (ref `modeling_llama.py:1112-1133`)."""

    def __init__(
        self,
        num_embeddings: int,
        embedding_dim: int,
        dtype: torch.dtype = torch.float32,
        tp_degree: Optional[int] = None,
        shard_along_embedding: bool = False,
        pad: bool = True,
    ):
        super().__init__()
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        self.tp_degree = tp_degree or ps.get_tensor_model_parallel_size()
        self.shard_along_embedding = shard_along_embedding
        if shard_along_embedding:
            per = divide(embedding_dim, self.tp_degree)
            self.weight = nn.Parameter(torch.empty(num_embeddings, per, dtype=dtype))
            mark_parallel(self.weight, partition_dim=1, num_partitions=self.tp_degree)
            self.vocab_start = 0
            self.vocab_size_per_partition = num_embeddings
        else:
            self.vocab_size_per_partition = divide(num_embeddings, self.tp_degree)
            self.weight = nn.Parameter(
                torch.empty(self.vocab_size_per_partition, embedding_dim, dtype=dtype)
            )
            mark_parallel(self.weight, partition_dim=0, num_partitions=self.tp_degree)
            self.vocab_start = None  # resolved at forward from the live rank

    def forward(self, ids: torch.Tensor, sequence_parallel: bool = False) -> torch.Tensor:
        if self.shard_along_embedding:
            y = F.embedding(ids, self.weight)
            return mappings.gather_from_tensor_parallel_region(y, dim=-1)
        tp = ps.get_tensor_model_parallel_size()
        if tp == 1:
            return F.embedding(ids, self.weight)
        rank = ps.get_tensor_model_parallel_rank()
        start = rank * self.vocab_size_per_partition
        in_shard = (ids >= start) & (ids < start + self.vocab_size_per_partition)
        local = torch.where(in_shard, ids - start, torch.zeros_like(ids))
        y = F.embedding(local, self.weight)
        y = y * in_shard.unsqueeze(-1).to(y.dtype)
        if sequence_parallel:
            # SP entry: partial sums reduce-scattered along S (ref C2,
            # model_base.py:1509-1514)
            return mappings.reduce_scatter_along_dim(y, dim=1)
        return mappings.reduce_from_tensor_parallel_region(y)
