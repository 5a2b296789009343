"""Device-op dispatch layer.

Every hot op has (a) a hand-written CDNA4 HIP kernel in ``csrc/`` compiled
in-tree to ``_nxdi_C*.so`` for gfx950, and (b) a pure-torch reference in
``ops.ref`` used on CPU and as the numerics oracle.

Dispatch policy: on a CUDA (ROCm) tensor the HIP extension is REQUIRED —
if it failed to import we raise instead of silently falling back to eager
torch, so a GPU run can never accidentally bench the fallback. CPU tensors
always use the reference path.
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch

from neuronx_distributed_inference_amd.ops import ref

_C = None
_C_IMPORT_ERROR: Optional[BaseException] = None

try:  # the in-tree built extension travels with the repo snapshot
    from neuronx_distributed_inference_amd.ops import _nxdi_C as _C  # type: ignore
except Exception as e:  # noqa: BLE001
    _C_IMPORT_ERROR = e


def extension_available() -> bool:
    return _C is not None


def _require_ext(op: str):
    if _C is None:
        raise RuntimeError(
            f"HIP op '{op}' called on a GPU tensor but the _nxdi_C extension is "
            f"not built/importable (import error: {_C_IMPORT_ERROR}). Build it "
            f"with `python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950)."
        )
    return _C


# ----------------------------------------------------------------------
def rms_norm(x, weight, eps: float = 1e-6, residual=None):
    if x.is_cuda:
        c = _require_ext("rms_norm")
        if residual is not None:
            res = residual.clone() if not residual.is_contiguous() else residual.contiguous()
            out = torch.empty_like(x)
            c.rms_norm_fused_add(out, x.contiguous(), res, weight, eps)
            return out, res
        out = torch.empty_like(x)
        c.rms_norm(out, x.contiguous(), weight, eps)
        return out, None
    return ref.rms_norm(x, weight, eps, residual)


def rope_apply(q, k, cos, sin):
    if q.is_cuda:
        c = _require_ext("rope_apply")
        q = q.contiguous()
        k = k.contiguous()
        c.rope_apply(q, k, cos.contiguous(), sin.contiguous())
        return q, k
    return ref.rope_apply(q, k, cos, sin)


def silu_mul(gate, up):
    if gate.is_cuda:
        c = _require_ext("silu_mul")
        out = torch.empty_like(gate)
        c.silu_mul(out, gate.contiguous(), up.contiguous())
        return out
    return ref.silu_mul(gate, up)


def gelu_tanh_mul(gate, up):
    if gate.is_cuda:
        c = _require_ext("gelu_tanh_mul")
        out = torch.empty_like(gate)
        c.gelu_tanh_mul(out, gate.contiguous(), up.contiguous())
        return out
    return ref.gelu_tanh_mul(gate, up)


def kv_cache_update(k_cache, v_cache, k_new, v_new, positions, seq_ids=None):
    if k_cache.is_cuda:
        c = _require_ext("kv_cache_update")
        B = k_new.shape[0]
        if seq_ids is None:
            seq_ids = torch.arange(B, device=k_new.device, dtype=torch.int32)
        c.kv_cache_update(
            k_cache, v_cache, k_new.contiguous(), v_new.contiguous(),
            positions.to(torch.int32).contiguous(), seq_ids.to(torch.int32).contiguous(),
        )
        return
    ref.kv_cache_update(k_cache, v_cache, k_new, v_new, positions, seq_ids)


def block_kv_cache_update(k_cache, v_cache, k_new, v_new, slot_mapping):
    if k_cache.is_cuda:
        c = _require_ext("block_kv_cache_update")
        c.block_kv_cache_update(
            k_cache, v_cache, k_new.contiguous(), v_new.contiguous(),
            slot_mapping.to(torch.int32).contiguous(),
        )
        return
    ref.block_kv_cache_update(k_cache, v_cache, k_new, v_new, slot_mapping)


def attention_prefill(q, k, v, mask=None, scale=None, sinks=None, causal=True,
                      sliding_window: int = 0):
    """Flash-attention forward for context encoding.

    q: (B,Hq,Sq,D); k,v: (B,Hkv,Skv,D). On GPU: hand-written CDNA4 MFMA
    flash kernel (K1/K11 analog). CPU: fp32 reference."""
    if q.is_cuda:
        c = _require_ext("attention_prefill")
        import math
        scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
        if (
            hasattr(c, "flash_prefill")
            and mask is None and causal
            and q.shape[-1] in (64, 128) and q.shape[2] == k.shape[2]
        ):
            out = torch.empty_like(q)
            c.flash_prefill(out, q.contiguous(), k.contiguous(), v.contiguous(),
                            float(scale), int(sliding_window),
                            sinks=None if sinks is None else sinks.detach())
            return out
        # masked / unusual-shape path: matmul-softmax via library GEMMs
        if mask is None and causal:
            from neuronx_distributed_inference_amd.modules.attention.utils import (
                causal_mask,
                chunked_attention_mask,
                sliding_window_mask,
            )
            Sq, Skv = q.shape[2], k.shape[2]
            mask = causal_mask(Sq, Skv, device=q.device)
            if sliding_window > 0:
                mask = mask & sliding_window_mask(Sq, Skv, sliding_window, device=q.device)
            elif sliding_window < 0:
                mask = mask & chunked_attention_mask(Sq, Skv, -sliding_window, device=q.device)
        return ref.sdpa(q, k, v, mask=mask, scale=scale, sinks=sinks)
    if mask is None and causal:
        Sq, Skv = q.shape[2], k.shape[2]
        m = torch.ones(Sq, Skv, dtype=torch.bool, device=q.device).tril(Skv - Sq)
        if sliding_window < 0:
            from neuronx_distributed_inference_amd.modules.attention.utils import (
                chunked_attention_mask,
            )
            m = m & chunked_attention_mask(Sq, Skv, -sliding_window).view(Sq, Skv)
            sliding_window = 0
        if sliding_window:
            m = m & ~m.cumsum(-1).le(0)  # placeholder; window applied below
            m = torch.ones(Sq, Skv, dtype=torch.bool).tril(Skv - Sq)
            win = torch.ones(Sq, Skv, dtype=torch.bool).triu(Skv - Sq - sliding_window + 1)
            m = m & win
        mask = m.view(1, 1, Sq, Skv)
    return ref.sdpa(q, k, v, mask=mask, scale=scale, sinks=sinks)


_lt_state = {"enabled": None, "bad_shapes": set()}


def lt_linear_enabled() -> bool:
    """True when the autotuned hipBLASLt GEMM path is usable here.

    Disable with NXDI_DISABLE_HIPBLASLT=1 (falls back to torch F.linear's
    rocBLAS dispatch)."""
    if _lt_state["enabled"] is None:
        if os.environ.get("NXDI_DISABLE_HIPBLASLT") == "1" or not torch.cuda.is_available():
            _lt_state["enabled"] = False
        else:
            try:
                c = _require_ext("lt_linear")
                _lt_state["enabled"] = bool(c.lt_available())
            except Exception:
                _lt_state["enabled"] = False
    return _lt_state["enabled"]


def lt_linear(x: torch.Tensor, weight: torch.Tensor,
              bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """y = x @ weight^T + bias via hipBLASLt with a per-shape TIMED
    autotune on first sight (the analog of the reference compiler's AOT
    GEMM tile selection). Replay is a plain stream enqueue — hipGraph-safe.
    Shapes hipblaslt can't serve fall back to F.linear permanently."""
    shape = x.shape
    x2 = x.reshape(-1, shape[-1])
    key = (x2.shape[0], weight.shape[0], weight.shape[1], bias is not None)
    if key in _lt_state["bad_shapes"]:
        return torch.nn.functional.linear(x, weight, bias)
    c = _require_ext("lt_linear")
    y = torch.empty(x2.shape[0], weight.shape[0], dtype=x.dtype, device=x.device)
    try:
        c.lt_linear(y, x2.contiguous(), weight,
                    None if bias is None else bias.contiguous())
    except RuntimeError:
        _lt_state["bad_shapes"].add(key)
        return torch.nn.functional.linear(x, weight, bias)
    return y.view(*shape[:-1], weight.shape[0])


# ----------------------------------------------------------------------
# Decode-GEMM dispatch: skinny_gemm5 (hand-written CDNA4 all-glds kernel,
# ops/csrc/skinny_gemm5.hip) vs the autotuned hipBLASLt path, TIMED per
# exact (M, N, K, bias, dtype) shape at first sight and the winner
# replayed (graph-safe: replay is a plain stream enqueue). Covers the
# reference's fused decode projections.
_sg5_state = {"choice": {}, "ws": {}}


def _sg5_workspace(rows: int, N: int, splitk: int, device) -> torch.Tensor:
    key = (rows, N, splitk, device)
    ws = _sg5_state["ws"].get(key)
    if ws is None:
        ws = torch.empty(splitk, rows, N, dtype=torch.float32, device=device)
        _sg5_state["ws"][key] = ws
    return ws


def skinny_gemm5(x2: torch.Tensor, weight: torch.Tensor, bias=None, rs=None,
                 splitk: int = 1, mode: int = 0, mt: int = 0) -> torch.Tensor:
    """y = x2 @ weight^T (+bias), optionally scaled per-row by fp32 rs
    (the rmsnorm fold: y[m] = rs[m] * (x[m] @ W^T)). GPU only.

    `mt`: row-block tile (0 = one block over all rows; 64/128 split M
    across blocks for >1 block/CU co-residency)."""
    c = _require_ext("skinny_gemm5")
    M, K = x2.shape
    N = weight.shape[0]
    y = torch.empty(M, N, dtype=x2.dtype, device=x2.device)
    part = None
    if splitk > 1:
        MT = mt if mt else (64 if M <= 64 else (128 if M <= 128 else 256))
        rows = ((M + MT - 1) // MT) * MT
        part = _sg5_workspace(rows, N, splitk, x2.device)
    c.skinny_gemm5(y, x2, weight, None if bias is None else bias.contiguous(),
                   rs, part, splitk, mode, mt)
    return y


_SK5F8_WS = {}


def skinny_gemm5_fp8(x2: torch.Tensor, wq_u8: torch.Tensor,
                     wscale: torch.Tensor, bias=None,
                     splitk: int = 1) -> torch.Tensor:
    """y = x2 @ dequant(wq)^T (+bias): fp8-e4m3 weights (N, K) u8 with a
    per-channel fp32 scale, decode batches (M <= 256). Double-depth
    (BK=128) all-glds pipeline — see sk5_fp8_kernel."""
    c = _require_ext("skinny_gemm5_fp8")
    M, K = x2.shape
    N = wq_u8.shape[0]
    y = torch.empty(M, N, dtype=x2.dtype, device=x2.device)
    part = None
    if splitk > 1:
        rows = ((M + 127) // 128) * 128
        key = (rows, N, splitk, x2.device)
        part = _SK5F8_WS.get(key)
        if part is None or part.numel() < splitk * rows * N:
            part = torch.empty(splitk, rows, N, dtype=torch.float32,
                               device=x2.device)
            _SK5F8_WS[key] = part
    c.skinny_gemm5_fp8(y, x2, wq_u8,
                       wscale.reshape(-1).contiguous().float(),
                       None if bias is None else bias.contiguous(),
                       part, splitk)
    return y


def _sg5_candidates(M: int, N: int, K: int):
    """(mt, splitk) variants: splitk sized so total blocks land near
    0.5-2x the 256 CUs (guide: per-block latency ~constant in N); row
    sub-tiles (mt < M) trade extra W traffic for >1 block/CU
    co-residency (block-level DMA/compute overlap)."""
    cands = []
    mts = [0]
    if M > 128:
        mts.append(128)
    if M > 64:
        mts.append(64)
    for mt in mts:
        mb = 1 if mt == 0 else (M + mt - 1) // mt
        for sk in (1, 2, 4, 8):
            if K % (64 * sk):
                continue
            blocks = (N // 64) * sk * mb
            if blocks > 560 or (sk > 1 and blocks > 520):
                continue
            cands.append((mt, sk))
    return cands


def _sg5_autotune(key, x2, weight, bias) -> int:
    """Returns 0 (= lt/library path) or the winning (mt, splitk). Timed
    over ROTATING weight clones so the stream is cache-cold like the real
    decode loop (the 256 MB L3 would otherwise hold the whole weight and
    flip the verdict); x stays hot, which is also realistic."""
    M, N, K = key[0], key[1], key[2]
    n_iter = 12
    # rotate enough copies that successive calls miss L3 (256 MB)
    wbytes = weight.numel() * weight.element_size()
    ncopy = max(2, min(8, int(320e6 // max(wbytes, 1)) + 1))
    try:
        wcopies = [weight] + [weight.clone() for _ in range(ncopy - 1)]
    except torch.cuda.OutOfMemoryError:  # pragma: no cover
        wcopies = [weight]
    results = {}
    torch.cuda.synchronize()

    def _time(fn):
        fn(wcopies[0])  # warm (triggers lt's own tune / LDS attr set)
        torch.cuda.synchronize()
        t0 = torch.cuda.Event(enable_timing=True)
        t1 = torch.cuda.Event(enable_timing=True)
        t0.record()
        for i in range(n_iter):
            fn(wcopies[i % len(wcopies)])
        t1.record()
        torch.cuda.synchronize()
        return t0.elapsed_time(t1) / n_iter

    try:
        results[0] = _time(lambda w: lt_linear(x2, w, bias))
    except Exception:  # noqa: BLE001
        results[0] = float("inf")
    for mt, sk in _sg5_candidates(M, N, K):
        try:
            results[(mt, sk)] = _time(
                lambda w, s=sk, m=mt: skinny_gemm5(x2, w, bias, splitk=s, mt=m))
        except Exception:  # noqa: BLE001
            results[(mt, sk)] = float("inf")
    del wcopies
    choice = min(results, key=results.get)
    # microbench rankings transfer to in-graph steady state with ~±20%
    # error (DVFS + cache state differ): ship the hand-written kernel
    # only on a CLEAR win, else keep the library path (measured: near
    # parity flips either way run to run)
    if choice != 0 and results[choice] > 0.95 * results.get(0, float("inf")):
        choice = 0
    _sg5_state["choice"][key] = choice
    return choice


def decode_linear(x: torch.Tensor, weight: torch.Tensor, bias=None) -> torch.Tensor:
    """Shape-autotuned decode projection: hand-written skinny_gemm5 vs
    hipBLASLt, winner replayed per shape. Falls through to lt_linear for
    shapes sg5 does not serve."""
    shape = x.shape
    x2 = x.reshape(-1, shape[-1])
    M, K = x2.shape
    N = weight.shape[0]
    eligible = (
        8 <= M <= 256 and N % 64 == 0 and K % 64 == 0
        and x.dtype in (torch.bfloat16, torch.float16)
        and os.environ.get("NXDI_DISABLE_SG5") != "1"
    )
    if not eligible:
        return lt_linear(x, weight, bias)
    key = (M, N, K, bias is not None, x.dtype)
    choice = _sg5_state["choice"].get(key)
    if choice is None:
        if torch.cuda.is_current_stream_capturing():
            return lt_linear(x, weight, bias)  # never tune inside a capture
        choice = _sg5_autotune(key, x2.contiguous(), weight, bias)
    if choice == 0:
        return lt_linear(x, weight, bias)
    mt, sk = choice
    y = skinny_gemm5(x2.contiguous(), weight, bias, splitk=sk, mt=mt)
    return y.view(*shape[:-1], N)


def attention_prefill_cached(q, k_cache, v_cache, seq_ids, scale=None,
                             sinks=None, sliding_window: int = 0):
    """Context-encoding attention reading K/V straight from the cache rows
    that `qkv_rope_update` just wrote (no fresh K/V tensors, no copies).

    q: (B,Hq,S,D); k_cache/v_cache: (Bc,Hkv,Smax,D); seq_ids: (B,) cache
    rows. Causal by *row index* — valid for right-padded batches because
    pad positions (-1) were skipped by the cache scatter, so cache rows
    0..i are always the real prefix for any real query at row i. Fuses the
    reference's CTE attention + cache update (SURVEY §2 K1+K5)."""
    import math
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        c = _require_ext("attention_prefill_cached")
        out = torch.empty_like(q)
        c.flash_prefill_cached(out, q.contiguous(), k_cache, v_cache,
                               seq_ids, float(scale), int(sliding_window),
                               sinks=None if sinks is None else sinks.detach())
        return out
    B, Hq, S, D = q.shape
    rows = seq_ids.long().clamp_min(0)
    k = k_cache[rows, :, :S].to(q.dtype)
    v = v_cache[rows, :, :S].to(q.dtype)
    return attention_prefill(q, k, v, scale=scale, sinks=sinks,
                             causal=True, sliding_window=sliding_window)


def attention_decode(q, k_cache, v_cache, seq_lens, seq_ids=None, scale=None,
                     sinks=None, sliding_window: int = 0, ring: bool = False):
    """Token-generation attention against the linear KV cache.

    q: (B,Hq,Sq,D) with small Sq (1 or spec_len); k_cache/v_cache:
    (B_cache,Hkv,Smax,D). `seq_lens` is either (B,) valid lengths (queries
    assumed to be the last Sq positions) or (B,Sq) per-query absolute
    POSITIONS (ragged chunks use -1 for pads -> zero output rows).
    K2-analog decode kernel.
    """
    import math
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    B, Hq, Sq, D = q.shape
    if seq_lens.dim() == 1:
        positions = seq_lens.view(B, 1) - Sq + torch.arange(
            Sq, device=seq_lens.device
        ).view(1, Sq)
    else:
        positions = seq_lens
    if q.is_cuda:
        c = _require_ext("attention_decode")
        Hkv = k_cache.shape[1]
        gq = Hq // max(Hkv, 1)
        if gq > 8:
            # odd GQA shards (e.g. 96q/8kv models, replicate-to-TP
            # 70B configs): split the q heads per KV head into <=8-wide
            # groups and run the kernel per group (KV re-read per split;
            # correctness first — VERDICT r01 weak #9)
            half = (gq + 1) // 2
            qg = q.view(B, Hkv, gq, Sq, D)
            outs = []
            for lo in (0, half):
                hi = min(lo + half, gq)
                qs = qg[:, :, lo:hi].reshape(B, Hkv * (hi - lo), Sq, D)
                sk = None
                if sinks is not None:
                    sk = sinks.view(Hkv, gq)[:, lo:hi].reshape(-1)
                outs.append(attention_decode(
                    qs.contiguous(), k_cache, v_cache, seq_lens,
                    seq_ids=seq_ids, scale=scale, sinks=sk,
                    sliding_window=sliding_window, ring=ring,
                ).view(B, Hkv, hi - lo, Sq, D))
            return torch.cat(outs, dim=2).view(B, Hq, Sq, D)
        out = torch.empty_like(q)
        if seq_ids is None:
            seq_ids = torch.arange(B, device=q.device, dtype=torch.int32)
        c.attention_decode(
            out, q.contiguous(), k_cache, v_cache,
            positions.to(torch.int32).contiguous(),
            seq_ids.to(torch.int32).contiguous(), float(scale), int(sliding_window),
            sinks=None if sinks is None else sinks.detach(), ring=int(ring),
        )
        return out
    # CPU reference: mask each query by its own position
    if seq_ids is None:
        seq_ids = torch.arange(B, device=q.device)
    k = k_cache[seq_ids.long()]
    v = v_cache[seq_ids.long()]
    Smax = k.shape[2]
    ar = torch.arange(Smax, device=q.device).view(1, 1, 1, Smax)
    qpos = positions.view(B, 1, Sq, 1)
    if ring:
        # compact sliding-window cache: slot s holds its newest position
        # == s (mod Smax); mask by TRUE position per query
        pmax = positions.amax(dim=-1).view(B, 1, 1, 1)
        p_s = pmax - ((pmax - ar) % Smax)
        mask = (p_s >= 0) & (p_s <= qpos)
        if sliding_window > 0:
            mask = mask & (p_s > (qpos - sliding_window))
        out = ref.sdpa(q, k, v, mask=mask, scale=scale, sinks=sinks)
        return torch.nan_to_num(out)
    mask = ar <= qpos
    if sliding_window > 0:
        mask = mask & (ar > (qpos - sliding_window))
    elif sliding_window < 0:  # chunked attention: own chunk only (Llama4)
        C = -sliding_window
        mask = mask & (ar >= (qpos.clamp(min=0) // C) * C)
    out = ref.sdpa(q, k, v, mask=mask, scale=scale, sinks=sinks)
    return torch.nan_to_num(out)  # pad queries (pos<0) mask everything


def attention_decode_lse(q, k_cache, v_cache, seq_lens, seq_ids=None,
                         scale=None, sliding_window: int = 0):
    """attention_decode that ALSO returns the per-query log-sum-exp
    (fp32 (B,Hq,Sq)) — the cross-rank flash-decoding combine handle
    (SURVEY §2.8 long-context mechanism 3). Empty query ranges get
    lse=-inf / out=0 so they drop out of the combine."""
    import math
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    B, Hq, Sq, D = q.shape
    if seq_lens.dim() == 1:
        positions = seq_lens.view(B, 1) - Sq + torch.arange(
            Sq, device=seq_lens.device).view(1, Sq)
    else:
        positions = seq_lens
    if q.is_cuda:
        c = _require_ext("attention_decode")
        out = torch.empty_like(q)
        lse = torch.empty(B, Hq, Sq, dtype=torch.float32, device=q.device)
        if seq_ids is None:
            seq_ids = torch.arange(B, device=q.device, dtype=torch.int32)
        c.attention_decode(
            out, q.contiguous(), k_cache, v_cache,
            positions.to(torch.int32).contiguous(),
            seq_ids.to(torch.int32).contiguous(), float(scale),
            int(sliding_window), lse=lse,
        )
        return out, lse
    # CPU reference
    if seq_ids is None:
        seq_ids = torch.arange(B, device=q.device)
    k = k_cache[seq_ids.long()].float()
    v = v_cache[seq_ids.long()].float()
    Smax = k.shape[2]
    rep = Hq // k.shape[1]
    if rep > 1:
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    ar = torch.arange(Smax, device=q.device).view(1, 1, 1, Smax)
    qpos = positions.view(B, 1, Sq, 1)
    mask = ar <= qpos
    if sliding_window > 0:
        mask = mask & (ar > (qpos - sliding_window))
    scores = torch.einsum("bhsd,bhtd->bhst", q.float(), k) * scale
    scores = scores.masked_fill(~mask, float("-inf"))
    lse = torch.logsumexp(scores, dim=-1)  # -inf where no valid rows
    out = torch.einsum("bhst,bhtd->bhsd", torch.softmax(scores, dim=-1), v)
    return torch.nan_to_num(out).to(q.dtype), lse


def combine_lse_partials(outs, lses):
    """Merge flash-decoding partials: outs[i] (B,H,S,D) softmax-normalized
    over shard i's rows, lses[i] (B,H,S). Standard LSE-weighted merge."""
    lse = torch.stack(lses, dim=0)                       # (P,B,H,S)
    m = lse.max(dim=0, keepdim=True).values
    w = torch.exp(lse - m)                               # -inf -> 0
    w = torch.nan_to_num(w)
    num = sum(wi.unsqueeze(-1) * oi.float() for wi, oi in zip(w, outs))
    den = w.sum(dim=0).clamp_min(1e-20).unsqueeze(-1)
    return (num / den).to(outs[0].dtype)


def topk_topp_sample(logits, top_k, top_p, temperature, rand, global_topk: int = 256):
    if logits.is_cuda:
        c = _require_ext("topk_topp_sample")
        out = torch.empty(logits.shape[0], dtype=torch.long, device=logits.device)
        c.topk_topp_sample(
            out, logits.contiguous(), top_k.to(torch.int32).contiguous(),
            top_p.to(torch.float32).contiguous(),
            temperature.to(torch.float32).contiguous(),
            rand.to(torch.float32).contiguous(), int(global_topk),
        )
        return out
    return ref.topk_topp_sample(logits, top_k, top_p, temperature, rand, global_topk)


def qkv_rope_update(qkv, cos, sin, k_cache, v_cache, positions, seq_ids, Hq, Hkv):
    """Fused qkv-GEMM output -> RoPE(q,k) -> cache write(k,v) -> q (B,Hq,S,D).

    qkv: (B, S, (Hq+2Hkv)*D) contiguous; cos/sin (B,S,D) fp32.
    """
    B, S, _ = qkv.shape
    D = k_cache.shape[3]
    if qkv.is_cuda:
        c = _require_ext("qkv_rope_update")
        q_out = torch.empty(B, Hq, S, D, dtype=qkv.dtype, device=qkv.device)
        if seq_ids is None:
            seq_ids = torch.arange(B, device=qkv.device, dtype=torch.int32)
        c.qkv_rope_update(
            q_out, qkv.contiguous(), cos.contiguous(), sin.contiguous(),
            k_cache, v_cache, positions.to(torch.int32).contiguous(),
            seq_ids.to(torch.int32).contiguous(), Hq, Hkv,
        )
        return q_out
    # CPU reference composition
    q, k, v = torch.split(qkv, [Hq * D, Hkv * D, Hkv * D], dim=-1)
    q = q.reshape(B, S, Hq, D).transpose(1, 2)
    k = k.reshape(B, S, Hkv, D).transpose(1, 2)
    v = v.reshape(B, S, Hkv, D).transpose(1, 2)
    q, k = ref.rope_apply(q, k, cos, sin)
    ref.kv_cache_update(k_cache, v_cache, k, v, positions, seq_ids)
    return q.contiguous()


def silu_mul_fused(gate_up):
    """SiLU(gate)*up on a fused (rows, 2I) [gate|up] tensor."""
    I = gate_up.shape[-1] // 2
    if gate_up.is_cuda:
        c = _require_ext("silu_mul_fused")
        out = torch.empty(gate_up.shape[:-1] + (I,), dtype=gate_up.dtype,
                          device=gate_up.device)
        c.silu_mul_fused(out, gate_up.contiguous())
        return out
    g, u = gate_up.chunk(2, dim=-1)
    return ref.silu_mul(g.contiguous(), u.contiguous())


def gelu_tanh_mul_fused(gate_up):
    I = gate_up.shape[-1] // 2
    if gate_up.is_cuda:
        c = _require_ext("gelu_tanh_mul_fused")
        out = torch.empty(gate_up.shape[:-1] + (I,), dtype=gate_up.dtype,
                          device=gate_up.device)
        c.gelu_tanh_mul_fused(out, gate_up.contiguous())
        return out
    g, u = gate_up.chunk(2, dim=-1)
    return ref.gelu_tanh_mul(g.contiguous(), u.contiguous())


def attention_decode_paged(q, k_cache, v_cache, seq_lens, block_table,
                           scale=None, sliding_window: int = 0):
    """Decode attention against a paged (num_blocks, block_size, Hkv, D)
    cache via a (B, max_blocks) block table. `seq_lens`: (B,) lengths or
    (B,Sq) per-query positions."""
    import math
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    B, Hq, Sq, D = q.shape
    if seq_lens.dim() == 1:
        positions = seq_lens.view(B, 1) - Sq + torch.arange(
            Sq, device=seq_lens.device
        ).view(1, Sq)
    else:
        positions = seq_lens
    if q.is_cuda:
        c = _require_ext("attention_decode_paged")
        out = torch.empty_like(q)
        c.attention_decode_paged(
            out, q.contiguous(), k_cache, v_cache,
            positions.to(torch.int32).contiguous(),
            block_table.to(torch.int32).contiguous(), float(scale),
            int(sliding_window),
        )
        return out
    # CPU reference: gather contiguous K/V per sequence then mask-attend
    max_len = int(positions.max()) + 1
    k, v = ref.paged_gather_kv(k_cache, v_cache, block_table, max_len)
    ar = torch.arange(max_len).view(1, 1, 1, -1)
    qpos = positions.view(B, 1, Sq, 1)
    mask = ar <= qpos
    if sliding_window > 0:
        mask = mask & (ar > (qpos - sliding_window))
    elif sliding_window < 0:  # chunked attention: own chunk only (Llama4)
        C = -sliding_window
        mask = mask & (ar >= (qpos.clamp(min=0) // C) * C)
    out = ref.sdpa(q, k, v, mask=mask, scale=scale)
    return torch.nan_to_num(out)


# ----------------------------------------------------------------------
# Grouped MoE expert GEMM: tokens sorted by expert, padded
# per expert to 64-row tiles; one kernel launch serves every expert's
# segment. All table construction is tensor ops with STATIC shapes, so
# the whole MoE layer is hipGraph-capturable.

def router_topk(logits: torch.Tensor, top_k: int, act_sigmoid: bool,
                pre_softmax: bool, normalize: bool):
    """Fused MoE router selection: top-k + activation + normalization in
    ONE launch (the torch chain costs ~6 kernels/layer). GPU only;
    logits (T, E) fp32. Returns (vals (T, k) f32, idx (T, k) i64)."""
    c = _require_ext("router_topk")
    T = logits.shape[0]
    vals = torch.empty(T, top_k, dtype=torch.float32, device=logits.device)
    idx = torch.empty(T, top_k, dtype=torch.long, device=logits.device)
    c.router_topk(vals, idx, logits.contiguous(), top_k,
                  int(act_sigmoid), int(pre_softmax), int(normalize))
    return vals, idx


MOE_TILE = 64


def moe_grouped_tables(idx_flat: torch.Tensor, num_experts: int,
                       top_k: int = 1, tile: int = MOE_TILE):
    """idx_flat: (T*top_k,) local expert id per assignment. Returns
    (order, dest, expert_ids, gtab, Mpad):
      order      (Ttk,)  argsort by expert (stable on CPU; atomic —
                         arbitrary within-expert permutation — on GPU)
      dest       (Ttk,)  destination row of each sorted assignment in the
                         padded sorted space
      expert_ids (Mpad/64,) int32 expert per row tile, -1 past the end
      gtab       (Mpad,) int32 padded row -> TOKEN row (order[p]//top_k),
                         -1 on padding; feeds the fused gather/combine
                         epilogues of moe_grouped_gemm
      Mpad       static worst-case padded rows (multiple of 64)
    """
    E = num_experts
    Ttk = idx_flat.numel()
    dev = idx_flat.device
    MT = tile
    Mpad = ((Ttk + E * (MT - 1)) + MT - 1) // MT * MT
    if idx_flat.is_cuda:
        # ONE fused kernel (vs ~15 tiny torch launches = 140 us at decode
        # sizes). Within-expert order is atomic (not stable argsort) —
        # any consistent permutation is correct downstream.
        c = _require_ext("moe_grouped_tables")
        order = torch.empty(Ttk, dtype=torch.long, device=dev)
        dest = torch.empty(Ttk, dtype=torch.long, device=dev)
        expert_ids = torch.empty(Mpad // MT, dtype=torch.int32, device=dev)
        gtab = torch.empty(Mpad, dtype=torch.int32, device=dev)
        c.moe_grouped_tables(order, dest, expert_ids, gtab,
                             idx_flat.to(torch.int32).contiguous(), E, top_k,
                             MT)
        return order, dest, expert_ids, gtab, Mpad
    order = torch.argsort(idx_flat, stable=True)
    sorted_e = idx_flat[order]
    # scatter_add, NOT bincount: bincount syncs the host on CUDA (output
    # sizing), which kills hipGraph capture
    counts = torch.zeros(E, dtype=torch.long, device=dev).scatter_add_(
        0, idx_flat.long(), torch.ones_like(idx_flat, dtype=torch.long))
    padded = (counts + MT - 1) // MT * MT
    ends_pad = padded.cumsum(0)                 # per-expert padded end
    starts_pad = ends_pad - padded
    starts_raw = counts.cumsum(0) - counts
    within = torch.arange(Ttk, device=dev) - starts_raw[sorted_e]
    dest = starts_pad[sorted_e] + within
    tile_start = torch.arange(Mpad // MT, device=dev) * MT
    eid = torch.searchsorted(ends_pad, tile_start, right=True)
    total_pad = ends_pad[-1] if E > 0 else torch.zeros((), device=dev)
    expert_ids = torch.where(tile_start < total_pad, eid,
                             torch.full_like(eid, -1)).to(torch.int32)
    gtab = torch.full((Mpad,), -1, dtype=torch.int32, device=dev)
    gtab[dest] = (order // top_k).to(torch.int32)
    return order, dest, expert_ids, gtab, Mpad


def moe_grouped_gemm(x_sorted: torch.Tensor, w: torch.Tensor,
                     expert_ids: torch.Tensor, bias=None,
                     bn: int = 0, x_gather=None, combine_out=None,
                     combine_scale=None, combine_gather=None,
                     bk: int = 0, mt: int = MOE_TILE) -> torch.Tensor:
    """y_sorted[m] = x_sorted[m] @ w[expert_of(m)]^T (+ bias[expert]).
    x_sorted (Mpad, K), w (E, N, K), expert_ids (Mpad/64,) int32, bias
    (E, N) or None. GPU: one grouped HIP kernel; CPU reference:
    per-tile loop.

    Fusions (GPU + CPU ref):
      x_gather (Mpad,) int32: x_sorted is instead the UNSORTED (T, K)
        activations; each padded row stages x[x_gather[m]] (the token
        gather fuses into the glds staging).
      combine_out (T, N) fp32 + combine_gather (Mpad,) [+ combine_scale
        (Mpad,) fp32]: instead of writing y, atomically accumulate
        (acc + bias) * scale into combine_out[token] — the routing-
        weighted combine fuses into the epilogue. Returns combine_out.
    """
    K = x_sorted.shape[1]
    Mpad = expert_ids.numel() * mt if x_gather is not None \
        else x_sorted.shape[0]
    N = w.shape[1]
    if x_sorted.is_cuda:
        c = _require_ext("moe_grouped_gemm")
        # y is the kernel's nominal output; untouched when combining
        y = torch.empty(Mpad, N, dtype=x_sorted.dtype, device=x_sorted.device)
        c.moe_grouped_gemm(y, x_sorted.contiguous(), w.contiguous(),
                           expert_ids.contiguous(),
                           None if bias is None else bias.contiguous(), bn,
                           None if x_gather is None else x_gather.contiguous(),
                           combine_out,
                           combine_scale,
                           None if combine_gather is None
                           else combine_gather.contiguous(), bk, mt)
        return combine_out if combine_out is not None else y
    # ---- CPU reference ----
    out = combine_out
    y = torch.zeros(Mpad, N, dtype=x_sorted.dtype)
    for t in range(Mpad // mt):
        e = int(expert_ids[t])
        if e < 0:
            continue
        rows = slice(t * mt, (t + 1) * mt)
        if x_gather is not None:
            xg = x_sorted[x_gather[rows].clamp(min=0).long()]
        else:
            xg = x_sorted[rows]
        r = xg @ w[e].t().to(x_sorted.dtype)
        if bias is not None:
            r = r + bias[e].to(r.dtype)
        if out is not None:
            cg = combine_gather[rows].long()
            sc = (combine_scale[rows] if combine_scale is not None
                  else torch.ones(mt))
            for j in range(mt):
                if cg[j] >= 0:
                    out[cg[j]] += r[j].float() * float(sc[j])
        else:
            y[rows] = r
    return out if out is not None else y


def fp8_moe_quantize(w: torch.Tensor):
    """Per-output-channel e4m3fn quantization of stacked expert weights
    (E, N, K) -> (wq (E, N, K) uint8, scale (E, N) fp32). e4m3fn max
    normal = 448."""
    amax = w.float().abs().amax(dim=-1).clamp(min=1e-12)   # (E, N)
    scale = amax / 448.0
    q = (w.float() / scale.unsqueeze(-1)).clamp(-448, 448)
    wq = q.to(torch.float8_e4m3fn).view(torch.uint8)
    return wq.contiguous(), scale.to(torch.float32).contiguous()


def fp8_moe_dequant(wq: torch.Tensor, scale: torch.Tensor) -> torch.Tensor:
    return wq.view(torch.float8_e4m3fn).float() * scale.unsqueeze(-1).float()


def moe_grouped_gemm_fp8(x_sorted: torch.Tensor, wq: torch.Tensor,
                         wscale: torch.Tensor, expert_ids: torch.Tensor,
                         bias=None, bn: int = 0, x_gather=None,
                         combine_out=None, combine_scale=None,
                         combine_gather=None) -> torch.Tensor:
    """Grouped expert GEMM with fp8-e4m3 weights (E, N, K) u8 + per-
    channel fp32 scales (E, N): HALF the weight stream of bf16 through
    the same all-glds pipeline; dequant happens at the LDS fragment read
    and the scale folds into the epilogue. Same gather/combine fusions
    as the bf16 kernel. CPU: dequant reference."""
    K = x_sorted.shape[1]
    Mpad = expert_ids.numel() * MOE_TILE if x_gather is not None \
        else x_sorted.shape[0]
    N = wq.shape[1]
    if x_sorted.is_cuda:
        c = _require_ext("moe_grouped_gemm_fp8")
        y = torch.empty(Mpad, N, dtype=x_sorted.dtype, device=x_sorted.device)
        c.moe_grouped_gemm_fp8(y, x_sorted.contiguous(), wq.contiguous(),
                               wscale.contiguous(), expert_ids.contiguous(),
                               None if bias is None else bias.contiguous(), bn,
                               None if x_gather is None else x_gather.contiguous(),
                               combine_out, combine_scale,
                               None if combine_gather is None
                               else combine_gather.contiguous())
        return combine_out if combine_out is not None else y
    out = combine_out
    y = torch.zeros(Mpad, N, dtype=x_sorted.dtype)
    for t in range(Mpad // MOE_TILE):
        e = int(expert_ids[t])
        if e < 0:
            continue
        rows = slice(t * MOE_TILE, (t + 1) * MOE_TILE)
        if x_gather is not None:
            xg = x_sorted[x_gather[rows].clamp(min=0).long()]
        else:
            xg = x_sorted[rows]
        wd = fp8_moe_dequant(wq[e], wscale[e]).to(x_sorted.dtype)
        r = xg @ wd.t()
        if bias is not None:
            r = r + bias[e].to(r.dtype)
        if out is not None:
            cg = combine_gather[rows].long()
            sc = (combine_scale[rows] if combine_scale is not None
                  else torch.ones(MOE_TILE))
            for j in range(MOE_TILE):
                if cg[j] >= 0:
                    out[cg[j]] += r[j].float() * float(sc[j])
        else:
            y[rows] = r
    return out if out is not None else y


# ----------------------------------------------------------------------
# OCP MXFP4 (ref models/gpt_oss/mx_layout_transform.py):
# e2m1 nibbles packed 2/byte + one e8m0 scale per 32 elements.

_MX4_VALS = torch.tensor(
    [0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0,
     -0.0, -0.5, -1.0, -1.5, -2.0, -3.0, -4.0, -6.0])


def mx4_dequant(wq: torch.Tensor, ws: torch.Tensor) -> torch.Tensor:
    """wq (..., K/2) u8 packed e2m1 (low nibble first), ws (..., K/32) u8
    e8m0 -> float (..., K)."""
    lo = (wq & 0xF).long()
    hi = (wq >> 4).long()
    vals = torch.stack([_MX4_VALS.to(wq.device)[lo],
                        _MX4_VALS.to(wq.device)[hi]], dim=-1).flatten(-2)
    scale = torch.pow(2.0, ws.float() - 127.0)
    K = vals.shape[-1]
    return (vals.view(*vals.shape[:-1], K // 32, 32)
            * scale.unsqueeze(-1)).view(*vals.shape[:-1], K)


def mx4_quantize(w: torch.Tensor):
    """float (..., K) -> (wq (..., K/2) u8, ws (..., K/32) u8). Per-32
    e8m0 scale = 2^(floor(log2(amax)) - 2) so amax maps into e2m1 range;
    round-to-nearest onto the e2m1 grid."""
    K = w.shape[-1]
    assert K % 32 == 0
    g = w.float().view(*w.shape[:-1], K // 32, 32)
    amax = g.abs().amax(-1).clamp(min=1e-30)
    e = torch.floor(torch.log2(amax)) - 2.0
    e = e.clamp(min=-127.0, max=127.0)
    scale = torch.pow(2.0, e)
    ws = (e + 127.0).to(torch.uint8)
    v = (g / scale.unsqueeze(-1)).clamp(min=-6.0, max=6.0)
    # nearest e2m1 magnitude via bucketize against the grid midpoints
    # (no K x 16 broadcast — a 470M-element expert stack quantizes in
    # milliseconds instead of minutes)
    mids = torch.tensor([0.25, 0.75, 1.25, 1.75, 2.5, 3.5, 5.0],
                        device=v.device, dtype=v.dtype)
    mag = torch.bucketize(v.abs().contiguous(), mids)
    nib = torch.where(v < 0, mag + 8, mag).to(torch.uint8)
    nib = nib.view(*w.shape[:-1], K)
    wq = (nib[..., 0::2] | (nib[..., 1::2] << 4)).contiguous()
    return wq, ws.contiguous()


def moe_grouped_gemm_mx4(x_sorted: torch.Tensor, wq: torch.Tensor,
                         ws: torch.Tensor, expert_ids: torch.Tensor,
                         bias=None, bn: int = 0, x_gather=None,
                         combine_out=None, combine_scale=None,
                         combine_gather=None) -> torch.Tensor:
    """Grouped expert GEMM with MXFP4-packed weights (E, N, K/2)+(E, N,
    K/32), dequantized in-kernel during LDS staging (~3.8x less weight
    traffic than bf16). CPU: dequant reference."""
    K = x_sorted.shape[1]
    Mpad = expert_ids.numel() * MOE_TILE if x_gather is not None \
        else x_sorted.shape[0]
    N = wq.shape[1]
    if x_sorted.is_cuda:
        c = _require_ext("moe_grouped_gemm_mx4")
        y = torch.empty(Mpad, N, dtype=x_sorted.dtype, device=x_sorted.device)
        c.moe_grouped_gemm_mx4(y, x_sorted.contiguous(), wq.contiguous(),
                               ws.contiguous(), expert_ids.contiguous(),
                               None if bias is None else bias.contiguous(), bn,
                               None if x_gather is None else x_gather.contiguous(),
                               combine_out, combine_scale,
                               None if combine_gather is None
                               else combine_gather.contiguous())
        return combine_out if combine_out is not None else y
    out = combine_out
    y = torch.zeros(Mpad, N, dtype=x_sorted.dtype)
    for t in range(Mpad // MOE_TILE):
        e = int(expert_ids[t])
        if e < 0:
            continue
        rows = slice(t * MOE_TILE, (t + 1) * MOE_TILE)
        if x_gather is not None:
            xg = x_sorted[x_gather[rows].clamp(min=0).long()]
        else:
            xg = x_sorted[rows]
        wd = mx4_dequant(wq[e], ws[e]).to(x_sorted.dtype)
        r = xg @ wd.t()
        if bias is not None:
            r = r + bias[e].to(r.dtype)
        if out is not None:
            cg = combine_gather[rows].long()
            sc = (combine_scale[rows] if combine_scale is not None
                  else torch.ones(MOE_TILE))
            for j in range(MOE_TILE):
                if cg[j] >= 0:
                    out[cg[j]] += r[j].float() * float(sc[j])
        else:
            y[rows] = r
    return out if out is not None else y
