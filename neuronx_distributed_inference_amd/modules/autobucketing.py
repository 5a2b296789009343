"""Bucket-ladder generation.

Analog of the reference's `modules/autobucketing.py` (:8-341): powers-of-2
ladders of sequence lengths (context encoding) and the fixed-shape token-gen
bucket, used both to pad inputs and to decide which hipGraph capture to
dispatch to. 2-D [batch, seq] buckets for prefix caching / chunked prefill
follow the same shapes as the reference (:22-146).
"""

from __future__ import annotations

import math
from typing import List, Optional, Tuple


def generate_buckets(min_length: int, max_length: int) -> List[int]:
    """Powers-of-2 ladder from min to max inclusive (ref :8-19)."""
    if min_length >= max_length:
        return [max_length]
    lo = max(int(math.log2(min_length)), 0)
    hi = int(math.ceil(math.log2(max_length)))
    out = [2 ** i for i in range(lo, hi + 1) if 2 ** i >= min_length]
    out = [b for b in out if b < max_length] + [max_length]
    return sorted(set(out))


def generate_2d_buckets(
    batch_buckets: List[int], seq_buckets: List[int]
) -> List[Tuple[int, int]]:
    return [(b, s) for b in batch_buckets for s in seq_buckets]


def generate_buckets_for_cte(
    max_context_length: int,
    max_length: int,
    buckets: Optional[List[int]] = None,
    enable: bool = True,
    min_bucket: int = 128,
) -> List[int]:
    """Context-encoding seq-length buckets (ref :149-225)."""
    if buckets:
        return sorted(set(buckets))
    if not enable:
        return [max_context_length]
    return generate_buckets(min(min_bucket, max_context_length), max_context_length)


def generate_buckets_for_tkg(
    max_length: int,
    buckets: Optional[List[int]] = None,
    enable: bool = True,
    min_bucket: int = 128,
) -> List[int]:
    """Token-generation buckets over the ATTENDED sequence length — the KV
    window the decode kernel scans (ref :226-281)."""
    if buckets:
        return sorted(set(buckets))
    if not enable:
        return [max_length]
    return generate_buckets(min(min_bucket, max_length), max_length)


def generate_buckets_for_fused_spec(max_length: int, enable: bool = True) -> List[int]:
    return generate_buckets_for_tkg(max_length, enable=enable)


def generate_batch_buckets(
    max_batch: int,
    buckets: Optional[List[int]] = None,
    enable: bool = True,
    max_graphs: int = 4,
) -> List[int]:
    """Decode BATCH sub-buckets (ref batch x seq 2-D buckets,
    autobucketing.py:203-223): a serving engine at partial occupancy
    replays the smallest captured graph covering its active batch instead
    of always paying full-batch cost. Powers-of-2 ladder, capped at
    `max_graphs` rungs ending at max_batch."""
    if buckets:
        out = sorted(set(int(b) for b in buckets))
        if out[-1] != max_batch:
            out.append(max_batch)
        return out
    if not enable or max_batch <= 1:
        return [max_batch]
    ladder = generate_buckets(1, max_batch)
    return ladder[-max_graphs:]


def get_target_bucket(length: int, buckets: List[int], strategy: str = "first_fit") -> int:
    """Bucket selection (ref model_wrapper.get_target_bucket :831 +
    pad_inputs strategies :730-829):
      first_fit  — smallest bucket >= length (default)
      second_fit — one rung above first fit when it exists (headroom for
                   in-place growth before the next re-bucket)
      max        — always the largest bucket
    """
    bs = sorted(buckets)
    if strategy == "max":
        return bs[-1]
    for i, b in enumerate(bs):
        if b >= length:
            if strategy == "second_fit" and i + 1 < len(bs):
                return bs[i + 1]
            return b
    raise ValueError(f"length {length} exceeds largest bucket {max(buckets)}")
