"""Benchmark harness.

Analog of the reference's `utils/benchmark.py` (:1-516): e2e + per-submodel
latency percentiles via pre/post hooks (`LatencyCollector` :484-493,
registered after warmup :120-133) and throughput =
n_runs * max_length * batch / total_time (:496-516), written to
`benchmark_report.json`.
"""

from __future__ import annotations

import json
import time
from typing import Dict, List, Optional

import torch

from neuronx_distributed_inference_amd.utils.hf_adapter import (
    GenerationConfig,
    HuggingFaceGenerationAdapter,
)


class LatencyCollector:
    """Wall-clock collector wrapped around a callable submodel."""

    def __init__(self):
        self.latencies: List[float] = []
        self._t0 = None

    def pre(self):
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self._t0 = time.perf_counter()

    def post(self):
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self.latencies.append(time.perf_counter() - self._t0)

    def percentile(self, p: float) -> float:
        if not self.latencies:
            return 0.0
        s = sorted(self.latencies)
        idx = min(int(len(s) * p / 100.0), len(s) - 1)
        return s[idx]

    def report(self) -> Dict[str, float]:
        if not self.latencies:
            return {}
        return {
            "latency_ms_p50": self.percentile(50) * 1e3,
            "latency_ms_p90": self.percentile(90) * 1e3,
            "latency_ms_p95": self.percentile(95) * 1e3,
            "latency_ms_p99": self.percentile(99) * 1e3,
            "latency_ms_p100": self.percentile(100) * 1e3,
            "latency_ms_avg": sum(self.latencies) / len(self.latencies) * 1e3,
        }


class _WrappedSubmodel:
    def __init__(self, inner, collector: LatencyCollector):
        self._inner = inner
        self._collector = collector

    def __call__(self, *a, **kw):
        self._collector.pre()
        out = self._inner(*a, **kw)
        self._collector.post()
        return out

    def __getattr__(self, name):
        return getattr(self._inner, name)


def benchmark_sampling(
    model,
    input_ids: torch.Tensor,
    attention_mask: Optional[torch.Tensor] = None,
    n_runs: int = 5,
    max_new_tokens: int = 64,
    generation_config: Optional[GenerationConfig] = None,
    report_path: Optional[str] = "benchmark_report.json",
) -> Dict:
    """Run `n_runs` full generate() calls, collect e2e + per-submodel
    latencies, return (and write) the report."""
    gc = generation_config or GenerationConfig(max_new_tokens=max_new_tokens)
    adapter = HuggingFaceGenerationAdapter(model)
    e2e = LatencyCollector()
    ctx_col, tkg_col = LatencyCollector(), LatencyCollector()
    orig_cte, orig_tkg = model.context_encoding_model, model.token_generation_model
    if orig_cte is not None:
        model.context_encoding_model = _WrappedSubmodel(orig_cte, ctx_col)
    if orig_tkg is not None:
        model.token_generation_model = _WrappedSubmodel(orig_tkg, tkg_col)
    try:
        # warmup
        adapter.generate(input_ids, attention_mask=attention_mask, generation_config=gc)
        ctx_col.latencies.clear()
        tkg_col.latencies.clear()
        total_t0 = time.perf_counter()
        out = None
        for _ in range(n_runs):
            e2e.pre()
            out = adapter.generate(
                input_ids, attention_mask=attention_mask, generation_config=gc
            )
            e2e.post()
        total = time.perf_counter() - total_t0
    finally:
        model.context_encoding_model = orig_cte
        model.token_generation_model = orig_tkg

    batch = input_ids.shape[0]
    gen_len = out.shape[1] - input_ids.shape[1] if out is not None else max_new_tokens
    report = {
        "e2e_model": {
            **e2e.report(),
            "throughput": n_runs * (input_ids.shape[1] + gen_len) * batch / total,
            "output_token_throughput": n_runs * gen_len * batch / total,
        },
        "context_encoding_model": ctx_col.report(),
        "token_generation_model": tkg_col.report(),
    }
    if report_path:
        with open(report_path, "w") as f:
            json.dump(report, f, indent=2)
    return report
