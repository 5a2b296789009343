"""hipGraph capture engine.

The MI355X replacement for the reference's trace -> neuronx-cc -> NEFF
pipeline (`models/model_wrapper.py` + ModelBuilder): each (submodel, bucket)
is captured once as a hipGraph (`torch.cuda.CUDAGraph` == hipGraph on ROCm)
with static input/output buffers, then replayed per step — one kernel-launch
boundary instead of hundreds of eager launches, which is where decode
latency lives. Capture happens at load()/warmup() time in seconds (vs
minutes per bucket for neuronx-cc); nothing needs serializing, so the
compiled-model artifact is just config + sharded weights.

KV-cache updates inside the captured region are in-place scatters into
persistent cache tensors, so replay reads what it wrote — the aliasing
machinery of the reference (`model_wrapper.py:1588-1626`) has no analog
here by construction.

Capture rules honored here: RCCL collectives are capturable on ROCm
(graph-captured all-reduce), allocations inside capture go to the graph
private pool, and nothing in the captured callable may sync the host.
"""

from __future__ import annotations

import logging
from typing import Callable, Dict, List, Optional, Tuple

import torch

logger = logging.getLogger("nxdi_amd")


class GraphRunner:
    """Captures `fn(*tensors)` at fixed shapes; replays with buffer copy-in.

    Outputs are the capture's static tensors — consumers must copy/consume
    before the next replay (the generation loop does).
    """

    def __init__(self, fn: Callable, example_inputs: Tuple[torch.Tensor, ...],
                 warmup_iters: int = 2, pool=None):
        self.fn = fn
        self.static_inputs = tuple(
            t.clone() if isinstance(t, torch.Tensor) else t for t in example_inputs
        )
        stream = torch.cuda.Stream()
        stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(stream):
            for _ in range(warmup_iters):
                out = fn(*self.static_inputs)
        torch.cuda.current_stream().wait_stream(stream)
        torch.cuda.synchronize()

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph, pool=pool):
            self.static_outputs = fn(*self.static_inputs)
        self.pool = self.graph.pool()

    def __call__(self, *inputs):
        for s, x in zip(self.static_inputs, inputs):
            if isinstance(s, torch.Tensor) and x is not None:
                s.copy_(x, non_blocking=True)
        self.graph.replay()
        return self.static_outputs


class BucketedGraphModel:
    """Per-bucket GraphRunner store for one submodel, sharing one memory
    pool across buckets (buckets replay one-at-a-time)."""

    def __init__(self):
        self.runners: Dict[tuple, GraphRunner] = {}
        self._pool = None

    def has(self, key) -> bool:
        return key in self.runners

    def capture(self, key, fn, example_inputs, warmup_iters: int = 2):
        logger.info("capturing hipGraph for bucket %s", key)
        r = GraphRunner(fn, example_inputs, warmup_iters=warmup_iters, pool=self._pool)
        if self._pool is None:
            self._pool = r.pool
        self.runners[key] = r
        return r

    def run(self, key, *inputs):
        return self.runners[key](*inputs)
