"""Module/function test harness.

Analog of the reference's `utils/testing.py` (build_function/build_module
:123-267 wrap a callable for device execution; validate_accuracy :67-120
compares against a CPU callable; CPU distributed env helpers :40-64).

MI355X translation of tier 2: "build_function -> hipGraph
capture vs eager CPU" — wrap any callable, optionally capture it as a
hipGraph at fixed shapes, and validate against a CPU fp32 reference.
"""

from __future__ import annotations

import os
from typing import Callable, Optional, Sequence

import torch
import torch.nn as nn


class FunctionModule(nn.Module):
    def __init__(self, fn: Callable):
        super().__init__()
        self.fn = fn

    def forward(self, *args):
        return self.fn(*args)


def build_function(fn: Callable, example_inputs: Sequence[torch.Tensor],
                   device: Optional[torch.device] = None, capture: bool = False):
    """Return a callable running `fn` on `device` (hipGraph-captured when
    requested and a GPU is present)."""
    device = device or (
        torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    )
    mod = FunctionModule(fn).to(device)
    inputs = tuple(
        t.to(device) if isinstance(t, torch.Tensor) else t for t in example_inputs
    )
    if capture and device.type == "cuda":
        from neuronx_distributed_inference_amd.runtime.graph_engine import GraphRunner

        return GraphRunner(mod, inputs), inputs
    return mod, inputs


def validate_accuracy(
    device_fn: Callable,
    cpu_fn: Callable,
    example_inputs: Sequence[torch.Tensor],
    atol: float = 3e-2,
    rtol: float = 3e-2,
):
    """Run both callables on the same inputs (device vs CPU fp32) and
    assert closeness (ref validate_accuracy + assert_close)."""
    dev_inputs = tuple(
        t.cuda() if torch.cuda.is_available() and isinstance(t, torch.Tensor) else t
        for t in example_inputs
    )
    cpu_inputs = tuple(
        t.float().cpu() if isinstance(t, torch.Tensor) and t.is_floating_point()
        else (t.cpu() if isinstance(t, torch.Tensor) else t)
        for t in example_inputs
    )
    got = device_fn(*dev_inputs)
    want = cpu_fn(*cpu_inputs)
    if isinstance(got, (tuple, list)):
        for g, w in zip(got, want):
            if g is not None and w is not None:
                torch.testing.assert_close(g.float().cpu(), w.float().cpu(),
                                           atol=atol, rtol=rtol)
    else:
        torch.testing.assert_close(got.float().cpu(), want.float().cpu(),
                                   atol=atol, rtol=rtol)
    return got


def init_cpu_env(world_size: int = 1, rank: int = 0, port: int = 29555):
    """gloo world for CPU logic tests (ref init_cpu_env :40-64)."""
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", str(port))
    if not dist.is_initialized():
        dist.init_process_group("gloo", rank=rank, world_size=world_size)


def destroy_cpu_env():
    import torch.distributed as dist

    from neuronx_distributed_inference_amd.parallel import state as ps

    ps.destroy_model_parallel()
    if dist.is_initialized():
        dist.destroy_process_group()
