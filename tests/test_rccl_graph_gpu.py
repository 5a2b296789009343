"""RCCL-inside-hipGraph pre-flight.

Multi-GPU decode graphs capture RCCL collectives (all-reduce after every
row-parallel GEMM, 2x/layer). Before any 8-GPU run, verify on ONE GPU
with a self-communicator (world_size=1 over the nccl/RCCL backend) that:

 * torch.distributed collectives enqueue inside a torch.cuda.graph
   capture without erroring,
 * replay produces the correct reduced values with fresh inputs copied
   into the captured buffers,
 * the TP layer path (ColumnParallel -> RowParallel with its all-reduce)
   captures and replays end to end.

Ref: SURVEY C1 (`reduce_from_tensor_model_parallel_region`,
attention_base.py:1356-1358); the reference compiles CCops into the
graph via neuronx-cc (`--enable-ccop-compute-overlap`).
"""

import os

import pytest
import torch
import torch.distributed as dist

pytestmark = pytest.mark.gpu


def _init_world1():
    if dist.is_initialized():
        return
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29611")
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    dist.init_process_group("nccl", rank=0, world_size=1)


@pytest.fixture(scope="module", autouse=True)
def _teardown_world():
    """Destroy the RCCL communicator before interpreter exit — RCCL's
    atexit teardown after a graph-captured communicator can dump core,
    which would fail the whole GPU test session."""
    yield
    if dist.is_initialized():
        torch.cuda.synchronize()
        dist.destroy_process_group()


def test_allreduce_inside_hipgraph():
    _init_world1()
    dev = torch.device("cuda:0")
    x = torch.ones(64, 4096, dtype=torch.bfloat16, device=dev)
    # warm the communicator outside capture (RCCL lazy-inits channels on
    # first use; first-call init is not capturable)
    dist.all_reduce(x)
    torch.cuda.synchronize()

    static_in = torch.zeros_like(x)
    g = torch.cuda.CUDAGraph()
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        dist.all_reduce(static_in)  # capture warmup on side stream
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()

    with torch.cuda.graph(g):
        dist.all_reduce(static_in)
        out = static_in * 2.0

    for fill in (1.0, 3.0):
        static_in.fill_(fill)
        g.replay()
        torch.cuda.synchronize()
        # world=1 all-reduce is identity; out = 2 * in
        assert torch.all(out == 2.0 * fill), f"fill={fill}"


def test_tp_layers_capture_with_rccl():
    """Column->Row parallel pair (with the row-parallel all-reduce) under
    graph capture, world=1 RCCL."""
    _init_world1()
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.parallel.layers import (
        ColumnParallelLinear,
        RowParallelLinear,
    )

    if not ps.model_parallel_is_initialized():
        ps.initialize_model_parallel(tensor_model_parallel_size=1)
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    col = ColumnParallelLinear(256, 512, dtype=torch.bfloat16).to(dev)
    row = RowParallelLinear(512, 256, dtype=torch.bfloat16).to(dev)
    x = torch.randn(8, 256, dtype=torch.bfloat16, device=dev)

    want = row(col(x))
    torch.cuda.synchronize()

    static_x = x.clone()
    g = torch.cuda.CUDAGraph()
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        _ = row(col(static_x))
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    with torch.cuda.graph(g):
        y = row(col(static_x))
    static_x.copy_(x)
    g.replay()
    torch.cuda.synchronize()
    torch.testing.assert_close(y, want, atol=2e-2, rtol=2e-2)
