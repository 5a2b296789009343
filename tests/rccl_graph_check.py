"""RCCL-inside-hipGraph pre-flight, run as a SUBPROCESS by
test_rccl_graph_gpu.py: RCCL's interpreter-exit teardown after
graph-captured communicators can core-dump (observed when other CUDA
test modules ran first), which would kill a whole pytest session. The
child prints RCCL_GRAPH_CHECK_OK and `os._exit(0)`s before any atexit
teardown runs."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402


def main():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29611")
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    dist.init_process_group("nccl", rank=0, world_size=1)
    dev = torch.device("cuda:0")

    # --- all-reduce inside a captured graph, replayed with fresh data ---
    x = torch.ones(64, 4096, dtype=torch.bfloat16, device=dev)
    dist.all_reduce(x)  # warm the communicator (lazy channel init)
    torch.cuda.synchronize()
    static_in = torch.zeros_like(x)
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        dist.all_reduce(static_in)
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        dist.all_reduce(static_in)
        out = static_in * 2.0
    for fill in (1.0, 3.0):
        static_in.fill_(fill)
        g.replay()
        torch.cuda.synchronize()
        assert torch.all(out == 2.0 * fill), f"fill={fill}"

    # --- TP layer pair (col -> row with its all-reduce) under capture ---
    from neuronx_distributed_inference_amd.parallel import state as ps
    from neuronx_distributed_inference_amd.parallel.layers import (
        ColumnParallelLinear,
        RowParallelLinear,
    )

    if not ps.model_parallel_is_initialized():
        ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(0)
    col = ColumnParallelLinear(256, 512, dtype=torch.bfloat16).to(dev)
    row = RowParallelLinear(512, 256, dtype=torch.bfloat16).to(dev)
    with torch.no_grad():  # parallel layers allocate torch.empty weights
        col.weight.normal_(0, 0.05)
        row.weight.normal_(0, 0.05)
    xin = torch.randn(8, 256, dtype=torch.bfloat16, device=dev)
    want = row(col(xin))
    torch.cuda.synchronize()
    static_x = xin.clone()
    s2 = torch.cuda.Stream()
    s2.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s2):
        _ = row(col(static_x))
    torch.cuda.current_stream().wait_stream(s2)
    torch.cuda.synchronize()
    g2 = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g2):
        y = row(col(static_x))
    static_x.copy_(xin)
    g2.replay()
    torch.cuda.synchronize()
    torch.testing.assert_close(y, want, atol=2e-2, rtol=2e-2)

    print("RCCL_GRAPH_CHECK_OK", flush=True)
    # skip interpreter teardown entirely (graph-captured RCCL
    # communicators can crash in atexit teardown)
    os._exit(0)


if __name__ == "__main__":
    main()
