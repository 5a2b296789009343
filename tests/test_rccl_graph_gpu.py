"""RCCL-inside-hipGraph pre-flight.

Multi-GPU decode graphs capture RCCL collectives (all-reduce after every
row-parallel GEMM, 2x/layer). Before any 8-GPU run, verify on ONE GPU
with a self-communicator (world_size=1 over the nccl/RCCL backend) that
torch.distributed collectives capture into a torch.cuda.graph, replay
with correct values, and that the TP layer path (ColumnParallel ->
RowParallel with its all-reduce) captures end to end.

Ref: SURVEY C1 (`reduce_from_tensor_model_parallel_region`,
attention_base.py:1356-1358); the reference compiles CCops into the
graph via neuronx-cc (`--enable-ccop-compute-overlap`).

The checks run in a SUBPROCESS (tests/rccl_graph_check.py) because
RCCL's interpreter-exit teardown after graph-captured communicators can
core-dump and would take the whole pytest session with it; the child
`os._exit(0)`s after printing its verdict.
"""

import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu


def test_rccl_collectives_inside_hipgraph():
    script = os.path.join(os.path.dirname(__file__), "rccl_graph_check.py")
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    r = subprocess.run([sys.executable, script], capture_output=True,
                       text=True, timeout=300, env=env)
    assert "RCCL_GRAPH_CHECK_OK" in r.stdout, (
        f"rc={r.returncode}\nstdout:\n{r.stdout[-2000:]}\n"
        f"stderr:\n{r.stderr[-2000:]}")
