"""Multi-node / multi-GPU launcher.

Analog of the reference's `scripts/nxdi_distributed_launcher.py` (:29-155):
builds the torchrun command for one-process-per-GPU RCCL execution, forwards
RCCL/xGMI env, and supports multi-node rank windows via --node-rank /
--nnodes (ref examples/multi_node.md start_rank_id scheme).

    python scripts/nxdi_distributed_launcher.py --nproc-per-node 8 -- \
        inference_demo --model-type llama run ...
"""

from __future__ import annotations

import argparse
import os
import shlex
import subprocess
import sys

FORWARD_ENV_PREFIXES = ("NCCL_", "RCCL_", "HSA_", "HIP_", "ROCR_", "NXDI_", "GLOO_")


def build_torchrun_command(args, cmd):
    out = [
        sys.executable, "-m", "torch.distributed.run",
        f"--nnodes={args.nnodes}",
        f"--nproc-per-node={args.nproc_per_node}",
        f"--node-rank={args.node_rank}",
        f"--master-addr={args.master_addr}",
        f"--master-port={args.master_port}",
    ]
    return out + cmd


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--nproc-per-node", type=int, default=8)
    p.add_argument("--nnodes", type=int, default=1)
    p.add_argument("--node-rank", type=int, default=0)
    p.add_argument("--master-addr", default="127.0.0.1")
    p.add_argument("--master-port", type=int, default=29510)
    p.add_argument("cmd", nargs=argparse.REMAINDER,
                   help="command after `--` (module or script + args)")
    args = p.parse_args(argv)
    cmd = args.cmd
    if cmd and cmd[0] == "--":
        cmd = cmd[1:]
    if not cmd:
        p.error("no command given")
    if cmd[0] == "inference_demo":
        cmd = ["-m", "neuronx_distributed_inference_amd.inference_demo"] + cmd[1:]
    env = dict(os.environ)
    # dmabuf IPC is required for RCCL cross-process sharing on this stack
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    full = build_torchrun_command(args, cmd)
    print("+", " ".join(shlex.quote(c) for c in full), file=sys.stderr)
    return subprocess.call(full, env=env)


if __name__ == "__main__":
    sys.exit(main())
