"""Profiling helpers.

Analog of the reference's `utils/profiling.py` (:34-122, which shells out to
`neuron-profile capture/view`): wraps rocprofv3 kernel-trace/stats capture
around a command or generates the command line, and parses the kernel-stats
CSV into the summary dict committed under profiles/.

Note the pool rule encoded here: PMC counter collection must run in its own
invocation — rocprofv3 refuses --pmc combined with the trace domains.
"""

from __future__ import annotations

import csv
import glob
import os
import shlex
import subprocess
import sys
from typing import Dict, List, Optional


def rocprof_command(
    cmd: List[str],
    output_dir: str,
    pmc: Optional[List[str]] = None,
    output_format: str = "csv",
) -> List[str]:
    base = ["rocprofv3"]
    if pmc:
        base += ["--pmc"] + list(pmc)          # counters: own run, no traces
    else:
        base += ["--kernel-trace", "--stats"]
    base += ["--output-format", output_format, "-d", output_dir, "--"]
    return base + cmd


def capture(cmd: List[str], output_dir: str, pmc: Optional[List[str]] = None,
            cwd: str = "/tmp") -> int:
    env = dict(os.environ, TMPDIR="/tmp")
    full = rocprof_command(cmd, output_dir, pmc=pmc)
    print("+", " ".join(shlex.quote(c) for c in full), file=sys.stderr)
    return subprocess.call(full, cwd=cwd, env=env)


def parse_kernel_stats(output_dir: str, top: int = 25) -> List[Dict]:
    """Read the *_kernel_stats.csv rocprofv3 wrote, sorted by total time."""
    files = glob.glob(os.path.join(output_dir, "**", "*_kernel_stats.csv"),
                      recursive=True)
    if not files:
        return []
    rows = list(csv.DictReader(open(sorted(files)[-1])))
    rows.sort(key=lambda r: -float(r.get("TotalDurationNs", 0)))
    out = []
    for r in rows[:top]:
        out.append({
            "name": r["Name"],
            "calls": int(r["Calls"]),
            "total_us": float(r["TotalDurationNs"]) / 1e3,
            "avg_us": float(r["AverageNs"]) / 1e3,
            "pct": float(r["Percentage"]),
        })
    return out


def write_summary_markdown(stats: List[Dict], path: str, title: str):
    with open(path, "w") as f:
        f.write(f"# {title}\n\n| kernel | calls | total_us | avg_us | % |\n")
        f.write("|---|---|---|---|---|\n")
        for s in stats:
            f.write(f'| {s["name"][:80]} | {s["calls"]} | {s["total_us"]:.0f} '
                    f'| {s["avg_us"]:.2f} | {s["pct"]:.2f} |\n')
