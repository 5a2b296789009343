#!/bin/bash
# ksplit sweep for skinny_gemm2: quantifies the atomic-partial write-traffic
# term found by the TCC probe (write EA requests scale with ksplit).
set -u
R=${GRAFT_REPO_ROOT:-/root/repo}
O=$R/gpurun_out/tcc
mkdir -p "$O"
cd /tmp && export TMPDIR=/tmp
for ks in 1 2 4 8 16 32; do
  timeout 150 rocprofv3 --output-format csv --kernel-trace --stats -d "$O/ks$ks" -o s \
    -- python "$R/scripts/prof_skinny_tcc.py" skinny2$ks > "$O/ks$ks.log" 2>&1 || true
  grep -m1 "skinny_gemm2" "$O/ks$ks/s_kernel_stats.csv" 2>/dev/null | head -1
done
find "$O" -name '*.db' -delete 2>/dev/null
echo SWEEP-DONE
