#!/bin/bash
# TCC counter probe for the skinny-GEMM 1 TB/s plateau (run on the GPU box).
# Counter names are validated against `rocprofv3 --list-avail` before use so a
# bad config never reaches the hardware (a rejected config hangs rocprofv3).
set -u
R=${GRAFT_REPO_ROOT:-/root/repo}
O=$R/gpurun_out/tcc
mkdir -p "$O"
cd /tmp && export TMPDIR=/tmp

rocprofv3 --list-avail > "$O/avail.txt" 2>&1 || true
pick() { grep -qw "$1" "$O/avail.txt" && printf '%s ' "$1"; }
P1=$(pick TCC_EA0_RDREQ_sum; pick TCC_HIT_sum; pick TCC_MISS_sum)
P2=$(pick TCC_EA0_RDREQ_32B_sum; pick TCC_EA0_WRREQ_sum)
P3=$(pick SQ_BUSY_CYCLES; pick SQ_WAVE_CYCLES; pick SQ_INSTS_VALU)
echo "P1=[$P1] P2=[$P2] P3=[$P3]" | tee "$O/picked.txt"

for m in skinny skinny2 lib lt; do
  timeout 150 rocprofv3 --output-format csv --kernel-trace --stats -d "$O/stats_$m" -o s \
    -- python "$R/scripts/prof_skinny_tcc.py" $m > "$O/stats_$m.log" 2>&1 || true
done
for m in skinny2 lib; do
  [ -n "$P1" ] && timeout 150 rocprofv3 --output-format csv --pmc $P1 -d "$O/p1_$m" -o p \
    -- python "$R/scripts/prof_skinny_tcc.py" $m > "$O/p1_$m.log" 2>&1 || true
  [ -n "$P2" ] && timeout 150 rocprofv3 --output-format csv --pmc $P2 -d "$O/p2_$m" -o p \
    -- python "$R/scripts/prof_skinny_tcc.py" $m > "$O/p2_$m.log" 2>&1 || true
done
[ -n "$P3" ] && timeout 150 rocprofv3 --output-format csv --pmc $P3 -d "$O/p3_skinny2" -o p \
  -- python "$R/scripts/prof_skinny_tcc.py" skinny2 > "$O/p3_skinny2.log" 2>&1 || true

# compact: keep only csv summaries, drop anything big
find "$O" -name '*.db' -delete 2>/dev/null
echo PROBE-DONE
