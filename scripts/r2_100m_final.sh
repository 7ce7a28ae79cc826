#!/bin/bash
# Final config-#3 validation: the exact driver-style bench invocation at
# 100M (full pipeline incl. cpu_baseline, saves the index), then a trimmed
# rocprofv3 evidence pass on the saved index at the chosen MaxCheck, then
# (time permitting) the KDT plateau pin.
set -x
REPO=/root/repo
OUT=$REPO/gpurun_out
mkdir -p "$OUT"
cd $REPO
export BENCH_INDEX_DIR=/tmp/b100

timeout 1750 python3 bench.py --workload bkt_100m_d100_i8_cos \
    --steps 20 --warmup 5 > "$OUT/f_100m_bench.log" 2>&1
grep -E "sweep|built|uploaded|truth|baseline|\{" "$OUT/f_100m_bench.log" | tail -16 | tee "$OUT/f_summary.txt"

MC=$(python3 - <<'EOF'
import json
line = None
for l in open("/root/repo/gpurun_out/f_100m_bench.log"):
    if l.startswith('{"metric"'):
        line = l
print(json.loads(line)["config"]["max_check"] if line else 4096)
EOF
)

cd /tmp && export TMPDIR=/tmp
IDX=/tmp/b100
QF=$IDX/bench_queries.bin

rocprofv3 -L 2>/dev/null | grep -iE "dot|mfma|valu_inst" | head -40 > "$OUT/ev_counters.txt"

timeout 700 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/kt -- \
    python3 $REPO/scripts/profile_search.py $IDX $QF $MC 1 3 > /tmp/kt.log 2>&1
grep profile_search /tmp/kt.log > "$OUT/ev_kt_100m.txt"
for f in $(find /tmp/kt -name "*stats*.csv"); do cat "$f" >> "$OUT/ev_kt_100m.txt"; done

timeout 700 rocprofv3 --pmc FETCH_SIZE --output-format csv -d /tmp/fs -- \
    python3 $REPO/scripts/profile_search.py $IDX $QF $MC 0 1 > /tmp/fs.log 2>&1
grep profile_search /tmp/fs.log > "$OUT/ev_fetch_100m.txt"
for f in $(find /tmp/fs -name "*.csv"); do
    head -1 "$f" >> "$OUT/ev_fetch_100m.txt"
    grep -h "bkt_search" "$f" | tail -6 >> "$OUT/ev_fetch_100m.txt"
done

timeout 700 rocprofv3 --pmc FETCH_SIZE --output-format csv -d /tmp/fc -- \
    python3 $REPO/scripts/calib_truth.py $IDX 2 > /tmp/fc.log 2>&1
grep calib_truth /tmp/fc.log > "$OUT/ev_fetchcalib_100m.txt"
for f in $(find /tmp/fc -name "*.csv"); do
    head -1 "$f" >> "$OUT/ev_fetchcalib_100m.txt"
    grep -h "truth_kernel" "$f" >> "$OUT/ev_fetchcalib_100m.txt"
done

SPTAG_AMD_PROF=1 timeout 500 python3 $REPO/scripts/profile_search.py \
    $IDX $QF $MC 0 1 > "$OUT/ev_phase_100m.txt" 2>&1
tail -16 "$OUT/ev_phase_100m.txt"

# KDT plateau pin (best-effort tail; outputs stream to gpurun_out)
cd $REPO
timeout 900 python3 scripts/kdt_pin.py b > "$OUT/f_kdt_pin.log" 2>&1
grep "kdt_pin" "$OUT/f_kdt_pin.log" | tail -8
echo done
