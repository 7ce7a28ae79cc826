#!/bin/bash
# rocprofv3 evidence pass for the BKT search kernel (run on the GPU box via
# gpurun). Big trace/csv files stay in /tmp; only the small summaries are
# written to gpurun_out/ for committing under profiles/.
#
# Guide rules (MI355X_MICROARCH.md): run from /tmp with TMPDIR=/tmp; collect
# PMC counters in their own passes, never combined with trace domains.
set -x
cd /tmp && export TMPDIR=/tmp
REPO=/root/repo
OUT=$REPO/gpurun_out
mkdir -p "$OUT"

WORKLOAD=${WORKLOAD:-bkt_10m_d128_f32_l2}
MC=${MC:-2048}

# 1) kernel trace + stats (the per-kernel duration table)
timeout 900 rocprofv3 --kernel-trace --stats -d /tmp/prof_kt -- \
    python3 $REPO/bench.py --workload $WORKLOAD --mc $MC --steps 3 --warmup 1 \
    --no-cpu-baseline > /tmp/prof_kt_run.log 2>&1
tail -4 /tmp/prof_kt_run.log | head -2 > "$OUT/prof_kt_bench.json"
# the stats table is printed at the end of the run log
grep -E "NAME|bkt_search|truth_kernel|gather_rows|KERNEL" /tmp/prof_kt_run.log \
    | head -40 > "$OUT/prof_kt_stats.txt"
cp /tmp/prof_kt_run.log "$OUT/prof_kt_full.log" 2>/dev/null
ls -la /tmp/prof_kt/* >> "$OUT/prof_kt_stats.txt" 2>/dev/null
# kernel stats csv if produced
find /tmp/prof_kt -name "*stats*" -exec grep -l bkt {} \; | while read f; do
    grep -E "Name|bkt_search" "$f" | head -6 >> "$OUT/prof_kt_stats.txt"
done

# 2) PMC pass A: SQ wave-state counters (issue vs parked vs active)
timeout 900 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY \
    -d /tmp/prof_sq -- \
    python3 $REPO/bench.py --workload bkt_1m_d128_f32_l2 --mc $MC --steps 2 \
    --warmup 1 --no-cpu-baseline > /tmp/prof_sq_run.log 2>&1
echo "rc=$?" > "$OUT/prof_sq.txt"
for f in $(find /tmp/prof_sq -name "*.csv"); do
    head -1 "$f" >> "$OUT/prof_sq.txt"
    grep bkt_search "$f" | tail -8 >> "$OUT/prof_sq.txt"
done

# 3) PMC pass B: HBM fetch bytes
timeout 900 rocprofv3 --pmc FETCH_SIZE -d /tmp/prof_tcc -- \
    python3 $REPO/bench.py --workload bkt_1m_d128_f32_l2 --mc $MC --steps 2 \
    --warmup 1 --no-cpu-baseline > /tmp/prof_tcc_run.log 2>&1
echo "rc=$?" > "$OUT/prof_tcc.txt"
for f in $(find /tmp/prof_tcc -name "*.csv"); do
    head -1 "$f" >> "$OUT/prof_tcc.txt"
    grep bkt_search "$f" | tail -8 >> "$OUT/prof_tcc.txt"
done
tail -2 /tmp/prof_sq_run.log >> "$OUT/prof_sq.txt"
tail -2 /tmp/prof_tcc_run.log >> "$OUT/prof_tcc.txt"
du -sh /tmp/prof_* >> "$OUT/prof_sizes.txt"
echo done
