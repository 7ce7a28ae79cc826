#!/bin/bash
# rocprofv3 evidence pass for the BKT search kernel (run on the GPU box via
# gpurun). The index is built once unprofiled; the profiled process only
# loads + searches, so the trace holds search kernels only.
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
IDX=/tmp/profidx

timeout 600 python3 $REPO/scripts/prep_profile_index.py $WORKLOAD $IDX \
    > /tmp/prep.log 2>&1 || { tail -5 /tmp/prep.log; exit 1; }

# 1) kernel trace + stats
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/kt -- \
    python3 $REPO/scripts/profile_search.py $IDX $IDX/queries.npy $MC 1 3 \
    > /tmp/kt_run.log 2>&1
grep profile_search /tmp/kt_run.log > "$OUT/prof_kt_stats.txt"
for f in $(find /tmp/kt -name "*stats*.csv"); do
    cat "$f" >> "$OUT/prof_kt_stats.txt"
done

# 2) PMC pass A: SQ wave-state counters
timeout 600 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY \
    --output-format csv -d /tmp/sq -- \
    python3 $REPO/scripts/profile_search.py $IDX $IDX/queries.npy $MC 0 1 \
    > /tmp/sq_run.log 2>&1
grep profile_search /tmp/sq_run.log > "$OUT/prof_sq.txt"
for f in $(find /tmp/sq -name "*.csv"); do
    head -1 "$f" >> "$OUT/prof_sq.txt"
    grep -h bkt_search "$f" | tail -12 >> "$OUT/prof_sq.txt"
done

# 3) PMC pass B: HBM fetch bytes
timeout 600 rocprofv3 --pmc FETCH_SIZE --output-format csv -d /tmp/tcc -- \
    python3 $REPO/scripts/profile_search.py $IDX $IDX/queries.npy $MC 0 1 \
    > /tmp/tcc_run.log 2>&1
grep profile_search /tmp/tcc_run.log > "$OUT/prof_tcc.txt"
for f in $(find /tmp/tcc -name "*.csv"); do
    head -1 "$f" >> "$OUT/prof_tcc.txt"
    grep -h bkt_search "$f" | tail -12 >> "$OUT/prof_tcc.txt"
done
tail -3 /tmp/kt_run.log /tmp/sq_run.log /tmp/tcc_run.log > "$OUT/prof_logs_tail.txt"
echo done
