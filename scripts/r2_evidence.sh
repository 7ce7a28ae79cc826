#!/bin/bash
# Round-2 evidence pass over a PREBUILT index folder (rocprofv3 per the
# guide: PMC passes separate from trace passes; run from /tmp).
#   usage: WORKLOAD=... IDX=... MC=... QFILE=... bash scripts/r2_evidence.sh
set -x
cd /tmp && export TMPDIR=/tmp
REPO=/root/repo
OUT=$REPO/gpurun_out
mkdir -p "$OUT"
WORKLOAD=${WORKLOAD:-bkt_10m_d128_f32_l2}
IDX=${IDX:-/tmp/bench_index}
MC=${MC:-2048}
QFILE=${QFILE:-$IDX/bench_queries.bin}
TAG=${TAG:-$WORKLOAD}

# counter inventory (for the int8 dot / MFMA counter names)
rocprofv3 -L 2>/dev/null | grep -iE "dot|mfma|valu_inst" | head -40 > "$OUT/ev_counters_$TAG.txt"

# 1) kernel trace + stats
timeout 900 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/kt -- \
    python3 $REPO/scripts/profile_search.py $IDX $QFILE $MC 1 3 \
    > /tmp/kt_run.log 2>&1
grep profile_search /tmp/kt_run.log > "$OUT/ev_kt_$TAG.txt"
for f in $(find /tmp/kt -name "*stats*.csv"); do cat "$f" >> "$OUT/ev_kt_$TAG.txt"; done

# 2) SQ wave-state counters
timeout 900 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY \
    --output-format csv -d /tmp/sq -- \
    python3 $REPO/scripts/profile_search.py $IDX $QFILE $MC 0 1 \
    > /tmp/sq_run.log 2>&1
grep profile_search /tmp/sq_run.log > "$OUT/ev_sq_$TAG.txt"
for f in $(find /tmp/sq -name "*.csv"); do
    head -1 "$f" >> "$OUT/ev_sq_$TAG.txt"
    grep -hE "bkt_search|kdt_search" "$f" | tail -8 >> "$OUT/ev_sq_$TAG.txt"
done

# 3) FETCH_SIZE: search pass + known-bytes calibration (truth kernel)
timeout 900 rocprofv3 --pmc FETCH_SIZE --output-format csv -d /tmp/fs -- \
    python3 $REPO/scripts/profile_search.py $IDX $QFILE $MC 0 1 \
    > /tmp/fs_run.log 2>&1
grep profile_search /tmp/fs_run.log > "$OUT/ev_fetch_$TAG.txt"
for f in $(find /tmp/fs -name "*.csv"); do
    head -1 "$f" >> "$OUT/ev_fetch_$TAG.txt"
    grep -hE "bkt_search|kdt_search" "$f" | tail -8 >> "$OUT/ev_fetch_$TAG.txt"
done
timeout 900 rocprofv3 --pmc FETCH_SIZE --output-format csv -d /tmp/fc -- \
    python3 $REPO/scripts/calib_truth.py $IDX 2 \
    > /tmp/fc_run.log 2>&1
grep calib_truth /tmp/fc_run.log > "$OUT/ev_fetchcalib_$TAG.txt"
for f in $(find /tmp/fc -name "*.csv"); do
    head -1 "$f" >> "$OUT/ev_fetchcalib_$TAG.txt"
    grep -h "truth_kernel" "$f" >> "$OUT/ev_fetchcalib_$TAG.txt"
done

# 4) WRITE_SIZE on the search
timeout 900 rocprofv3 --pmc WRITE_SIZE --output-format csv -d /tmp/ws -- \
    python3 $REPO/scripts/profile_search.py $IDX $QFILE $MC 0 1 \
    > /tmp/ws_run.log 2>&1
for f in $(find /tmp/ws -name "*.csv"); do
    head -1 "$f" >> "$OUT/ev_write_$TAG.txt"
    grep -hE "bkt_search|kdt_search" "$f" | tail -8 >> "$OUT/ev_write_$TAG.txt"
done

# 5) phase breakdown (PROF kernel)
SPTAG_AMD_PROF=1 timeout 600 python3 $REPO/scripts/profile_search.py \
    $IDX $QFILE $MC 0 1 > "$OUT/ev_phase_$TAG.txt" 2>&1
tail -16 "$OUT/ev_phase_$TAG.txt"
echo done
