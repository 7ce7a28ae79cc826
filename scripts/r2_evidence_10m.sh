#!/bin/bash
# Evidence pass for the driver-default workload (10M f32 L2, mc 2048) and
# the int8 path (10M i8 cosine, mc 2048): kernel stats, SQ counters,
# FETCH_SIZE + truth-kernel calibration, phase breakdown, counter list.
set -x
REPO=/root/repo
OUT=$REPO/gpurun_out
mkdir -p "$OUT"
cd $REPO

timeout 500 python3 scripts/prep_profile_index.py bkt_10m_d128_f32_l2 /tmp/idx_f32 \
    > /tmp/p1.log 2>&1 || tail -3 /tmp/p1.log
timeout 500 python3 scripts/prep_profile_index.py bkt_10m_d100_i8_cos /tmp/idx_i8 \
    > /tmp/p2.log 2>&1 || tail -3 /tmp/p2.log

WORKLOAD=bkt_10m_d128_f32_l2 IDX=/tmp/idx_f32 MC=2048 \
    QFILE=/tmp/idx_f32/queries.npy TAG=10m_f32 \
    bash $REPO/scripts/r2_evidence.sh 2>&1 | tail -3

# int8: kernel stats + FETCH + dot/MFMA-relevant SQ instruction counters
cd /tmp && export TMPDIR=/tmp
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/ki -- \
    python3 $REPO/scripts/profile_search.py /tmp/idx_i8 /tmp/idx_i8/queries.npy 2048 1 3 \
    > /tmp/ki.log 2>&1
grep profile_search /tmp/ki.log > "$OUT/ev_kt_10m_i8.txt"
for f in $(find /tmp/ki -name "*stats*.csv"); do cat "$f" >> "$OUT/ev_kt_10m_i8.txt"; done
timeout 500 rocprofv3 --pmc FETCH_SIZE --output-format csv -d /tmp/fi -- \
    python3 $REPO/scripts/profile_search.py /tmp/idx_i8 /tmp/idx_i8/queries.npy 2048 0 1 \
    > /tmp/fi.log 2>&1
grep profile_search /tmp/fi.log > "$OUT/ev_fetch_10m_i8.txt"
for f in $(find /tmp/fi -name "*.csv"); do
    head -1 "$f" >> "$OUT/ev_fetch_10m_i8.txt"
    grep -h "bkt_search" "$f" | tail -6 >> "$OUT/ev_fetch_10m_i8.txt"
done
# VALU instruction-mix counters on the int8 search (dot evidence)
DOTC=$(rocprofv3 -L 2>/dev/null | grep -oE "SQ_INSTS_VALU[A-Z_0-9]*" | head -6 | tr '\n' ' ')
echo "counters: $DOTC" > "$OUT/ev_dot_10m_i8.txt"
if [ -n "$DOTC" ]; then
    timeout 500 rocprofv3 --pmc $DOTC --output-format csv -d /tmp/dc -- \
        python3 $REPO/scripts/profile_search.py /tmp/idx_i8 /tmp/idx_i8/queries.npy 2048 0 1 \
        > /tmp/dc.log 2>&1
    for f in $(find /tmp/dc -name "*.csv"); do
        head -1 "$f" >> "$OUT/ev_dot_10m_i8.txt"
        grep -h "bkt_search" "$f" | tail -6 >> "$OUT/ev_dot_10m_i8.txt"
    done
fi
echo done
