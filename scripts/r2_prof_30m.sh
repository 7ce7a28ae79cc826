#!/bin/bash
# 1) per-phase cycle breakdown of the BKT search kernel (PROF variant)
# 2) 30M int8 recipe validation (NN-descent + search-refine)
set -x
REPO=/root/repo
OUT=$REPO/gpurun_out
mkdir -p "$OUT"
cd $REPO

timeout 400 python3 scripts/prep_profile_index.py bkt_10m_d128_f32_l2 /tmp/idx_f32 \
    > /tmp/prep1.log 2>&1 || tail -3 /tmp/prep1.log
SPTAG_AMD_PROF=1 timeout 300 python3 scripts/profile_search.py \
    /tmp/idx_f32 /tmp/idx_f32/queries.npy 2048 1 1 \
    > "$OUT/prof_phase_f32.txt" 2>&1
tail -15 "$OUT/prof_phase_f32.txt"

timeout 400 python3 scripts/prep_profile_index.py bkt_10m_d100_i8_cos /tmp/idx_i8 \
    > /tmp/prep2.log 2>&1 || tail -3 /tmp/prep2.log
SPTAG_AMD_PROF=1 timeout 300 python3 scripts/profile_search.py \
    /tmp/idx_i8 /tmp/idx_i8/queries.npy 4096 1 1 \
    > "$OUT/prof_phase_i8.txt" 2>&1
tail -15 "$OUT/prof_phase_i8.txt"

timeout 1100 python3 bench.py --workload bkt_30m_d100_i8_cos --steps 5 \
    --warmup 2 --no-cpu-baseline > "$OUT/r2_30m.log" 2>&1
tail -25 "$OUT/r2_30m.log"
echo done
