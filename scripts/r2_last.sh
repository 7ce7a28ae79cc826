#!/bin/bash
# Last GPU call of the round, in priority order:
#  1. full GPU parity suite (gates the v7 kernel)
#  2. v7 timing on the driver-default workload
#  3. config-#5 per-shard mechanism check (force-shard, no collectives)
#  4. KDT plateau pin part B (reference CPU searcher on MY 10M x 768 index)
set -x
REPO=/root/repo
OUT=$REPO/gpurun_out
mkdir -p "$OUT"
cd $REPO

timeout 700 python3 -m pytest tests -m gpu -x -q > "$OUT/last_pytest.log" 2>&1
rc=$?
tail -2 "$OUT/last_pytest.log"
echo "pytest rc=$rc" | tee "$OUT/last_summary.txt"

timeout 450 python3 scripts/prep_profile_index.py bkt_10m_d128_f32_l2 /tmp/idx_f32 \
    > /tmp/p1.log 2>&1 || tail -3 /tmp/p1.log
timeout 250 python3 scripts/profile_search.py /tmp/idx_f32 /tmp/idx_f32/queries.npy 2048 2 5 \
    2>&1 | sed 's/^/v7 f32 /' | tee -a "$OUT/last_summary.txt"
SPTAG_AMD_PROF=1 timeout 250 python3 scripts/profile_search.py /tmp/idx_f32 /tmp/idx_f32/queries.npy 2048 0 1 \
    > "$OUT/last_phase_f32.txt" 2>&1
tail -13 "$OUT/last_phase_f32.txt" | tee -a "$OUT/last_summary.txt"

timeout 400 python3 bench.py --workload bkt_30m_d100_i8_cos --force-shard 2/8 \
    --steps 3 --warmup 1 --srefine 0 --refine 1 --no-cpu-baseline \
    > "$OUT/last_shard.log" 2>&1
grep -E "sweep|built|\{" "$OUT/last_shard.log" | tail -6 | tee -a "$OUT/last_summary.txt"

timeout 800 python3 scripts/kdt_pin.py b > "$OUT/f_kdt_pin.log" 2>&1
grep "kdt_pin" "$OUT/f_kdt_pin.log" | tail -8 | tee -a "$OUT/last_summary.txt"
echo done
