#!/bin/bash
# Step 4: (A) 10M int8 new-data ceiling check (descent only)
#         (B) 30M descent + deep wide-prune search-refine (k=512, mc=4096)
set -x
REPO=/root/repo
OUT=$REPO/gpurun_out
mkdir -p "$OUT"
cd $REPO

timeout 700 python3 bench.py --workload bkt_10m_d100_i8_cos --steps 3 --warmup 1 \
    --refine 2 --srefine 0 --no-cpu-baseline > "$OUT/s4_10m_descent.log" 2>&1
grep -E "sweep|built|\{" "$OUT/s4_10m_descent.log" | tail -9 | tee "$OUT/s4_summary.txt"

timeout 1800 python3 bench.py --workload bkt_30m_d100_i8_cos --steps 3 --warmup 1 \
    --srefine 1 --srefine-k 512 --srefine-mc 4096 --no-cpu-baseline \
    > "$OUT/s4_30m_sref.log" 2>&1
grep -E "sweep|built|search-refine round 1/1|\{" "$OUT/s4_30m_sref.log" | tail -10 | tee -a "$OUT/s4_summary.txt"
echo done
