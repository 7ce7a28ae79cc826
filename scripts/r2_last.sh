#!/bin/bash
# Last GPU call of the round: KDT plateau pin (part B: reference CPU
# searcher on MY 10M x 768 index) + config-#5 per-shard mechanism check.
set -x
REPO=/root/repo
OUT=$REPO/gpurun_out
mkdir -p "$OUT"
cd $REPO

# shard mechanism: shard 2 of 8 of the 30M int8 workload on one process
# (no collectives; shard-local recall vs shard-local truth) — the code
# path the driver's --gpus 8 SCALE run and config #5 use per rank.
timeout 420 python3 bench.py --workload bkt_30m_d100_i8_cos --force-shard 2/8 \
    --steps 3 --warmup 1 --srefine 0 --refine 1 --no-cpu-baseline \
    > "$OUT/last_shard.log" 2>&1
grep -E "sweep|built|\{" "$OUT/last_shard.log" | tail -6 | tee "$OUT/last_summary.txt"

timeout 1100 python3 scripts/kdt_pin.py b > "$OUT/f_kdt_pin.log" 2>&1
grep "kdt_pin" "$OUT/f_kdt_pin.log" | tail -8 | tee -a "$OUT/last_summary.txt"
echo done
