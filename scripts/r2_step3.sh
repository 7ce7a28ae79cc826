#!/bin/bash
# Round-2 step 3: parity gate (seed-warming kernel), then 30M recipe
# experiments on the FIXED int8 data:
#   (c) NN-descent only        (d) NN-descent + wide-prune search-refine
set -x
REPO=/root/repo
OUT=$REPO/gpurun_out
mkdir -p "$OUT"
cd $REPO

timeout 900 python3 -m pytest tests -m gpu -x -q > "$OUT/s3_pytest.log" 2>&1
rc=$?
tail -3 "$OUT/s3_pytest.log"
echo "pytest rc=$rc" | tee "$OUT/s3_summary.txt"
if [ $rc -ne 0 ]; then echo "PARITY RED - aborting"; exit 1; fi

timeout 1100 python3 bench.py --workload bkt_30m_d100_i8_cos --steps 3 --warmup 1 \
    --srefine 0 --no-cpu-baseline > "$OUT/s3_30m_descent.log" 2>&1
grep -E "sweep|built|\{" "$OUT/s3_30m_descent.log" | tail -9 | tee -a "$OUT/s3_summary.txt"

timeout 1500 python3 bench.py --workload bkt_30m_d100_i8_cos --steps 3 --warmup 1 \
    --no-cpu-baseline > "$OUT/s3_30m_sref.log" 2>&1
grep -E "sweep|built|\{" "$OUT/s3_30m_sref.log" | tail -9 | tee -a "$OUT/s3_summary.txt"
echo done
