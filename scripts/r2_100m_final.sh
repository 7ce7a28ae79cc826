#!/bin/bash
# Final config-#3 validation: the exact driver-style bench invocation at
# 100M (full pipeline incl. cpu_baseline, saves the index), followed by
# the rocprofv3 evidence pass on the saved index at the chosen MaxCheck.
set -x
REPO=/root/repo
OUT=$REPO/gpurun_out
mkdir -p "$OUT"
cd $REPO
export BENCH_INDEX_DIR=/tmp/b100

timeout 1900 python3 bench.py --workload bkt_100m_d100_i8_cos \
    --steps 20 --warmup 5 > "$OUT/f_100m_bench.log" 2>&1
grep -E "sweep|built|refine|\{" "$OUT/f_100m_bench.log" | tail -14 | tee "$OUT/f_summary.txt"

MC=$(python3 - <<'EOF'
import json
line = None
for l in open("/root/repo/gpurun_out/f_100m_bench.log"):
    if l.startswith('{"metric"'):
        line = l
print(json.loads(line)["config"]["max_check"] if line else 4096)
EOF
)
WORKLOAD=bkt_100m_d100_i8_cos IDX=/tmp/b100 MC=$MC \
    QFILE=/tmp/b100/bench_queries.bin TAG=100m \
    bash $REPO/scripts/r2_evidence.sh 2>&1 | tail -5
echo done
