#!/bin/bash
# Round-2 step 2: parity-gate the tiered-SPT/flat-dpq kernel, time it on the
# prebuilt 10M f32 config, then 30M builder-stage experiments:
#   (a) NN-descent only (srefine=0)  (b) NN-descent + 2 exact-metric srefine
set -x
REPO=/root/repo
OUT=$REPO/gpurun_out
mkdir -p "$OUT"
cd $REPO

timeout 900 python3 -m pytest tests -m gpu -x -q > "$OUT/s2_pytest.log" 2>&1
rc=$?
tail -3 "$OUT/s2_pytest.log"
echo "pytest rc=$rc" | tee "$OUT/s2_summary.txt"
if [ $rc -ne 0 ]; then echo "PARITY RED - aborting"; exit 1; fi

timeout 500 python3 scripts/prep_profile_index.py bkt_10m_d128_f32_l2 /tmp/idx_f32 \
    > /tmp/prep1.log 2>&1 || tail -3 /tmp/prep1.log
timeout 300 python3 scripts/profile_search.py /tmp/idx_f32 /tmp/idx_f32/queries.npy 2048 2 5 \
    2>&1 | sed 's/^/v6 f32 mc2048 /' | tee -a "$OUT/s2_summary.txt"
SPTAG_AMD_PROF=1 timeout 300 python3 scripts/profile_search.py /tmp/idx_f32 /tmp/idx_f32/queries.npy 2048 1 1 \
    > "$OUT/s2_phase_f32.txt" 2>&1
tail -12 "$OUT/s2_phase_f32.txt" | tee -a "$OUT/s2_summary.txt"

timeout 1200 python3 bench.py --workload bkt_30m_d100_i8_cos --steps 3 --warmup 1 \
    --srefine 0 --no-cpu-baseline > "$OUT/s2_30m_nodescent.log" 2>&1
grep -E "sweep|built|\{" "$OUT/s2_30m_nodescent.log" | tail -8 | tee -a "$OUT/s2_summary.txt"

timeout 1500 python3 bench.py --workload bkt_30m_d100_i8_cos --steps 3 --warmup 1 \
    --srefine 2 --no-cpu-baseline > "$OUT/s2_30m_sref2.log" 2>&1
grep -E "sweep|built|\{" "$OUT/s2_30m_sref2.log" | tail -8 | tee -a "$OUT/s2_summary.txt"
echo done
