#!/bin/bash
# Round-2 kernel v5 A/B on one MI355X: parity gate first, then per-flag
# timing on prebuilt 10M indexes (f32 L2 and int8 cosine).
set -x
REPO=/root/repo
OUT=$REPO/gpurun_out
mkdir -p "$OUT"
cd $REPO

timeout 900 python3 -m pytest tests -m gpu -x -q > "$OUT/ab_pytest.log" 2>&1
echo "pytest rc=$?" | tee "$OUT/ab_v5.txt"

prep() {
    timeout 900 python3 scripts/prep_profile_index.py "$1" "$2" \
        > /tmp/prep_$1.log 2>&1 || tail -5 /tmp/prep_$1.log
}

run_ab() {  # $1=idx dir, $2=mc
    for spec in 0 1 2 3; do
        SPTAG_AMD_SPEC=$spec timeout 600 python3 scripts/profile_search.py \
            "$1" "$1/queries.npy" "$2" 2 5 2>&1 | \
            sed "s/^/spec=$spec /" | tee -a "$OUT/ab_v5.txt"
    done
    # occupancy experiment: smaller LDS frontier heap (more waves/CU)
    for cap in 768 1024 1280; do
        SPTAG_AMD_SPEC=3 SPTAG_AMD_NG_CAP=$cap timeout 600 \
            python3 scripts/profile_search.py "$1" "$1/queries.npy" "$2" 2 5 \
            2>&1 | sed "s/^/spec=3 cap=$cap /" | tee -a "$OUT/ab_v5.txt"
    done
}

prep bkt_10m_d128_f32_l2 /tmp/idx_f32
echo "== f32 L2 10M mc=2048 ==" | tee -a "$OUT/ab_v5.txt"
run_ab /tmp/idx_f32 2048

prep bkt_10m_d100_i8_cos /tmp/idx_i8
echo "== int8 cos 10M mc=4096 ==" | tee -a "$OUT/ab_v5.txt"
for spec in 0 1 3; do
    SPTAG_AMD_SPEC=$spec timeout 600 python3 scripts/profile_search.py \
        /tmp/idx_i8 /tmp/idx_i8/queries.npy 4096 2 5 2>&1 | \
        sed "s/^/i8 spec=$spec /" | tee -a "$OUT/ab_v5.txt"
done
echo done
