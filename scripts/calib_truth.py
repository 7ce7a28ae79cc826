#!/usr/bin/env python3
"""FETCH_SIZE calibration target: runs the brute-force truth kernel over a
prebuilt index — a KNOWN algorithmic read volume (nq x n x dim x esz, the
5.12 GB vector blob streamed once per query, working set far beyond the
256 MB L3) with the same 64 B-granule per-16-lane access shape as the
search kernel's distance loads. rocprofv3 --pmc FETCH_SIZE over this run
gives the gfx950 correction factor for that access shape
(MI355X_MICROARCH.md: FETCH_SIZE is exactly 1/2 of true bytes only for
128 B requests; other widths must be calibrated).

Usage: calib_truth.py <index_dir> <nq>
"""
import sys
import os

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import sptag_amd  # noqa: E402


def main():
    idx_dir, nq = sys.argv[1], int(sys.argv[2])
    ix = sptag_amd.AnnIndex.Load(idx_dir)
    rng = np.random.default_rng(7)
    # query content is irrelevant to the byte count (the read volume is
    # the streamed vector blob)
    if ix.valuetype == 0:
        q = rng.standard_normal((nq, ix.dim)).astype(np.float32)
    else:
        q = rng.integers(-100, 101, (nq, ix.dim)).astype(np.int8)
    v, d = ix.Truth(q, 10)
    esz = 4 if ix.valuetype == 0 else 1
    alg = nq * ix.n * ix.dim * esz
    print(f"calib_truth: nq={nq} n={ix.n} dim={ix.dim} esz={esz} "
          f"algorithmic_read_bytes={alg}")


if __name__ == "__main__":
    main()
