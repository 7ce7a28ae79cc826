#!/usr/bin/env python3
"""Minimal search-only harness for rocprofv3 passes: loads a prebuilt index
folder (no torch, no build kernels in the trace) and runs warmup+steps of
the batched search. Usage:
    profile_search.py <index_dir> <queries.npy> <mc> <warmup> <steps>
"""
import sys
import os
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import sptag_amd  # noqa: E402


def main():
    idx_dir, qfile, mc, warmup, steps = sys.argv[1:6]
    mc, warmup, steps = int(mc), int(warmup), int(steps)
    if qfile.endswith(".npy"):
        q = np.load(qfile)
    else:   # reference DEFAULT binary format [int32 n][int32 d][rows]
        with open(qfile, "rb") as f:
            n, d = np.frombuffer(f.read(8), dtype=np.int32)
            import sptag_amd as _s
            ix0 = _s.AnnIndex.Load(idx_dir)
            dt = np.float32 if ix0.valuetype == 0 else np.int8
            del ix0
            q = np.frombuffer(f.read(), dtype=dt).reshape(n, d)
    ix = sptag_amd.AnnIndex.Load(idx_dir)
    k = 10
    for _ in range(warmup):
        ix.BatchSearch(q, k, mc)
    t0 = time.time()
    for _ in range(steps):
        v, d = ix.BatchSearch(q, k, mc)
    dt = time.time() - t0
    ms, checked, popped = ix.LastStats()
    print(f"profile_search: {steps} steps, {dt/steps*1e3:.2f} ms/step wall, "
          f"{ms:.2f} ms kernel, checked={checked} popped={popped} "
          f"qps={q.shape[0]*steps/dt:.0f}")


if __name__ == "__main__":
    main()
