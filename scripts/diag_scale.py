#!/usr/bin/env python3
"""Scale diagnosis: build the int8-cosine workload at a given n, then
compare GPU search vs the ORACLE (reference-exact CPU) on the SAME index
arrays. Separates kernel-at-scale bugs from index-quality-at-scale bugs.
Usage: diag_scale.py <n>"""
import sys
import os
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import bench
import sptag_amd
from sptag_amd.build import build_index_arrays
from oracle.pyoracle import OrcIndex


def main():
    n = int(sys.argv[1])
    ntrees = int(sys.argv[2]) if len(sys.argv) > 2 else 2
    refine = int(sys.argv[3]) if len(sys.argv) > 3 else 0
    cand = int(sys.argv[4]) if len(sys.argv) > 4 else 96
    srefine = int(sys.argv[5]) if len(sys.argv) > 5 else 0
    cfg = dict(bench.CONFIGS["bkt_100m_d100_i8_cos"])
    cfg["n"] = n
    x, q, lo = bench.gen_data(cfg, 0, 1, "cuda:0", torch)
    t0 = time.time()
    arrays = build_index_arrays(x.cpu().numpy(), "Cosine", cand=cand,
                                ntrees=ntrees, refine_rounds=refine,
                                search_refine_rounds=srefine,
                                device="cuda:0")
    print(f"built {n} in {time.time()-t0:.0f}s", flush=True)
    torch.cuda.empty_cache()

    nq = 300
    qs = q[:nq]
    xs = torch.as_tensor(arrays["vectors"], device="cuda:0")
    tv, _ = bench.shard_truth(xs, qs, 10, "Cosine", torch, 0)
    tv = tv.cpu().numpy()

    gix = sptag_amd.AnnIndex.FromArrays(arrays["vectors"], arrays["tree_start"],
                                        arrays["tree_nodes"], arrays["graph"],
                                        "Cosine")
    q_np = qs.cpu().numpy()
    for mc in [4096, 16384]:
        gv, _ = gix.BatchSearch(q_np, 10, mc)
        print(f"GPU    mc={mc}: recall={bench.recall_at_k(gv, tv, 10):.4f}",
              flush=True)

    if os.environ.get("DIAG_ORACLE") != "1":
        return   # kernel==oracle was established (agreement 1.000 at 30M)
    oix = OrcIndex.from_arrays(arrays["vectors"], arrays["tree_start"],
                               arrays["tree_nodes"], arrays["graph"], "Cosine")
    for mc in [4096, 16384]:
        ov, _ = oix.search_batch(q_np, 10, mc, nthreads=0)
        agree = (ov == gv).mean() if mc == 16384 else None
        print(f"ORACLE mc={mc}: recall={bench.recall_at_k(ov, tv, 10):.4f}"
              + (f" (gpu-agreement {agree:.3f})" if agree is not None else ""),
              flush=True)


if __name__ == "__main__":
    main()
