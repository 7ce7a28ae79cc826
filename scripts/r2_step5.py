#!/usr/bin/env python3
"""30M recipe close-out: build once (descent-2 + srefine-1 at k=512,
mc=2048 — the cheaper refine), then sweep NumberOfInitialDynamicPivots x
MaxCheck on the SAME index to find the cheapest setting reaching
recall@10 >= 0.95. Seed pivots are a reference search parameter
(ParameterDefinitionList.h), saved into the ini, so the CPU baseline
comparison stays apples-to-apples."""
import os
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import bench  # noqa: E402
import sptag_amd  # noqa: E402
from sptag_amd.build import build_index_arrays  # noqa: E402


def main():
    cfg = dict(bench.CONFIGS["bkt_30m_d100_i8_cos"])
    x, q, lo = bench.gen_data(cfg, 0, 1, "cuda:0", torch)
    t0 = time.time()
    arrays = build_index_arrays(
        x.cpu().numpy(), cfg["metric"], cand=cfg["cand"],
        ntrees=cfg["ntrees"], refine_rounds=cfg["refine"],
        search_refine_rounds=1, srefine_k=512, srefine_mc=2048,
        fill_pruned=True, device="cuda:0", verbose=True)
    print(f"[s5] build {time.time()-t0:.0f}s", flush=True)
    torch.cuda.empty_cache()
    ix = sptag_amd.AnnIndex.FromArrays(
        arrays["vectors"], arrays["tree_start"], arrays["tree_nodes"],
        arrays["graph"], cfg["metric"])
    xs = torch.as_tensor(arrays["vectors"], device="cuda:0")
    del arrays
    torch.cuda.empty_cache()
    tv, _ = bench.shard_truth(xs, q, cfg["k"], cfg["metric"], torch, 0)
    tv = tv.cpu().numpy()
    q_np = q.cpu().numpy()
    for pivots in (50, 128, 256, 512):
        ix.SetSearchParams(init_pivots=pivots)
        for mc in (2048, 4096, 8192, 16384):
            t0 = time.time()
            v, _ = ix.BatchSearch(q_np, cfg["k"], mc)
            dt = time.time() - t0
            r = bench.recall_at_k(v, tv, cfg["k"])
            print(f"[s5] pivots={pivots} mc={mc}: recall@10={r:.4f} "
                  f"qps={cfg['nq']/dt:.0f}", flush=True)


if __name__ == "__main__":
    main()
