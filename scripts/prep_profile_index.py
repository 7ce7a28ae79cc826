#!/usr/bin/env python3
"""Build the bench workload's index once (unprofiled) and save it plus the
query set for the rocprofv3 passes. Usage: prep_profile_index.py <workload>
<out_dir>."""
import os
import sys

import numpy as np
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import bench  # noqa: E402
import sptag_amd  # noqa: E402
from sptag_amd.build import build_index_arrays  # noqa: E402


def main():
    workload, out_dir = sys.argv[1:3]
    cfg = bench.CONFIGS[workload]
    x, q, lo = bench.gen_data(cfg, 0, 1, "cuda:0", torch)
    arrays = build_index_arrays(x.cpu().numpy(), cfg["metric"], ntrees=4,
                                refine_rounds=0, device="cuda:0")
    torch.cuda.empty_cache()
    ix = sptag_amd.AnnIndex.FromArrays(
        arrays["vectors"], arrays["tree_start"], arrays["tree_nodes"],
        arrays["graph"], cfg["metric"])
    os.makedirs(out_dir, exist_ok=True)
    ix.Save(out_dir)
    np.save(os.path.join(out_dir, "queries.npy"), q.cpu().numpy())
    print("saved", out_dir)


if __name__ == "__main__":
    main()
