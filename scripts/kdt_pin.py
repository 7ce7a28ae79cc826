#!/usr/bin/env python3
"""KDT config-#4 plateau pinning.

Part A (GPU, 1M x 768): my KDT builder at kdt_trees = 1/2/4 — does seed
diversity lift the no-better-propagation recall plateau?

Part B (10M x 768): build+save my index, run the REFERENCE CPU searcher
(oracle/_ref/indexsearcher) on the SAME index folder with a bounded query
sample, and compute its recall against exact truth — pins that the 0.70
plateau is a property of (index, reference algorithm), not of the GPU
kernel (which is bit-exact against the reference on KDT goldens).

Usage: kdt_pin.py [a|b|ab]
"""
import os
import subprocess
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import bench  # noqa: E402
import sptag_amd  # noqa: E402
from sptag_amd.build import build_index_arrays  # noqa: E402

SWEEP = [2048, 4096, 8192, 16384, 32768]


def recall_curve(ix, q, tv, k, sweep=None):
    out = {}
    for mc in (sweep or SWEEP):
        v, _ = ix.BatchSearch(q.cpu().numpy(), k, mc)
        out[mc] = bench.recall_at_k(v, tv, k)
    return out


def part_a():
    cfg = bench.CONFIGS["kdt_1m_d768_f32_cos"]
    x, q, lo = bench.gen_data(cfg, 0, 1, "cuda:0", torch)
    x_np = x.cpu().numpy()
    for trees in (1, 2, 4):
        t0 = time.time()
        arrays = build_index_arrays(x_np, "Cosine", algo="KDT",
                                    kdt_trees=trees, cand=128,
                                    device="cuda:0")
        torch.cuda.empty_cache()
        ix = sptag_amd.AnnIndex.FromArraysKDT(
            arrays["vectors"], arrays["tree_start"], arrays["tree_nodes"],
            arrays["graph"], "Cosine")
        xs = torch.as_tensor(arrays["vectors"], device="cuda:0")
        tv, _ = bench.shard_truth(xs, q, cfg["k"], "Cosine", torch, 0)
        tv = tv.cpu().numpy()
        cur = recall_curve(ix, q, tv, cfg["k"])
        print(f"[kdt_pin A] trees={trees} build={time.time()-t0:.0f}s "
              f"recall@10: " + " ".join(f"mc{m}={r:.4f}"
                                        for m, r in cur.items()),
              flush=True)
        del ix, arrays, xs
        torch.cuda.empty_cache()


def part_b():
    cfg = bench.CONFIGS["kdt_10m_d768_f32_cos"]
    x, q, lo = bench.gen_data(cfg, 0, 1, "cuda:0", torch)
    x_np = x.cpu().numpy()
    t0 = time.time()
    arrays = build_index_arrays(x_np, "Cosine", algo="KDT", kdt_trees=2,
                                cand=128, device="cuda:0", verbose=True)
    torch.cuda.empty_cache()
    ix = sptag_amd.AnnIndex.FromArraysKDT(
        arrays["vectors"], arrays["tree_start"], arrays["tree_nodes"],
        arrays["graph"], "Cosine")
    print(f"[kdt_pin B] 10M build {time.time()-t0:.0f}s", flush=True)
    xs = torch.as_tensor(arrays["vectors"], device="cuda:0")
    tv, _ = bench.shard_truth(xs, q, cfg["k"], "Cosine", torch, 0)
    tv = tv.cpu().numpy()
    cur = recall_curve(ix, q, tv, cfg["k"], sweep=(4096, 8192, 16384))
    print("[kdt_pin B] GPU recall@10: " +
          " ".join(f"mc{m}={r:.4f}" for m, r in cur.items()), flush=True)

    idx = "/tmp/kdtpin_idx"
    os.makedirs(idx, exist_ok=True)
    ix.Save(idx)
    nq = 1500   # bounded sample for the CPU reference
    qs = q[:nq].cpu().numpy()
    qfile = os.path.join(idx, "q.bin")
    with open(qfile, "wb") as f:
        f.write(np.int32(nq).tobytes())
        f.write(np.int32(qs.shape[1]).tobytes())
        f.write(qs.tobytes())
    import multiprocessing
    cores = multiprocessing.cpu_count()
    ref = os.path.join(REPO, "oracle", "_ref", "indexsearcher")
    for mc in (8192, 16384):
        outb = os.path.join(idx, f"ref_mc{mc}.bin")
        r = subprocess.run([ref, "-d", str(qs.shape[1]), "-v", "Float",
                            "-f", "DEFAULT", "-i", qfile, "-x", idx,
                            "-k", "10", "-m", str(mc), "-t", str(cores),
                            "-of", "1", "-o", outb],
                           capture_output=True, text=True, timeout=1200)
        raw = open(outb, "rb").read()
        rec = np.frombuffer(raw[8:], dtype=np.dtype([("vid", np.int32),
                                                     ("dist", np.float32)]))
        rv = rec["vid"].reshape(nq, 10)
        rr = bench.recall_at_k(rv, tv[:nq], 10)
        print(f"[kdt_pin B] REFERENCE CPU on same index mc={mc}: "
              f"recall@10={rr:.4f} (cores={cores})", flush=True)


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "ab"
    if "a" in which:
        part_a()
    if "b" in which:
        part_b()
