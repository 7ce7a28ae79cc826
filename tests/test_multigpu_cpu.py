"""Sharded-search semantics on CPU (gloo, world_size=2).

Validates the multi-GPU path of bench.py without GPUs: the same dataset is
range-sharded by contiguous VID ranges, every rank searches its shard (the
oracle stands in for the GPU backend — same results by the parity
contract), per-shard top-k lists are all-gathered and merged. The merged
result must equal the single-index search's result set quality, and the
merged truth must equal global truth (exact by construction — SURVEY.md
§8e: union of shard top-ks covers the global top-k).
"""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

import bench
from sptag_amd.build import build_index_arrays
from oracle.pyoracle import OrcIndex

WORLD = 2


def _make_data(n=6000, d=24, nq=64):
    g = torch.Generator()
    g.manual_seed(2016)
    centers = torch.rand((100, d), generator=g) * 255.0
    lab = torch.randint(0, 100, (n,), generator=g)
    x = centers[lab] + torch.randn((n, d), generator=g) * 32.0
    qlab = torch.randint(0, 100, (nq,), generator=g)
    q = centers[qlab] + torch.randn((nq, d), generator=g) * 32.0
    return x.numpy(), q.numpy()


def _rank_main(rank, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29517"
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)

    x, q = _make_data()
    n = x.shape[0]
    k = 10
    lo = n * rank // WORLD
    hi = n * (rank + 1) // WORLD
    shard = x[lo:hi]

    arrays = build_index_arrays(shard, "L2", device="cpu", ntrees=2,
                                tpt_leaf=500, cand=128, refine_rounds=0)
    ix = OrcIndex.from_arrays(shard, arrays["tree_start"],
                              arrays["tree_nodes"], arrays["graph"], "L2")
    vids, dists = ix.search_batch(q, k, 2048, nthreads=2)
    gvids = np.where(vids >= 0, vids + lo, -1).astype(np.int32)

    tv = torch.from_numpy(gvids)
    td = torch.from_numpy(dists)
    agv = [torch.zeros_like(tv) for _ in range(WORLD)]
    agd = [torch.zeros_like(td) for _ in range(WORLD)]
    dist.all_gather(agv, tv)
    dist.all_gather(agd, td)

    # shard truth gather
    stv, std_ = ix.truth(q, k, nthreads=2)
    stv = np.where(stv >= 0, stv + lo, -1).astype(np.int32)
    tgv = [torch.zeros_like(torch.from_numpy(stv)) for _ in range(WORLD)]
    tgd = [torch.zeros_like(torch.from_numpy(std_)) for _ in range(WORLD)]
    dist.all_gather(tgv, torch.from_numpy(stv))
    dist.all_gather(tgd, torch.from_numpy(std_))

    if rank == 0:
        mv, md = bench.merge_topk(torch.cat(agv, 1).numpy(),
                                  torch.cat(agd, 1).numpy(), k)
        ttv, ttd = bench.merge_topk(torch.cat(tgv, 1).numpy(),
                                    torch.cat(tgd, 1).numpy(), k)
        # merged truth == exact global truth
        gix = OrcIndex.from_arrays(x, arrays["tree_start"],
                                   arrays["tree_nodes"],
                                   np.zeros((n, 4), np.int32), "L2")
        fullv, fulld = gix.truth(q, k, nthreads=2)
        np.testing.assert_array_equal(ttv, fullv)
        np.testing.assert_array_equal(ttd, fulld)
        # merged search reaches the recall gate
        recall = bench.recall_at_k(mv, fullv, k)
        np.save(os.path.join(out_dir, "recall.npy"), np.array([recall]))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_sharded_search_merge(tmp_path):
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_rank_main, args=(r, str(tmp_path)))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(500)
        assert p.exitcode == 0, p.exitcode
    recall = float(np.load(tmp_path / "recall.npy")[0])
    assert recall > 0.95, recall
