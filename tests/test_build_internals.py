"""Unit tests of the builder's internal pieces (build.py): the TP-tree
partition, candidate merge, and RNG prune — each checked against a
straightforward numpy restatement of the same rule."""
import numpy as np
import torch

from sptag_amd.build import (_tpt_leaves, _merge_candidates, _rng_prune,
                             _ragged_arange, build_kdt_tree)


def test_ragged_arange():
    np.testing.assert_array_equal(_ragged_arange([3, 1, 2]),
                                  [0, 1, 2, 0, 0, 1])
    assert _ragged_arange([]).size == 0
    assert _ragged_arange([5]).tolist() == [0, 1, 2, 3, 4]


def test_tpt_leaves_partition():
    g = torch.Generator()
    g.manual_seed(1)
    x = torch.rand((5000, 8), generator=g)
    perm, bounds = _tpt_leaves(x, 200, g)
    # bounds tile perm exactly; every point appears once
    total = sum(c for _, c in bounds)
    assert total == 5000
    assert sorted(perm.tolist()) == list(range(5000))
    assert all(c <= 200 for _, c in bounds)
    starts = sorted(s for s, _ in bounds)
    ends = sorted(s + c for s, c in bounds)
    assert starts[0] == 0 and ends[-1] == 5000


def test_merge_candidates_dedupe_and_order():
    ids_a = torch.tensor([[3, 1, -1, 7]], dtype=torch.int32)
    dst_a = torch.tensor([[1.0, 2.0, float("inf"), 9.0]])
    ids_b = torch.tensor([[1, 5, 0, 3]], dtype=torch.int32)
    dst_b = torch.tensor([[2.0, 0.5, 4.0, 1.0]])
    self_ids = torch.tensor([0], dtype=torch.int32)
    ci, cd = _merge_candidates(ids_a, dst_a, ids_b, dst_b, 6, self_ids)
    # nearest first, ids deduped (1 and 3 appear once), self (0) dropped,
    # padded to width 6 with -1/inf
    assert ci.shape == (1, 6)
    got = [(int(i), float(d)) for i, d in zip(ci[0], cd[0]) if i >= 0]
    assert got == [(5, 0.5), (3, 1.0), (1, 2.0), (7, 9.0)]
    assert ci[0, 4] == -1 and not torch.isfinite(cd[0, 4])


def naive_rng_prune(x, cand_ids, cand_dst, degree, factor=1.0):
    """direct restatement of RebuildNeighbors (ascending candidates,
    accept c iff factor*dist(b,c) >= dist(q,c) for all accepted b)."""
    out = np.full(degree, -1, dtype=np.int32)
    cnt = 0
    for j in range(len(cand_ids)):
        if cnt >= degree or not np.isfinite(cand_dst[j]):
            break
        good = True
        for b in out[:cnt]:
            d = ((x[b] - x[cand_ids[j]]) ** 2).sum()
            if factor * d < cand_dst[j]:
                good = False
                break
        if good:
            out[cnt] = cand_ids[j]
            cnt += 1
    return out


def test_rng_prune_matches_naive():
    g = torch.Generator()
    g.manual_seed(3)
    x = torch.rand((500, 8), generator=g)
    q_rows = torch.arange(20)
    # candidates: 24 random others per row, sorted by true distance
    cid = torch.randint(20, 500, (20, 24), generator=g, dtype=torch.int32)
    d = ((x[cid.long()] - x[q_rows][:, None, :]) ** 2).sum(-1)
    order = d.argsort(1)
    cid = torch.gather(cid, 1, order)
    cdd = torch.gather(d, 1, order)
    pruned = _rng_prune(x, cid, cdd, 8, 1.0, "cpu").numpy()
    xn = x.numpy()
    for i in range(20):
        want = naive_rng_prune(xn, cid[i].numpy(), cdd[i].numpy(), 8)
        np.testing.assert_array_equal(pruned[i], want, err_msg=f"row {i}")


def test_kdt_tree_routes_every_point():
    """descending the built kd-tree by its own split rules reaches every
    vector exactly once (leaf ids -(v+1) partition the dataset)."""
    g = torch.Generator()
    g.manual_seed(5)
    x = torch.rand((2000, 12), generator=g).numpy().astype(np.float32)
    ts, nodes = build_kdt_tree(x, ntrees=2, device="cpu")
    n = x.shape[0]
    for t in range(2):
        seen = set()
        stack = [int(ts[t])]
        while stack:
            nd = stack.pop()
            if nd < 0:
                v = -nd - 1
                if v < n:
                    assert v not in seen
                    seen.add(v)
                continue
            stack.append(int(nodes[nd, 0]))
            stack.append(int(nodes[nd, 1]))
        assert len(seen) == n, (t, len(seen))


def test_rng_prune_fill_pruned():
    """fill_pruned pads leftover degree slots with the nearest REJECTED
    candidates (billion-scale build knob); the RNG-accepted prefix must be
    identical to the strict prune, and the filled row stays ascending."""
    torch.manual_seed(5)
    n, c, d, deg = 40, 24, 8, 12
    x = torch.randn(n, d)
    cdd0 = torch.rand(n, c).sort(dim=1).values * 10
    cid = torch.stack([torch.randperm(n)[:c] for _ in range(n)]).int()
    strict = _rng_prune(x, cid, cdd0, deg, 1.0, "cpu").numpy()
    filled = _rng_prune(x, cid, cdd0, deg, 1.0, "cpu",
                        fill_pruned=True).numpy()
    for i in range(n):
        s = strict[i][strict[i] >= 0]
        f = filled[i][filled[i] >= 0]
        assert len(f) >= len(s)
        assert len(f) == min(deg, c)  # all slots used (valid pool is full)
        # accepted set preserved (as a subsequence of the filled row)
        it = iter(f.tolist())
        assert all(v in it for v in s.tolist()) or set(s) <= set(f)
        # filled row ascending by the pool's distance order
        pos = {int(v): j for j, v in enumerate(cid[i].tolist())}
        order = [pos[int(v)] for v in f.tolist()]
        assert order == sorted(order)
