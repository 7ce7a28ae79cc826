"""Builder tests (CPU; torch on host): the produced tree/graph arrays must
be structurally valid, loadable by the oracle (pinned to the reference
loader), and good enough that deep search reaches high recall."""
import numpy as np
import pytest

from oracle.pyoracle import OrcIndex
from sptag_amd.build import build_index_arrays, build_bkt_tree, normalize_base


def make_clustered(n, d, ncenters, seed=2016, dtype=np.float32, sigma=40.0,
                   nq=0):
    """Overlapping Gaussian mixture (SIFT-like clusterability — BASELINE.md
    synthetic-data note). Returns (data, queries) drawn from the SAME
    mixture."""
    rng = np.random.default_rng(seed)
    centers = rng.random((ncenters, d), dtype=np.float32) * 255.0
    lab = rng.integers(0, ncenters, n)
    x = centers[lab] + rng.standard_normal((n, d)).astype(np.float32) * sigma
    q = None
    if nq:
        qlab = rng.integers(0, ncenters, nq)
        q = (centers[qlab] +
             rng.standard_normal((nq, d)).astype(np.float32) * sigma).astype(dtype)
    return x.astype(dtype), q


def check_tree(tree_nodes, n):
    cent, cs, ce = tree_nodes[:, 0], tree_nodes[:, 1], tree_nodes[:, 2]
    N = len(tree_nodes)
    # sentinel
    assert cent[-1] == -1
    # every vector appears exactly once as a leaf
    leaves = cent[(cs < 0) & (cent >= 0)]
    assert len(np.unique(leaves)) == n, (len(np.unique(leaves)), n)
    # children ranges valid and non-overlapping (DFS property not required)
    internal = np.where(cs > 0)[0]
    assert (ce[internal] <= N).all()
    assert (ce[internal] > cs[internal]).all()
    # all internal centerids (except root) are valid vector ids
    assert (cent[internal][1:] < n).all()


def test_tree_structure():
    x, _ = make_clustered(5000, 16, 50)
    tree_start, tree_nodes = build_bkt_tree(x, device="cpu", big_cluster=512)
    assert tree_start.tolist() == [0]
    check_tree(tree_nodes, 5000)


@pytest.mark.parametrize("dm", ["L2", "Cosine"])
def test_build_and_search_recall(dm):
    n, d = 20000, 32
    x, q = make_clustered(n, d, 200, nq=200)
    arrays = build_index_arrays(x, dm, device="cpu", ntrees=4, tpt_leaf=500,
                                refine_rounds=1)
    check_tree(arrays["tree_nodes"], n)
    g = arrays["graph"]
    assert g.shape == (n, 32)
    assert (g < n).all()
    # no self-loops
    assert not (g == np.arange(n)[:, None]).any()

    ix = OrcIndex.from_arrays(arrays["vectors"], arrays["tree_start"],
                              arrays["tree_nodes"], arrays["graph"], dm)
    if dm == "Cosine":
        q = q / np.linalg.norm(q, axis=1, keepdims=True)
    tv, td = ix.truth(q, 10, nthreads=4)
    vids, _ = ix.search_batch(q, 10, 2048, nthreads=4)
    hits = sum(len(set(tv[i]).intersection(vids[i])) for i in range(len(q)))
    recall = hits / (len(q) * 10)
    assert recall > 0.95, recall


def test_normalize_int8_matches_reference_truncation():
    rng = np.random.default_rng(3)
    x = rng.integers(-100, 101, (100, 100)).astype(np.int8)
    out = normalize_base(x, "Cosine")
    norms = np.linalg.norm(out.astype(np.float64), axis=1)
    assert (np.abs(norms - 127) < 8).all()


def test_kdt_build_and_search_recall():
    from sptag_amd.build import build_kdt_tree
    from oracle.pyoracle import OrcIndex
    n, d = 20000, 32
    x, q = make_clustered(n, d, 200, nq=200)
    arrays = build_index_arrays(x, "L2", algo="KDT", device="cpu", ntrees=4,
                                tpt_leaf=500, cand=128)
    kn = arrays["tree_nodes"]
    assert kn.shape[1] == 4
    # every vector appears exactly once as a leaf
    leaves = np.concatenate([kn[:, 0], kn[:, 1]])
    leaves = -leaves[leaves < 0] - 1
    leaves = leaves[leaves < n]
    assert len(np.unique(leaves)) == n
    ix = OrcIndex.from_arrays_kdt(x, arrays["tree_start"], kn,
                                  arrays["graph"], "L2")
    tv, _ = ix.truth(q, 10, nthreads=4)
    vids, _ = ix.search_batch(q, 10, 2048, nthreads=4)
    hits = sum(len(set(tv[i]).intersection(vids[i])) for i in range(len(q)))
    assert hits / (len(q) * 10) > 0.9, hits / (len(q) * 10)
