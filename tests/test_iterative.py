"""Iterative (streaming) search parity.

Golden fixtures (iter_b8_c5.bin) were produced by driving the REFERENCE's
own ResultIterator (oracle/_ref/iterprobe, linked against the reference
objects): per query, 5 Next(8) calls recording count, the sticky
relaxed-monotonicity flag, ids and distances. The oracle restatement and
the GPU iterator must both reproduce them bit-exactly."""
import os

import numpy as np
import pytest

from conftest import GOLDEN, golden_fixtures, load_golden
from oracle.pyoracle import OrcIndex

import sptag_amd

FIXTURES = ["f32_l2_n10k_d32", "i8_l2_n10k_d100"]


def load_iter_golden(name):
    raw = open(os.path.join(GOLDEN, name, "iter_b8_c5.bin"), "rb").read()
    nq, batch, ncalls = np.frombuffer(raw[:12], dtype=np.int32)
    rec = np.frombuffer(raw[12:], dtype=np.int32).reshape(nq, ncalls,
                                                          2 + 2 * batch)
    return int(nq), int(batch), int(ncalls), rec


@pytest.mark.parametrize("name", FIXTURES)
def test_oracle_iterative_bit_exact(name):
    g = load_golden(name)
    nq, batch, ncalls, rec = load_iter_golden(name)
    ix = OrcIndex.load(g["index"])
    for i in range(nq):
        it = ix.iterate(g["queries"][i])
        for c in range(ncalls):
            cnt, vids, dists, rel = it.next(batch)
            gold = rec[i, c]
            assert cnt == gold[0], (name, i, c)
            assert rel == gold[1], (name, i, c)
            np.testing.assert_array_equal(vids[:cnt], gold[2::2][:cnt])
            np.testing.assert_array_equal(dists[:cnt],
                                          gold[3::2].view(np.float32)[:cnt])


@pytest.mark.gpu
@pytest.mark.parametrize("name", FIXTURES)
def test_gpu_iterative_bit_exact(name):
    from sptag_amd import AnnIndex
    g = load_golden(name)
    nq, batch, ncalls, rec = load_iter_golden(name)
    ix = AnnIndex.Load(g["index"])
    it = ix.Iterate(g["queries"], max_check=8192)
    for c in range(ncalls):
        vids, dists, counts, relaxed = it.Next(batch)
        np.testing.assert_array_equal(counts, rec[:, c, 0],
                                      err_msg=f"{name} call {c} counts")
        np.testing.assert_array_equal(relaxed, rec[:, c, 1],
                                      err_msg=f"{name} call {c} relaxed")
        for i in range(nq):
            cnt = counts[i]
            np.testing.assert_array_equal(vids[i, :cnt], rec[i, c, 2::2][:cnt])
            np.testing.assert_array_equal(
                dists[i, :cnt], rec[i, c, 3::2].view(np.float32)[:cnt])
    it.Close()


@pytest.mark.gpu
def test_kdt_iterate_refused_like_reference():
    """Reference parity for the refusal itself: KDT::Index<T>::GetIterator
    logs 'ITERATIVE NOT SUPPORT FOR KDT' and returns null
    (src/Core/KDT/KDTIndex.cpp:322-346) — the backend's iterator create
    rejects KDT handles the same way."""
    names = [n for n in golden_fixtures() if n.startswith("kdt")]
    if not names:
        pytest.skip("no KDT fixtures")
    g = load_golden(names[0])
    ix = sptag_amd.AnnIndex.Load(g["index"])
    with pytest.raises(Exception):
        ix.Iterate(g["queries"][:4], max_check=512)
