"""Online add parity: golden post-add graphs were produced by the
REFERENCE's own AddIndex (oracle/_ref/addprobe). The oracle restatement and
the GPU-backed sptag_amd_add must reproduce every graph row bit-exactly."""
import ctypes
import json
import os

import numpy as np
import pytest

from conftest import GOLDEN, load_golden
from oracle.pyoracle import OrcIndex, load_library

FIXTURES = ["f32_l2_n10k_d32", "i8_l2_n10k_d100"]


def load_add_fixture(name):
    d = os.path.join(GOLDEN, name)
    meta = json.load(open(os.path.join(d, "meta.json")))
    dtype = np.float32 if meta["valuetype"] == "Float" else np.int8
    with open(os.path.join(d, "add_vectors.bin"), "rb") as f:
        n, dim = np.frombuffer(f.read(8), dtype=np.int32)
        add = np.frombuffer(f.read(), dtype=dtype).reshape(n, dim)
    with open(os.path.join(d, "postadd_graph.bin"), "rb") as f:
        gn, deg = np.frombuffer(f.read(8), dtype=np.int32)
        gref = np.frombuffer(f.read(), dtype=np.int32).reshape(gn, deg)
    return add, gref


@pytest.mark.parametrize("name", FIXTURES)
def test_oracle_add_bit_exact(name):
    g = load_golden(name)
    add, gref = load_add_fixture(name)
    lib = load_library()
    lib.orc_add.restype = ctypes.c_int
    lib.orc_add.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int32,
                            ctypes.c_int32, ctypes.c_int]
    lib.orc_graph_ptr.restype = ctypes.POINTER(ctypes.c_int32)
    lib.orc_graph_ptr.argtypes = [ctypes.c_void_p]
    ix = OrcIndex.load(g["index"])
    assert lib.orc_add(ix._h, add.ctypes.data_as(ctypes.c_void_p),
                       add.shape[0], 500, 0) == 0
    mine = np.ctypeslib.as_array(lib.orc_graph_ptr(ix._h), shape=gref.shape)
    np.testing.assert_array_equal(mine, gref, err_msg=name)


@pytest.mark.gpu
@pytest.mark.parametrize("name", FIXTURES)
def test_gpu_add_bit_exact(name, tmp_path):
    from sptag_amd import AnnIndex
    g = load_golden(name)
    add, gref = load_add_fixture(name)
    ix = AnnIndex.Load(g["index"])
    ix.Add(add)
    assert ix.n == gref.shape[0]
    out = tmp_path / "postadd"
    ix.Save(str(out))
    with open(out / "graph.bin", "rb") as f:
        gn, deg = np.frombuffer(f.read(8), dtype=np.int32)
        mine = np.frombuffer(f.read(), dtype=np.int32).reshape(gn, deg)
    np.testing.assert_array_equal(mine, gref, err_msg=name)


def test_add_fails_loudly_without_gpu():
    import sptag_amd
    if sptag_amd.gpu_available():
        pytest.skip("GPU present")
    from sptag_amd import AnnIndex, SptagAmdError
    g = load_golden("f32_l2_n10k_d32")
    add, _ = load_add_fixture("f32_l2_n10k_d32")
    ix = AnnIndex.Load(g["index"])
    with pytest.raises(SptagAmdError) as e:
        ix.Add(add)
    assert e.value.code == -3
