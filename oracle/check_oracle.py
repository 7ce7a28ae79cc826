#!/usr/bin/env python3
"""Quick oracle-vs-golden check (the formal version lives in tests/).
Compares orc_search output with the reference binaries' golden results."""
import ctypes
import json
import os
import sys

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)
GOLDEN = os.path.join(REPO, "tests", "golden")

lib = ctypes.CDLL(os.path.join(HERE, "liboracle.so"))
lib.orc_load_index.restype = ctypes.c_void_p
lib.orc_load_index.argtypes = [ctypes.c_char_p]
lib.orc_search_batch.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                 ctypes.c_int32, ctypes.c_int32, ctypes.c_int32,
                                 ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p]
lib.orc_free_index.argtypes = [ctypes.c_void_p]


def read_default(path, dtype):
    with open(path, "rb") as f:
        n, d = np.frombuffer(f.read(8), dtype=np.int32)
        data = np.frombuffer(f.read(), dtype=dtype).reshape(n, d)
    return data


def check(name):
    d = os.path.join(GOLDEN, name)
    meta = json.load(open(os.path.join(d, "meta.json")))
    dtype = np.float32 if meta["valuetype"] == "Float" else np.int8
    queries = read_default(os.path.join(d, "queries.bin"), dtype)
    idx = lib.orc_load_index(os.path.join(d, "index").encode())
    assert idx, f"load failed: {name}"
    k, nq = meta["k"], meta["nq"]
    total = exact = id_exact = 0
    for mc in meta["maxchecks"]:
        raw = open(os.path.join(d, f"results_mc{mc}.bin"), "rb").read()
        hdr = np.frombuffer(raw[:8], dtype=np.int32)
        assert hdr[0] == nq and hdr[1] == k, (hdr, nq, k)
        rec = np.frombuffer(raw[8:], dtype=np.dtype([("vid", np.int32), ("dist", np.float32)]))
        rec = rec.reshape(nq, k)
        vids = np.empty((nq, k), dtype=np.int32)
        dists = np.empty((nq, k), dtype=np.float32)
        lib.orc_search_batch(idx, queries.ctypes.data_as(ctypes.c_void_p),
                             nq, k, mc, 4,
                             vids.ctypes.data_as(ctypes.c_void_p),
                             dists.ctypes.data_as(ctypes.c_void_p))
        same = (vids == rec["vid"]).all(axis=1) & (dists == rec["dist"]).all(axis=1)
        sameid = (vids == rec["vid"]).all(axis=1)
        total += nq
        exact += same.sum()
        id_exact += sameid.sum()
        bad = np.where(~same)[0][:3]
        for q in bad:
            print(f"  {name} mc={mc} q={q}:")
            print(f"    ref vid {rec['vid'][q]} dist {rec['dist'][q]}")
            print(f"    orc vid {vids[q]} dist {dists[q]}")
    lib.orc_free_index(idx)
    print(f"{name}: {exact}/{total} bit-exact rows, {id_exact}/{total} id-exact")
    return exact == total


def main():
    names = sys.argv[1:] or sorted(os.listdir(GOLDEN))
    ok = True
    for name in names:
        if os.path.isdir(os.path.join(GOLDEN, name)):
            ok &= check(name)
    print("ALL BIT-EXACT" if ok else "MISMATCHES PRESENT")
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
