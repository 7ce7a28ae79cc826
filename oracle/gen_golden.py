#!/usr/bin/env python3
"""Generate golden parity fixtures under tests/golden/ using the REFERENCE
binaries in oracle/_ref (compiled from /root/reference by oracle/Makefile).

Run in the dev container (where /root/reference exists):
    python3 oracle/gen_golden.py

For each fixture we commit:
  index/            the reference-built index folder (vectors/tree/graph/
                    deletes + indexloader.ini) — build is NOT deterministic
                    (SURVEY.md §8c), so golden answers are pinned to this
                    saved index, never to a rebuild.
  queries.bin       queries in DEFAULT format [int32 n][int32 dim][data]
  results_mc{M}.bin reference indexsearcher output, binary format
                    [int32 nq][int32 K] + nq*K*{int32 vid, float dist}
                    (reference src/IndexSearcher/main.cpp:314-331)
  meta.json         shapes/types/maxchecks

Fixture search results verified byte-stable across runs and thread counts
(SURVEY.md §8c), so they are exact golden answers for any backend that
claims bit-parity.
"""
import json
import os
import subprocess
import sys

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)
REF = os.path.join(HERE, "_ref")
GOLDEN = os.path.join(REPO, "tests", "golden")

K = 10
MAXCHECKS = [512, 2048, 8192]


def write_default(path, arr):
    """DEFAULT file format: [int32 rows][int32 cols][row-major data]
    (reference Dataset.h:146 / Helper VectorSetReaders/DefaultReader)."""
    with open(path, "wb") as f:
        f.write(np.int32(arr.shape[0]).tobytes())
        f.write(np.int32(arr.shape[1]).tobytes())
        f.write(arr.tobytes())


def run(cmd):
    print("+", " ".join(cmd))
    subprocess.run(cmd, check=True, cwd=REPO)


def gen_fixture(name, data, queries, vtype, dist, maxchecks=MAXCHECKS, k=K,
                algo="BKT"):
    d = os.path.join(GOLDEN, name)
    idx = os.path.join(d, "index")
    os.makedirs(d, exist_ok=True)
    write_default(os.path.join(d, "data.bin"), data)
    write_default(os.path.join(d, "queries.bin"), queries)

    run([os.path.join(REF, "indexbuilder"),
         "-d", str(data.shape[1]), "-v", vtype, "-f", "DEFAULT",
         "-i", os.path.join(d, "data.bin"), "-o", idx, "-a", algo,
         "-t", "4",
         "Index.DistCalcMethod=" + dist])

    for mc in maxchecks:
        run([os.path.join(REF, "indexsearcher"),
             "-d", str(data.shape[1]), "-v", vtype, "-f", "DEFAULT",
             "-i", os.path.join(d, "queries.bin"), "-x", idx,
             "-k", str(k), "-m", str(mc), "-t", "2", "-of", "1",
             "-o", os.path.join(d, f"results_mc{mc}.bin")])

    # data.bin duplicates index/vectors.bin content (normalized for cosine);
    # keep queries only + the index folder to stay small.
    os.remove(os.path.join(d, "data.bin"))
    with open(os.path.join(d, "meta.json"), "w") as f:
        json.dump({"n": int(data.shape[0]), "dim": int(data.shape[1]),
                   "valuetype": vtype, "distmethod": dist, "k": k, "algo": algo,
                   "maxchecks": maxchecks, "nq": int(queries.shape[0])}, f)
    print(f"fixture {name} done")


def gen_fixture_ex(name, data, queries, vtype, dist, maxchecks, algo="BKT",
                   k=K, extra=()):
    """gen_fixture with extra builder parameters (Section.Name=Value argv,
    reference IndexSearcher/IndexBuilder main.cpp:362-389)."""
    d = os.path.join(GOLDEN, name)
    idx = os.path.join(d, "index")
    os.makedirs(d, exist_ok=True)
    write_default(os.path.join(d, "data.bin"), data)
    write_default(os.path.join(d, "queries.bin"), queries)
    run([os.path.join(REF, "indexbuilder"), "-d", str(data.shape[1]),
         "-v", vtype, "-f", "DEFAULT", "-i", os.path.join(d, "data.bin"),
         "-o", idx, "-a", algo, "-t", "4",
         "Index.DistCalcMethod=" + dist, *extra])
    for mc in maxchecks:
        run([os.path.join(REF, "indexsearcher"), "-d", str(data.shape[1]),
             "-v", vtype, "-f", "DEFAULT", "-i", os.path.join(d, "queries.bin"),
             "-x", idx, "-k", str(k), "-m", str(mc), "-t", "2", "-of", "1",
             "-o", os.path.join(d, f"results_mc{mc}.bin")])
    os.remove(os.path.join(d, "data.bin"))
    with open(os.path.join(d, "meta.json"), "w") as f:
        json.dump({"n": int(data.shape[0]), "dim": int(data.shape[1]),
                   "valuetype": vtype, "distmethod": dist, "k": k, "algo": algo,
                   "maxchecks": maxchecks, "nq": int(queries.shape[0])}, f)
    print(f"fixture {name} done")


def gen_iter_and_add_goldens():
    """Iterative-search and online-add golden answers, produced by driving
    the reference's OWN ResultIterator / AddIndex (iterprobe / addprobe
    harnesses linked against the oracle/_ref objects)."""
    rng = np.random.default_rng(555)
    for name in ["f32_l2_n10k_d32", "i8_l2_n10k_d100"]:
        d = os.path.join(GOLDEN, name)
        meta = json.load(open(os.path.join(d, "meta.json")))
        run([os.path.join(REF, "iterprobe"), os.path.join(d, "index"),
             os.path.join(d, "queries.bin"), "8", "5",
             os.path.join(d, "iter_b8_c5.bin")])
        dim = meta["dim"]
        if meta["valuetype"] == "Float":
            add = (rng.random((64, dim), dtype=np.float32) * 100).astype(np.float32)
        else:
            add = rng.integers(-100, 101, (64, dim)).astype(np.int8)
        write_default(os.path.join(d, "add_vectors.bin"), add)
        out = os.path.join("/tmp", "postadd_" + name)
        run([os.path.join(REF, "addprobe"), os.path.join(d, "index"),
             os.path.join(d, "add_vectors.bin"), out])
        import shutil
        shutil.copy(os.path.join(out, "graph.bin"),
                    os.path.join(d, "postadd_graph.bin"))


def main():
    os.makedirs(GOLDEN, exist_ok=True)
    rng = np.random.default_rng(2016)

    # 1. f32 L2, uniform — the basic float case (BASELINE config #1 shape, small)
    data = rng.random((10000, 32), dtype=np.float32)
    queries = rng.random((100, 32), dtype=np.float32)
    gen_fixture("f32_l2_n10k_d32", data, queries, "Float", "L2")

    # 2. int8 cosine, SPACEV-shaped (config #3, small). Builder normalizes
    #    base vectors to norm 127 in place (BKTIndex.cpp:749-756).
    data = rng.integers(-100, 101, (20000, 100)).astype(np.int8)
    queries = rng.integers(-100, 101, (200, 100)).astype(np.int8)
    gen_fixture("i8_cos_n20k_d100", data, queries, "Int8", "Cosine")

    # 3. int8 L2 (config #5 dtype/metric, small): integer distances — the
    #    identical-top-k-IDs parity case.
    data = rng.integers(-100, 101, (10000, 100)).astype(np.int8)
    queries = rng.integers(-100, 101, (100, 100)).astype(np.int8)
    gen_fixture("i8_l2_n10k_d100", data, queries, "Int8", "L2")

    # 4. AlgoTest-style deterministic grid (reference Test/src/AlgoTest.cpp:161:
    #    vector i = (i,i,...,i), query j = (2j,...)): heavy distance ties and
    #    duplicate-free but collapse-prone clusters; known answers.
    n, dim = 2000, 10
    data = np.tile(np.arange(n, dtype=np.float32)[:, None], (1, dim))
    queries = np.tile((np.arange(100, dtype=np.float32) * 2)[:, None], (1, dim))
    gen_fixture("f32_l2_grid_ties", data, queries, "Float", "L2",
                maxchecks=[512, 2048], k=3)

    # 5. duplicate vectors — exercises the collapsed-center duplicate chain
    #    (BKTIndex.cpp:292-312, BKTree.h:598-608): 2000 base vectors, each
    #    repeated 4x.
    base = rng.random((2000, 16), dtype=np.float32)
    data = np.repeat(base, 4, axis=0)
    perm = rng.permutation(len(data))
    data = data[perm]
    queries = rng.random((100, 16), dtype=np.float32)
    gen_fixture("f32_l2_dups", data, queries, "Float", "L2",
                maxchecks=[512, 2048], k=10)

    # 7. KDT fixtures (config #4 algo): same traversal machinery, kd-tree
    #    seeds + no-better-propagation termination (KDTIndex.cpp:184-241)
    data = rng.standard_normal((10000, 64)).astype(np.float32)
    queries = rng.standard_normal((100, 64)).astype(np.float32)
    gen_fixture("kdt_f32_cos_n10k_d64", data, queries, "Float", "Cosine",
                maxchecks=[512, 2048, 8192], algo="KDT")
    data = rng.random((10000, 32), dtype=np.float32)
    queries = rng.random((100, 32), dtype=np.float32)
    gen_fixture("kdt_f32_l2_n10k_d32", data, queries, "Float", "L2",
                maxchecks=[512, 2048, 8192], algo="KDT")
    data = rng.integers(-100, 101, (10000, 100)).astype(np.int8)
    queries = rng.integers(-100, 101, (100, 100)).astype(np.int8)
    gen_fixture("kdt_i8_l2_n10k_d100", data, queries, "Int8", "L2",
                maxchecks=[2048, 8192], algo="KDT")

    # 6. f32 cosine (config #4 metric; KDT later — BKT for now), normalized
    #    gaussian rows, d=48 not a multiple of 16 to cover distance tails.
    data = rng.standard_normal((10000, 48)).astype(np.float32)
    queries = rng.standard_normal((100, 48)).astype(np.float32)
    gen_fixture("f32_cos_n10k_d48", data, queries, "Float", "Cosine",
                maxchecks=[512, 2048, 8192])

    # 8. builder-parameter coverage: multi-tree BKT, non-default degree,
    #    2-tree KDT, odd int8 dim (byte-wise distance path)
    rng2 = np.random.default_rng(777)
    data = rng2.random((8000, 32), dtype=np.float32) * 100
    queries = rng2.random((100, 32), dtype=np.float32) * 100
    gen_fixture_ex("f32_l2_bkt2trees", data, queries, "Float", "L2",
                   [512, 2048], extra=("Index.BKTNumber=2",))
    data = rng2.random((8000, 48), dtype=np.float32)
    queries = rng2.random((100, 48), dtype=np.float32)
    gen_fixture_ex("f32_l2_deg16", data, queries, "Float", "L2", [512, 2048],
                   extra=("Index.NeighborhoodSize=16",))
    data = rng2.integers(-100, 101, (8000, 100)).astype(np.int8)
    queries = rng2.integers(-100, 101, (100, 100)).astype(np.int8)
    gen_fixture_ex("kdt_i8_cos_2trees", data, queries, "Int8", "Cosine",
                   [2048, 8192], algo="KDT", extra=("Index.KDTNumber=2",))
    rng3 = np.random.default_rng(4242)
    data = rng3.integers(-100, 101, (8000, 37)).astype(np.int8)
    queries = rng3.integers(-100, 101, (100, 37)).astype(np.int8)
    gen_fixture_ex("i8_l2_odd_d37", data, queries, "Int8", "L2", [512, 2048])

    # 9. iterative + online-add golden answers
    gen_iter_and_add_goldens()


if __name__ == "__main__":
    sys.exit(main())
