"""ctypes wrapper over liboracle.so — TEST INFRASTRUCTURE ONLY.

Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
import this module (see sptag_oracle.h). The product path (sptag_amd) never
touches it.
"""
import ctypes
import os
import subprocess

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIBPATH = os.path.join(_HERE, "liboracle.so")
_LIB = None


def load_library():
    global _LIB
    if _LIB is not None:
        return _LIB
    if not os.path.exists(_LIBPATH):
        subprocess.run(["make", "-C", _HERE, "oracle"], check=True,
                       capture_output=True)
    lib = ctypes.CDLL(_LIBPATH)
    lib.orc_load_index.restype = ctypes.c_void_p
    lib.orc_load_index.argtypes = [ctypes.c_char_p]
    lib.orc_create_index.restype = ctypes.c_void_p
    lib.orc_create_index.argtypes = [
        ctypes.c_int32, ctypes.c_int32, ctypes.c_int, ctypes.c_int,
        ctypes.c_void_p, ctypes.c_int32, ctypes.c_void_p, ctypes.c_int32,
        ctypes.c_void_p, ctypes.c_int32, ctypes.c_void_p, ctypes.c_void_p]
    lib.orc_create_kdt_index.restype = ctypes.c_void_p
    lib.orc_create_kdt_index.argtypes = lib.orc_create_index.argtypes
    lib.orc_free_index.argtypes = [ctypes.c_void_p]
    lib.orc_search.restype = ctypes.c_int32
    lib.orc_search.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int32,
                               ctypes.c_int32, ctypes.c_void_p, ctypes.c_void_p]
    lib.orc_search_batch.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int32, ctypes.c_int32,
        ctypes.c_int32, ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p]
    lib.orc_truth.argtypes = lib.orc_search_batch.argtypes[:4] + [
        ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p]
    lib.orc_iter_create.restype = ctypes.c_void_p
    lib.orc_iter_create.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                    ctypes.c_int32]
    lib.orc_iter_next.restype = ctypes.c_int32
    lib.orc_iter_next.argtypes = [ctypes.c_void_p, ctypes.c_int32,
                                  ctypes.c_void_p, ctypes.c_void_p,
                                  ctypes.c_void_p]
    lib.orc_iter_free.argtypes = [ctypes.c_void_p]
    lib.orc_distance.restype = ctypes.c_float
    lib.orc_distance.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_void_p,
                                 ctypes.c_void_p, ctypes.c_int32]
    for f in ["num_vectors", "dim", "valuetype", "distmethod", "degree"]:
        fn = getattr(lib, "orc_" + f)
        fn.restype = ctypes.c_int32
        fn.argtypes = [ctypes.c_void_p]
    _LIB = lib
    return lib


class OrcIndex:
    def __init__(self, handle):
        self._lib = load_library()
        if not handle:
            raise RuntimeError("oracle index load failed")
        self._h = handle

    @classmethod
    def load(cls, folder):
        return cls(load_library().orc_load_index(str(folder).encode()))

    @classmethod
    def from_arrays(cls, vectors, tree_start, tree_nodes, graph, distmethod,
                    deleted=None):
        lib = load_library()
        vectors = np.ascontiguousarray(vectors)
        vt = 0 if vectors.dtype == np.float32 else 1
        dm = {"L2": 0, "Cosine": 1}.get(distmethod, distmethod)
        tree_start = np.ascontiguousarray(tree_start, dtype=np.int32)
        tree_nodes = np.ascontiguousarray(tree_nodes, dtype=np.int32)
        graph = np.ascontiguousarray(graph, dtype=np.int32)
        delp = None
        if deleted is not None:
            deleted = np.ascontiguousarray(deleted, dtype=np.uint8)
            delp = deleted.ctypes.data_as(ctypes.c_void_p)
        h = lib.orc_create_index(
            vectors.shape[0], vectors.shape[1], vt, dm,
            vectors.ctypes.data_as(ctypes.c_void_p),
            len(tree_start), tree_start.ctypes.data_as(ctypes.c_void_p),
            tree_nodes.size // 3, tree_nodes.ctypes.data_as(ctypes.c_void_p),
            graph.shape[1], graph.ctypes.data_as(ctypes.c_void_p), delp)
        return cls(h)

    @classmethod
    def from_arrays_kdt(cls, vectors, tree_start, kdt_nodes, graph, distmethod,
                        deleted=None):
        lib = load_library()
        vectors = np.ascontiguousarray(vectors)
        vt = 0 if vectors.dtype == np.float32 else 1
        dm = {"L2": 0, "Cosine": 1}.get(distmethod, distmethod)
        tree_start = np.ascontiguousarray(tree_start, dtype=np.int32)
        kdt_nodes = np.ascontiguousarray(kdt_nodes, dtype=np.int32)
        graph = np.ascontiguousarray(graph, dtype=np.int32)
        delp = None
        if deleted is not None:
            deleted = np.ascontiguousarray(deleted, dtype=np.uint8)
            delp = deleted.ctypes.data_as(ctypes.c_void_p)
        h = lib.orc_create_kdt_index(
            vectors.shape[0], vectors.shape[1], vt, dm,
            vectors.ctypes.data_as(ctypes.c_void_p),
            len(tree_start), tree_start.ctypes.data_as(ctypes.c_void_p),
            kdt_nodes.size // 4, kdt_nodes.ctypes.data_as(ctypes.c_void_p),
            graph.shape[1], graph.ctypes.data_as(ctypes.c_void_p), delp)
        return cls(h)

    def __del__(self):
        if getattr(self, "_h", None):
            self._lib.orc_free_index(self._h)
            self._h = None

    @property
    def n(self):
        return self._lib.orc_num_vectors(self._h)

    @property
    def dim(self):
        return self._lib.orc_dim(self._h)

    @property
    def valuetype(self):
        return self._lib.orc_valuetype(self._h)

    @property
    def distmethod(self):
        return self._lib.orc_distmethod(self._h)

    def _qarr(self, queries):
        dtype = np.float32 if self.valuetype == 0 else np.int8
        queries = np.ascontiguousarray(queries, dtype=dtype)
        if queries.ndim == 1:
            queries = queries[None, :]
        assert queries.shape[1] == self.dim
        return queries

    def search_batch(self, queries, k, max_check, nthreads=0):
        queries = self._qarr(queries)
        nq = queries.shape[0]
        vids = np.empty((nq, k), dtype=np.int32)
        dists = np.empty((nq, k), dtype=np.float32)
        self._lib.orc_search_batch(
            self._h, queries.ctypes.data_as(ctypes.c_void_p), nq, k, max_check,
            nthreads, vids.ctypes.data_as(ctypes.c_void_p),
            dists.ctypes.data_as(ctypes.c_void_p))
        return vids, dists

    def truth(self, queries, k, nthreads=0):
        queries = self._qarr(queries)
        nq = queries.shape[0]
        vids = np.empty((nq, k), dtype=np.int32)
        dists = np.empty((nq, k), dtype=np.float32)
        self._lib.orc_truth(
            self._h, queries.ctypes.data_as(ctypes.c_void_p), nq, k, nthreads,
            vids.ctypes.data_as(ctypes.c_void_p),
            dists.ctypes.data_as(ctypes.c_void_p))
        return vids, dists

    def iterate(self, query):
        return OrcIter(self, query)

    def distance(self, x, y):
        dtype = np.float32 if self.valuetype == 0 else np.int8
        x = np.ascontiguousarray(x, dtype=dtype)
        y = np.ascontiguousarray(y, dtype=dtype)
        return self._lib.orc_distance(self.valuetype, self.distmethod,
                                      x.ctypes.data_as(ctypes.c_void_p),
                                      y.ctypes.data_as(ctypes.c_void_p),
                                      x.size)


def recall_at_k(vids, truth_vids, truth_dists, k):
    """TruthSet::CalculateRecall semantics (reference TruthSet.h:167-201):
    a returned id counts if it is in the truth set OR its distance ties the
    k-th truth distance; here we use the common id-set intersection plus
    distance-tie tolerance via truth_dists."""
    nq = vids.shape[0]
    hits = 0
    for i in range(nq):
        tset = set(truth_vids[i, :k].tolist())
        # distance ties: any vid whose true distance equals the k-th truth
        # distance also counts; approximate by id-set here (exact ties are
        # handled by the caller where needed)
        hits += len(tset.intersection(vids[i, :k].tolist()))
    return hits / (nq * k)


class OrcIter:
    """reference ResultIterator restatement (see sptag_oracle.c)."""

    def __init__(self, index, query, max_check=8192):
        self._lib = index._lib
        self._index = index
        q = index._qarr(query)[0]
        self._h = self._lib.orc_iter_create(
            index._h, np.ascontiguousarray(q).ctypes.data_as(ctypes.c_void_p),
            max_check)

    def next(self, batch):
        vids = np.empty(batch, dtype=np.int32)
        dists = np.empty(batch, dtype=np.float32)
        rel = ctypes.c_int32()
        cnt = self._lib.orc_iter_next(self._h, batch,
                                      vids.ctypes.data_as(ctypes.c_void_p),
                                      dists.ctypes.data_as(ctypes.c_void_p),
                                      ctypes.byref(rel))
        return cnt, vids, dists, rel.value

    def __del__(self):
        if getattr(self, "_h", None):
            self._lib.orc_iter_free(self._h)
            self._h = None
