"""sptag_amd — MI355X-native SPTAG search backend (Python host mirror).

This module mirrors the reference Python wrapper's ``AnnIndex`` facade
(reference Wrappers/inc/CoreInterface.h:14: Load:72, Search:45,
BatchSearch:49) over the C-ABI in include/sptag_amd.h. Host logic lives in
the C++ library (sptag_amd/libsptag_amd.so); this wrapper only marshals
numpy arrays across the C boundary.

THE GPU IS THE PRODUCT PATH: importing works anywhere (so CPU-side tests can
check symbol exports), but every search raises loudly when the HIP device or
the extension is missing. There is no CPU fallback in this package.
"""
import ctypes
import os

import numpy as np

__all__ = ["AnnIndex", "MetadataSet", "load_library", "gpu_available",
           "SptagAmdError"]

_LIB = None
_LIBPATH = os.environ.get(
    "SPTAG_AMD_LIB",
    os.path.join(os.path.dirname(os.path.abspath(__file__)), "libsptag_amd.so"))

VT_FLOAT, VT_INT8 = 0, 1
DM_L2, DM_COSINE = 0, 1

_ERRNAMES = {
    0: "OK", -1: "IO", -2: "PARAM", -3: "NOGPU", -4: "UNSUPPORTED",
    -5: "OOM", -6: "INTERNAL",
}


class SptagAmdError(RuntimeError):
    def __init__(self, code, what=""):
        super().__init__(f"sptag_amd error {_ERRNAMES.get(code, code)} {what}")
        self.code = code


def load_library():
    """Load libsptag_amd.so; raises if the extension was not built."""
    global _LIB
    if _LIB is not None:
        return _LIB
    if not os.path.exists(_LIBPATH):
        raise SptagAmdError(
            -6, f"HIP extension missing: {_LIBPATH} — build it with "
                "`make -C sptag_amd/csrc` (or __graft_entry__.build()); "
                "this package has no CPU fallback")
    lib = ctypes.CDLL(_LIBPATH)
    lib.sptag_amd_load_index.restype = ctypes.c_void_p
    lib.sptag_amd_load_index.argtypes = [ctypes.c_char_p, ctypes.c_int]
    lib.sptag_amd_create_index.restype = ctypes.c_void_p
    lib.sptag_amd_create_index.argtypes = [
        ctypes.c_int32, ctypes.c_int32, ctypes.c_int, ctypes.c_int,
        ctypes.c_void_p, ctypes.c_int32, ctypes.c_void_p, ctypes.c_int32,
        ctypes.c_void_p, ctypes.c_int32, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_int]
    lib.sptag_amd_create_index_kdt.restype = ctypes.c_void_p
    lib.sptag_amd_create_index_kdt.argtypes = lib.sptag_amd_create_index.argtypes
    lib.sptag_amd_free_index.argtypes = [ctypes.c_void_p]
    lib.sptag_amd_search_batch.restype = ctypes.c_int
    lib.sptag_amd_search_batch.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int32, ctypes.c_int32,
        ctypes.c_int32, ctypes.c_void_p, ctypes.c_void_p]
    lib.sptag_amd_search_batch_device.restype = ctypes.c_int
    lib.sptag_amd_search_batch_device.argtypes = lib.sptag_amd_search_batch.argtypes
    lib.sptag_amd_last_stats.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_double),
        ctypes.POINTER(ctypes.c_longlong), ctypes.POINTER(ctypes.c_longlong)]
    lib.sptag_amd_truth.restype = ctypes.c_int
    lib.sptag_amd_truth.argtypes = lib.sptag_amd_search_batch.argtypes[:4] + [
        ctypes.c_void_p, ctypes.c_void_p]
    lib.sptag_amd_save_index.restype = ctypes.c_int
    lib.sptag_amd_save_index.argtypes = [ctypes.c_void_p, ctypes.c_char_p]
    for f in ["num_vectors", "dim", "valuetype", "distmethod", "degree",
              "default_maxcheck", "algo"]:
        fn = getattr(lib, "sptag_amd_" + f)
        fn.restype = ctypes.c_int32
        fn.argtypes = [ctypes.c_void_p]
    lib.sptag_amd_add.restype = ctypes.c_int
    lib.sptag_amd_add.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                  ctypes.c_int32, ctypes.c_int]
    lib.sptag_amd_delete.restype = ctypes.c_int
    lib.sptag_amd_delete.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                     ctypes.c_int32]
    lib.sptag_amd_delete_by_vector.restype = ctypes.c_int
    lib.sptag_amd_delete_by_vector.argtypes = lib.sptag_amd_delete.argtypes
    lib.sptag_amd_deleted_count.restype = ctypes.c_int64
    lib.sptag_amd_deleted_count.argtypes = [ctypes.c_void_p]
    lib.sptag_amd_iter_create.restype = ctypes.c_void_p
    lib.sptag_amd_iter_create.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                          ctypes.c_int32, ctypes.c_int32]
    lib.sptag_amd_iter_next.restype = ctypes.c_int
    lib.sptag_amd_iter_next.argtypes = [ctypes.c_void_p, ctypes.c_int32,
                                        ctypes.c_void_p, ctypes.c_void_p,
                                        ctypes.c_void_p, ctypes.c_void_p]
    lib.sptag_amd_iter_free.argtypes = [ctypes.c_void_p]
    lib.sptag_amd_set_search_params.restype = None
    lib.sptag_amd_set_search_params.argtypes = [
        ctypes.c_void_p, ctypes.c_int32, ctypes.c_int32, ctypes.c_int32,
        ctypes.c_int32]
    lib.sptag_amd_gpu_available.restype = ctypes.c_int
    lib.sptag_amd_build_info.restype = ctypes.c_char_p
    _LIB = lib
    return lib


def gpu_available():
    return bool(load_library().sptag_amd_gpu_available())


def _np_dtype(vt):
    return np.float32 if vt == VT_FLOAT else np.int8


class AnnIndex:
    """Mirror of the reference Python AnnIndex facade over the GPU backend.

    Reference method -> here:
      AnnIndex.Load(folder)            -> AnnIndex.Load(folder)
      AnnIndex.Search(query, k)        -> .Search(query, k)
      AnnIndex.BatchSearch(q, n, k, _) -> .BatchSearch(queries, k)
      (SaveIndex)                      -> .Save(folder)
    """

    def __init__(self, handle, owns=True):
        self._lib = load_library()
        self._h = handle
        self._owns = owns
        if not handle:
            raise SptagAmdError(-1, "index load/create failed (see stderr)")

    # -- construction -------------------------------------------------
    @classmethod
    def Load(cls, folder, device=0):
        lib = load_library()
        h = lib.sptag_amd_load_index(str(folder).encode(), device)
        ix = cls(h)
        ix.metadata = MetadataSet.from_index_folder(str(folder))
        return ix

    @classmethod
    def FromArrays(cls, vectors, tree_start, tree_nodes, graph, distmethod,
                   deleted=None, device=0):
        """Assemble from raw numpy blobs (file layouts without headers)."""
        lib = load_library()
        vectors = np.ascontiguousarray(vectors)
        vt = VT_FLOAT if vectors.dtype == np.float32 else VT_INT8
        if vectors.dtype not in (np.float32, np.int8):
            raise ValueError("dtype must be float32 or int8")
        n, dim = vectors.shape
        tree_start = np.ascontiguousarray(tree_start, dtype=np.int32)
        tree_nodes = np.ascontiguousarray(tree_nodes, dtype=np.int32)
        graph = np.ascontiguousarray(graph, dtype=np.int32)
        dm = {"L2": DM_L2, "Cosine": DM_COSINE}.get(distmethod, distmethod)
        delp = None
        if deleted is not None:
            deleted = np.ascontiguousarray(deleted, dtype=np.uint8)
            delp = deleted.ctypes.data_as(ctypes.c_void_p)
        h = lib.sptag_amd_create_index(
            n, dim, vt, dm, vectors.ctypes.data_as(ctypes.c_void_p),
            len(tree_start), tree_start.ctypes.data_as(ctypes.c_void_p),
            tree_nodes.size // 3, tree_nodes.ctypes.data_as(ctypes.c_void_p),
            graph.shape[1], graph.ctypes.data_as(ctypes.c_void_p),
            delp, device)
        return cls(h)

    @classmethod
    def FromArraysKDT(cls, vectors, tree_start, kdt_nodes, graph, distmethod,
                      deleted=None, device=0):
        """KDT variant: kdt_nodes is an int32 [N,4] array whose 4th column
        holds the float split_value bit pattern (KDTree.h:22)."""
        lib = load_library()
        vectors = np.ascontiguousarray(vectors)
        vt = VT_FLOAT if vectors.dtype == np.float32 else VT_INT8
        n, dim = vectors.shape
        tree_start = np.ascontiguousarray(tree_start, dtype=np.int32)
        kdt_nodes = np.ascontiguousarray(kdt_nodes, dtype=np.int32)
        graph = np.ascontiguousarray(graph, dtype=np.int32)
        dm = {"L2": DM_L2, "Cosine": DM_COSINE}.get(distmethod, distmethod)
        delp = None
        if deleted is not None:
            deleted = np.ascontiguousarray(deleted, dtype=np.uint8)
            delp = deleted.ctypes.data_as(ctypes.c_void_p)
        h = lib.sptag_amd_create_index_kdt(
            n, dim, vt, dm, vectors.ctypes.data_as(ctypes.c_void_p),
            len(tree_start), tree_start.ctypes.data_as(ctypes.c_void_p),
            kdt_nodes.size // 4, kdt_nodes.ctypes.data_as(ctypes.c_void_p),
            graph.shape[1], graph.ctypes.data_as(ctypes.c_void_p),
            delp, device)
        return cls(h)

    def __del__(self):
        if getattr(self, "_owns", False) and getattr(self, "_h", None):
            self._lib.sptag_amd_free_index(self._h)
            self._h = None

    # -- metadata -----------------------------------------------------
    @property
    def n(self):
        return self._lib.sptag_amd_num_vectors(self._h)

    @property
    def dim(self):
        return self._lib.sptag_amd_dim(self._h)

    @property
    def valuetype(self):
        return self._lib.sptag_amd_valuetype(self._h)

    @property
    def distmethod(self):
        return self._lib.sptag_amd_distmethod(self._h)

    @property
    def algo(self):
        """0 = BKT, 1 = KDT (from the loaded indexloader.ini)."""
        return self._lib.sptag_amd_algo(self._h)

    @property
    def degree(self):
        return self._lib.sptag_amd_degree(self._h)

    @property
    def default_maxcheck(self):
        return self._lib.sptag_amd_default_maxcheck(self._h)

    # -- search -------------------------------------------------------
    def BatchSearch(self, queries, k, max_check=0):
        """GPU batched search. Returns (vids int32 [nq,k], dists f32 [nq,k]),
        ascending by (dist, vid), vid=-1 padding — the contract of the
        reference batch overload VectorIndex.h:103."""
        queries = np.ascontiguousarray(queries, dtype=_np_dtype(self.valuetype))
        if queries.ndim == 1:
            queries = queries[None, :]
        nq = queries.shape[0]
        assert queries.shape[1] == self.dim, (queries.shape, self.dim)
        vids = np.empty((nq, k), dtype=np.int32)
        dists = np.empty((nq, k), dtype=np.float32)
        rc = self._lib.sptag_amd_search_batch(
            self._h, queries.ctypes.data_as(ctypes.c_void_p), nq, k, max_check,
            vids.ctypes.data_as(ctypes.c_void_p),
            dists.ctypes.data_as(ctypes.c_void_p))
        if rc != 0:
            raise SptagAmdError(rc, "search_batch")
        return vids, dists

    def BatchSearchDevice(self, d_queries_ptr, nq, k, d_vids_ptr, d_dists_ptr,
                          max_check=0):
        """Device-resident search: all pointers are HIP device addresses
        (e.g. torch cuda tensors' .data_ptr()); no PCIe copies inside."""
        rc = self._lib.sptag_amd_search_batch_device(
            self._h, ctypes.c_void_p(d_queries_ptr), nq, k, max_check,
            ctypes.c_void_p(d_vids_ptr), ctypes.c_void_p(d_dists_ptr))
        if rc != 0:
            raise SptagAmdError(rc, "search_batch_device")

    def LastStats(self):
        """(kernel_ms, checked, popped) of the last search call."""
        ms = ctypes.c_double()
        ch = ctypes.c_longlong()
        po = ctypes.c_longlong()
        self._lib.sptag_amd_last_stats(self._h, ctypes.byref(ms),
                                       ctypes.byref(ch), ctypes.byref(po))
        return ms.value, ch.value, po.value

    def Search(self, query, k, max_check=0):
        vids, dists = self.BatchSearch(query, k, max_check)
        return vids[0], dists[0]

    def BatchSearchWithMeta(self, queries, k, max_check=0):
        """reference AnnIndex.BatchSearchWithMetaData: results plus the
        per-result metadata blobs (empty when the index has none)."""
        vids, dists = self.BatchSearch(queries, k, max_check)
        meta = getattr(self, "metadata", None)
        blobs = [[meta.get(int(v)) if meta else b"" for v in row]
                 for row in vids]
        return vids, dists, blobs

    def Truth(self, queries, k):
        """Exact brute-force top-k on the GPU."""
        queries = np.ascontiguousarray(queries, dtype=_np_dtype(self.valuetype))
        if queries.ndim == 1:
            queries = queries[None, :]
        nq = queries.shape[0]
        vids = np.empty((nq, k), dtype=np.int32)
        dists = np.empty((nq, k), dtype=np.float32)
        rc = self._lib.sptag_amd_truth(
            self._h, queries.ctypes.data_as(ctypes.c_void_p), nq, k,
            vids.ctypes.data_as(ctypes.c_void_p),
            dists.ctypes.data_as(ctypes.c_void_p))
        if rc != 0:
            raise SptagAmdError(rc, "truth")
        return vids, dists

    def Add(self, vectors, normalized=False):
        """Online add (reference AddIndex semantics below the tree-rebuild
        threshold); runs GPU refine searches per added vector."""
        vectors = np.ascontiguousarray(vectors,
                                       dtype=_np_dtype(self.valuetype))
        if vectors.ndim == 1:
            vectors = vectors[None, :]
        rc = self._lib.sptag_amd_add(
            self._h, vectors.ctypes.data_as(ctypes.c_void_p),
            vectors.shape[0], 1 if normalized else 0)
        if rc != 0:
            raise SptagAmdError(rc, "add")

    def Delete(self, vids):
        """Flag vector ids as deleted (reference DeleteIndex semantics);
        searches exclude them from then on."""
        vids = np.ascontiguousarray(vids, dtype=np.int32)
        rc = self._lib.sptag_amd_delete(
            self._h, vids.ctypes.data_as(ctypes.c_void_p), vids.size)
        if rc != 0:
            raise SptagAmdError(rc, "delete")

    def DeleteByVector(self, vectors):
        """Delete exact duplicates of the given vectors (reference
        DeleteIndex(const void*, n) semantics: CEF search + dist < 1e-6)."""
        vectors = np.ascontiguousarray(vectors,
                                       dtype=_np_dtype(self.valuetype))
        if vectors.ndim == 1:
            vectors = vectors[None, :]
        rc = self._lib.sptag_amd_delete_by_vector(
            self._h, vectors.ctypes.data_as(ctypes.c_void_p), vectors.shape[0])
        if rc != 0:
            raise SptagAmdError(rc, "delete_by_vector")

    @property
    def deleted_count(self):
        return self._lib.sptag_amd_deleted_count(self._h)

    def SetSearchParams(self, init_pivots=0, other_pivots=0,
                        nobetter_threshold=0, max_check=0):
        """Reference SetParameter equivalents (NumberOfInitialDynamicPivots
        etc., BKTIndex.cpp:980); pass 0 to keep a value."""
        self._lib.sptag_amd_set_search_params(
            self._h, init_pivots, other_pivots, nobetter_threshold, max_check)

    def Iterate(self, queries, max_check=0):
        """Streaming search: mirrors the reference GetIterator/Next protocol
        (ResultIterator) for a batch of queries with device-persistent
        traversal state."""
        return IterBatch(self, queries, max_check)

    def Save(self, folder):
        os.makedirs(folder, exist_ok=True)
        rc = self._lib.sptag_amd_save_index(self._h, str(folder).encode())
        if rc != 0:
            raise SptagAmdError(rc, "save_index")


class MetadataSet:
    """Reference MemMetadataSet file pair (src/Core/MetadataSet.cpp:269-283:
    metadataIndex = [int32 count][uint64 offsets x (count+1)]; metadata =
    concatenated byte blobs). Post-search VID -> bytes lookup — host-side
    only, exactly as the reference's SearchIndex metadata copy
    (BKTIndex.cpp:611-618)."""

    def __init__(self, meta_file, metaindex_file):
        with open(metaindex_file, "rb") as f:
            self.count = int(np.frombuffer(f.read(4), dtype=np.int32)[0])
            self.offsets = np.frombuffer(f.read(8 * (self.count + 1)),
                                         dtype=np.uint64)
        self.blob = open(meta_file, "rb").read()

    def get(self, vid):
        if vid < 0 or vid >= self.count:
            return b""
        return self.blob[int(self.offsets[vid]):int(self.offsets[vid + 1])]

    @classmethod
    def from_index_folder(cls, folder):
        """Load the pair named by indexloader.ini's [MetaData] section
        (VectorIndex.cpp:618 LoadIndex), or None when the index has none."""
        ini = os.path.join(folder, "indexloader.ini")
        if not os.path.exists(ini):
            return None
        meta = metaidx = None
        section = None
        for line in open(ini):
            line = line.strip()
            if line.startswith("["):
                section = line
            elif section == "[MetaData]" and "=" in line:
                key, val = line.split("=", 1)
                if key == "MetaDataFilePath":
                    meta = val
                elif key == "MetaDataIndexPath":
                    metaidx = val
        if not meta or not metaidx:
            return None
        return cls(os.path.join(folder, meta), os.path.join(folder, metaidx))


class IterBatch:
    """Batch of per-query result iterators (reference ResultIterator
    semantics; state persists on the GPU between Next calls)."""

    def __init__(self, index, queries, max_check=0):
        self._lib = index._lib
        self._index = index   # keep the index alive
        queries = np.ascontiguousarray(queries,
                                       dtype=_np_dtype(index.valuetype))
        if queries.ndim == 1:
            queries = queries[None, :]
        self.nq = queries.shape[0]
        self._h = self._lib.sptag_amd_iter_create(
            index._h, queries.ctypes.data_as(ctypes.c_void_p), self.nq,
            max_check)
        if not self._h:
            raise SptagAmdError(-6, "iter_create failed (see stderr)")

    def Next(self, batch):
        """Returns (vids [nq,batch], dists, counts [nq], relaxed [nq]):
        the next `counts[i]` nearest results of query i, sorted; vid=-1
        padding; relaxed = the sticky relaxed-monotonicity flag."""
        vids = np.empty((self.nq, batch), dtype=np.int32)
        dists = np.empty((self.nq, batch), dtype=np.float32)
        counts = np.empty(self.nq, dtype=np.int32)
        relaxed = np.empty(self.nq, dtype=np.int32)
        rc = self._lib.sptag_amd_iter_next(
            self._h, batch, vids.ctypes.data_as(ctypes.c_void_p),
            dists.ctypes.data_as(ctypes.c_void_p),
            counts.ctypes.data_as(ctypes.c_void_p),
            relaxed.ctypes.data_as(ctypes.c_void_p))
        if rc != 0:
            raise SptagAmdError(rc, "iter_next")
        return vids, dists, counts, relaxed

    def Close(self):
        if getattr(self, "_h", None):
            self._lib.sptag_amd_iter_free(self._h)
            self._h = None

    def __del__(self):
        self.Close()
