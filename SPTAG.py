"""Drop-in `import SPTAG` compatibility shim over the MI355X backend.

Mirrors the reference SWIG wrapper's documented Python surface
(reference Wrappers/inc/CoreInterface.h:14 + docs/GettingStart.md:388-460):

    import SPTAG
    i = SPTAG.AnnIndex('BKT', 'Float', dim)
    i.SetBuildParam("DistCalcMethod", "L2", "Index")
    i.Build(x, x.shape[0], False)        # GPU builder + GPU-resident index
    i.Save(out)
    j = SPTAG.AnnIndex.Load(out)
    j.SetSearchParam("MaxCheck", "1024", "Index")
    ids, dists = j.Search(q, k)
    ids, dists, metas = j.SearchWithMetaData(q, k)
    i.Add(x2, x2.shape[0], False)
    i.Delete(xdel, xdel.shape[0])
    it = j.GetIterator(q)                # ResultIterator protocol
    ids, dists = it.Next(batch)

Vectors may be numpy arrays or raw bytes (the SWIG ByteArray form); bytes
are reinterpreted using the index's value type and dimension. SPANN,
quantizers and dump/merge are out of the hot-path scope and raise.
"""
import os
import sys

import numpy as np

_REPO = os.path.dirname(os.path.abspath(__file__))
if _REPO not in sys.path:
    sys.path.insert(0, _REPO)

import sptag_amd as _backend  # noqa: E402

__all__ = ["AnnIndex", "ResultIterator"]

_VTMAP = {"Float": np.float32, "Int8": np.int8}


class ResultIterator:
    """reference ResultIterator (inc/Core/ResultIterator.h) over one query."""

    def __init__(self, iter_batch):
        self._it = iter_batch
        self._relaxed = False

    def Next(self, batch):
        vids, dists, counts, relaxed = self._it.Next(batch)
        self._relaxed = bool(relaxed[0])
        n = int(counts[0])
        return vids[0][:n].tolist(), dists[0][:n].tolist()

    def GetRelaxedMono(self):
        return self._relaxed

    def Close(self):
        self._it.Close()


class AnnIndex:
    def __init__(self, p_algoType="BKT", p_valueType="Float", p_dimension=0):
        if p_algoType not in ("BKT", "KDT"):
            raise NotImplementedError(
                f"algo {p_algoType} outside the hot-path scope (BKT/KDT only)")
        if p_valueType not in _VTMAP:
            raise NotImplementedError(f"value type {p_valueType} unsupported")
        self._algo = p_algoType
        self._vt = p_valueType
        self._dim = int(p_dimension)
        self._params = {"DistCalcMethod": "Cosine"}  # reference default
        self._mc = 0
        self._ix = None

    # -- parameters (SetParameter semantics; the build/search split follows
    # the reference wrapper) --
    def SetBuildParam(self, name, value, section="Index"):
        self._params[name] = value

    def SetSearchParam(self, name, value, section="Index"):
        if name == "MaxCheck":
            self._mc = int(value)
        else:
            self._params[name] = value

    def _as_array(self, data, num):
        dtype = _VTMAP[self._vt]
        if isinstance(data, (bytes, bytearray, memoryview)):
            arr = np.frombuffer(data, dtype=dtype)
            return arr.reshape(num, -1) if num > 0 else arr.reshape(1, -1)
        arr = np.ascontiguousarray(data, dtype=dtype)
        return arr.reshape(num, -1) if arr.ndim == 1 and num > 1 else \
            (arr[None, :] if arr.ndim == 1 else arr)

    # -- build --
    def Build(self, p_data, p_num, p_normalized=False):
        from sptag_amd.build import build_index_arrays
        x = self._as_array(p_data, p_num)
        self._dim = x.shape[1]
        dist = self._params.get("DistCalcMethod", "Cosine")
        arrays = build_index_arrays(
            x, dist, algo=self._algo,
            degree=int(self._params.get("NeighborhoodSize", 32)),
            ntrees=int(self._params.get("TPTNumber", 32)) // 8 or 4,
            normalized=bool(p_normalized))
        if self._algo == "KDT":
            self._ix = _backend.AnnIndex.FromArraysKDT(
                arrays["vectors"], arrays["tree_start"], arrays["tree_nodes"],
                arrays["graph"], dist)
        else:
            self._ix = _backend.AnnIndex.FromArrays(
                arrays["vectors"], arrays["tree_start"], arrays["tree_nodes"],
                arrays["graph"], dist)
        return True

    def BuildWithMetaData(self, p_data, p_meta, p_num, p_withMetaIndex=False,
                          p_normalized=False):
        raise NotImplementedError("metadata-at-build lands with the metadata "
                                  "writer; attach metadata files to the saved "
                                  "folder and reload")

    # -- search --
    def _need(self):
        if self._ix is None:
            raise RuntimeError("index not built/loaded")
        return self._ix

    def Search(self, p_data, p_resultNum):
        q = self._as_array(p_data, 1)
        vids, dists = self._need().BatchSearch(q, p_resultNum, self._mc)
        return vids[0].tolist(), dists[0].tolist()

    def SearchWithMetaData(self, p_data, p_resultNum):
        q = self._as_array(p_data, 1)
        vids, dists, metas = self._need().BatchSearchWithMeta(
            q, p_resultNum, self._mc)
        return vids[0].tolist(), dists[0].tolist(), metas[0]

    def BatchSearch(self, p_data, p_vectorNum, p_resultNum, p_withMetaData=False):
        q = self._as_array(p_data, p_vectorNum)
        if p_withMetaData:
            return self._need().BatchSearchWithMeta(q, p_resultNum, self._mc)
        return self._need().BatchSearch(q, p_resultNum, self._mc)

    def GetIterator(self, p_target):
        q = self._as_array(p_target, 1)
        return ResultIterator(self._need().Iterate(q, self._mc))

    # -- updates --
    def Add(self, p_data, p_num, p_normalized=False):
        self._need().Add(self._as_array(p_data, p_num), bool(p_normalized))
        return True

    def Delete(self, p_data, p_num):
        self._need().DeleteByVector(self._as_array(p_data, p_num))
        return True

    # -- persistence --
    def Save(self, p_saveFile):
        self._need().Save(p_saveFile)
        return True

    @staticmethod
    def Load(p_loaderFile):
        ix = _backend.AnnIndex.Load(p_loaderFile)
        out = AnnIndex("KDT" if ix.algo == 1 else "BKT", "Float", ix.dim)
        out._vt = "Float" if ix.valuetype == _backend.VT_FLOAT else "Int8"
        out._dim = ix.dim
        out._ix = ix
        # reflect the loaded index's metric so later Build()/introspection
        # on this handle uses the right default
        out._params["DistCalcMethod"] = (
            "L2" if ix.distmethod == _backend.DM_L2 else "Cosine")
        return out

    def ReadyToServe(self):
        return self._ix is not None

    # -- out-of-scope reference surface: fail loudly, not silently --
    def BuildSPANN(self, *a, **k):
        raise NotImplementedError("SPANN is outside the hot-path scope")

    def LoadQuantizer(self, *a, **k):
        raise NotImplementedError("quantizers are outside the hot-path scope")
