"""C-ABI surface tests (CPU-side): the HIP library loads, exports every
symbol include/sptag_amd.h declares, index folders load without a GPU
(metadata + save), search fails LOUDLY without a GPU (no CPU fallback),
and save_index writes byte-identical binary files."""
import os
import re

import numpy as np
import pytest

from conftest import REPO, load_golden
import sptag_amd
from sptag_amd import AnnIndex, SptagAmdError


def declared_symbols():
    hdr = open(os.path.join(REPO, "include", "sptag_amd.h")).read()
    return re.findall(r"\b(sptag_amd_\w+)\s*\(", hdr)


def test_library_exports_all_header_symbols():
    lib = sptag_amd.load_library()
    syms = set(declared_symbols())
    assert len(syms) >= 12
    for s in syms:
        assert hasattr(lib, s), f"missing export {s}"


def test_load_metadata_without_gpu():
    g = load_golden("f32_l2_n10k_d32")
    ix = AnnIndex.Load(g["index"])
    assert ix.n == g["meta"]["n"]
    assert ix.dim == g["meta"]["dim"]
    assert ix.valuetype == sptag_amd.VT_FLOAT
    assert ix.distmethod == sptag_amd.DM_L2
    assert ix.degree == 32
    assert ix.default_maxcheck == 8192


def test_search_fails_loudly_without_gpu():
    if sptag_amd.gpu_available():
        pytest.skip("GPU present; loud-failure path not reachable")
    g = load_golden("f32_l2_n10k_d32")
    ix = AnnIndex.Load(g["index"])
    with pytest.raises(SptagAmdError) as e:
        ix.BatchSearch(g["queries"][:2], 10)
    assert e.value.code == -3  # NOGPU


def test_save_roundtrip_bytes(tmp_path):
    """Save must reproduce the reference's byte format (vectors/tree/graph
    identical; deletes semantically equal; ini reloadable)."""
    g = load_golden("i8_cos_n20k_d100")
    ix = AnnIndex.Load(g["index"])
    out = tmp_path / "saved"
    ix.Save(str(out))
    for f in ["vectors.bin", "tree.bin", "graph.bin"]:
        a = open(os.path.join(g["index"], f), "rb").read()
        b = open(out / f, "rb").read()
        assert a == b, f"{f} differs after save"
    ix2 = AnnIndex.Load(str(out))
    assert ix2.n == ix.n and ix2.dim == ix.dim
    assert ix2.distmethod == ix.distmethod

    # the oracle (pinned to the reference loader) must accept our folder too
    from oracle.pyoracle import OrcIndex
    oix = OrcIndex.load(str(out))
    assert oix.n == ix.n and oix.distmethod == ix.distmethod


@pytest.mark.skipif(
    not os.path.exists(os.path.join(REPO, "oracle", "_ref", "indexsearcher")),
    reason="reference binaries not built in this environment")
def test_reference_searcher_accepts_saved_folder(tmp_path):
    """Drop-in proof: the REFERENCE indexsearcher runs against a folder
    written by sptag_amd_save_index and reproduces its golden results."""
    import subprocess
    g = load_golden("f32_l2_n10k_d32")
    ix = AnnIndex.Load(g["index"])
    out = tmp_path / "saved"
    ix.Save(str(out))
    res = tmp_path / "res.bin"
    subprocess.run(
        [os.path.join(REPO, "oracle", "_ref", "indexsearcher"),
         "-d", str(ix.dim), "-v", "Float", "-f", "DEFAULT",
         "-i", os.path.join(g["dir"], "queries.bin"), "-x", str(out),
         "-k", "10", "-m", "2048", "-t", "2", "-of", "1", "-o", str(res)],
        check=True, capture_output=True, cwd=tmp_path)
    raw = open(res, "rb").read()
    rec = np.frombuffer(raw[8:], dtype=np.dtype([("vid", np.int32),
                                                 ("dist", np.float32)]))
    rec = rec.reshape(g["meta"]["nq"], 10)
    ref = g["results"][2048]
    np.testing.assert_array_equal(rec["vid"], ref["vid"])
    np.testing.assert_array_equal(rec["dist"], ref["dist"])


def test_kdt_save_roundtrip(tmp_path):
    """KDT index folders round-trip byte-identically and stay loadable by
    the oracle (pinned to the reference loader)."""
    g = load_golden("kdt_f32_l2_n10k_d32")
    ix = AnnIndex.Load(g["index"])
    out = tmp_path / "saved"
    ix.Save(str(out))
    for f in ["vectors.bin", "tree.bin", "graph.bin"]:
        a = open(os.path.join(g["index"], f), "rb").read()
        b = open(out / f, "rb").read()
        assert a == b, f"{f} differs after KDT save"
    from oracle.pyoracle import OrcIndex
    oix = OrcIndex.load(str(out))
    assert oix.n == ix.n


@pytest.mark.skipif(
    not os.path.exists(os.path.join(REPO, "oracle", "_ref", "indexsearcher")),
    reason="reference binaries not built in this environment")
def test_reference_searcher_accepts_saved_kdt_folder(tmp_path):
    import subprocess
    g = load_golden("kdt_f32_l2_n10k_d32")
    ix = AnnIndex.Load(g["index"])
    out = tmp_path / "saved"
    ix.Save(str(out))
    res = tmp_path / "res.bin"
    subprocess.run(
        [os.path.join(REPO, "oracle", "_ref", "indexsearcher"),
         "-d", str(ix.dim), "-v", "Float", "-f", "DEFAULT",
         "-i", os.path.join(g["dir"], "queries.bin"), "-x", str(out),
         "-k", "10", "-m", "2048", "-t", "2", "-of", "1", "-o", str(res)],
        check=True, capture_output=True, cwd=tmp_path)
    raw = open(res, "rb").read()
    rec = np.frombuffer(raw[8:], dtype=np.dtype([("vid", np.int32),
                                                 ("dist", np.float32)]))
    rec = rec.reshape(g["meta"]["nq"], 10)
    ref = g["results"][2048]
    np.testing.assert_array_equal(rec["vid"], ref["vid"])
    np.testing.assert_array_equal(rec["dist"], ref["dist"])


def test_error_paths(tmp_path):
    """C-ABI failure modes are loud, not silent."""
    import sptag_amd as sa
    lib = sa.load_library()
    assert lib.sptag_amd_load_index(str(tmp_path / "nonexistent").encode(), 0) is None
    # unsupported algo in ini
    d = tmp_path / "bad"
    d.mkdir()
    (d / "indexloader.ini").write_text("[Index]\nIndexAlgoType=SPANN\n")
    assert lib.sptag_amd_load_index(str(d).encode(), 0) is None


def test_metadata_set(tmp_path):
    """MemMetadataSet file-pair reader (reference MetadataSet.cpp:269-283)."""
    from sptag_amd import MetadataSet
    blobs = [b"alpha", b"", b"gamma-123"]
    offs = np.zeros(len(blobs) + 1, dtype=np.uint64)
    for i, b in enumerate(blobs):
        offs[i + 1] = offs[i] + len(b)
    (tmp_path / "metadata.bin").write_bytes(b"".join(blobs))
    with open(tmp_path / "metadataIndex.bin", "wb") as f:
        f.write(np.int32(len(blobs)).tobytes())
        f.write(offs.tobytes())
    ms = MetadataSet(str(tmp_path / "metadata.bin"),
                     str(tmp_path / "metadataIndex.bin"))
    assert [ms.get(i) for i in range(3)] == blobs
    assert ms.get(-1) == b"" and ms.get(3) == b""

    # an index folder whose ini names the pair
    import shutil
    g = load_golden("f32_l2_n10k_d32")
    idx2 = tmp_path / "idx"
    shutil.copytree(g["index"], idx2)
    with open(idx2 / "indexloader.ini", "a") as f:
        f.write("\n[MetaData]\nMetaDataFilePath=metadata.bin\n"
                "MetaDataIndexPath=metadataIndex.bin\n")
    shutil.copy(tmp_path / "metadata.bin", idx2 / "metadata.bin")
    shutil.copy(tmp_path / "metadataIndex.bin", idx2 / "metadataIndex.bin")
    from sptag_amd import AnnIndex
    ix = AnnIndex.Load(str(idx2))
    assert ix.metadata is not None
    assert ix.metadata.get(2) == b"gamma-123"


def test_load_tolerant_ini(tmp_path):
    """The ini parser at the drop-in seam accepts what the reference's
    SimpleIniReader accepts: case-insensitive section/key names, spaces
    around '=', ';' comment lines (src/Helper/SimpleIniReader.cpp)."""
    import shutil
    from conftest import golden_fixtures, load_golden
    names = [n for n in golden_fixtures() if n.startswith("f32_l2_n")]
    if not names:
        pytest.skip("no golden fixtures")
    g = load_golden(names[0])
    dst = tmp_path / "index"
    shutil.copytree(g["index"], dst)
    ini = (dst / "indexloader.ini").read_text()
    # mangle: comments, case, whitespace — still reference-legal
    mangled = ["; rewritten by test", "[index]"]
    for line in ini.splitlines():
        line = line.strip()
        if not line or line.startswith("["):
            continue
        if "=" in line:
            k, v = line.split("=", 1)
            mangled.append(f"  {k.lower()} = {v}")
    (dst / "indexloader.ini").write_text("\n".join(mangled) + "\n")
    ix = sptag_amd.AnnIndex.Load(str(dst))
    assert ix.n == g["meta"]["n"]
    assert ix.distmethod == sptag_amd.DM_L2
    assert ix.default_maxcheck == 8192
