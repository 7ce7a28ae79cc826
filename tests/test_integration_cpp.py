"""The compiled reference-side binding (integration/amd_bkt_index.cpp):
a SPTAG::VectorIndex subclass over the GPU C-ABI, driven through
base-pointer virtual dispatch by oracle/_ref/amd_searcher.

CPU (this container): the binary exists (built against the reference
headers by oracle/Makefile), loads a golden index through the virtual
interface, and fails LOUDLY without a GPU — no CPU fallback.

GPU: its results are byte-identical to the reference CPU searcher's
recorded answers (tests/golden/*/results_mc*.bin), and the per-query
virtual (VectorIndex.h:41) equals the one-launch batch virtual
(VectorIndex.h:103) — checked inside amd_searcher itself.
"""
import os
import subprocess

import numpy as np
import pytest

from conftest import GOLDEN, REPO, golden_fixtures, load_golden

AMD_SEARCHER = os.path.join(REPO, "oracle", "_ref", "amd_searcher")

pytestmark = pytest.mark.skipif(not os.path.exists(AMD_SEARCHER),
                                reason="amd_searcher not built "
                                       "(requires /root/reference)")


def _fixture_or_skip():
    names = [n for n in golden_fixtures() if n.startswith("f32_l2_n")]
    if not names:
        pytest.skip("no golden fixtures")
    return load_golden(names[0])


def test_loads_and_fails_loudly_without_gpu(tmp_path):
    import sptag_amd
    g = _fixture_or_skip()
    out = subprocess.run(
        [AMD_SEARCHER, g["index"], os.path.join(g["dir"], "queries.bin"),
         str(g["meta"]["k"]), "512", str(tmp_path / "out.bin")],
        capture_output=True, text=True, timeout=300)
    if sptag_amd.gpu_available():
        assert out.returncode == 0, out.stderr
    else:
        assert out.returncode != 0
        assert "no CPU fallback" in out.stderr
        assert "load failed" not in out.stderr


@pytest.mark.gpu
@pytest.mark.parametrize("name", [n for n in golden_fixtures()
                                  if not n.startswith("kdt")])
def test_virtual_dispatch_matches_reference_bytes(name, tmp_path):
    """amd_searcher output == the reference CPU searcher's recorded bytes,
    for every BKT golden fixture and every recorded MaxCheck."""
    g = load_golden(name)
    for mc in g["meta"]["maxchecks"]:
        outfile = str(tmp_path / f"out_{mc}.bin")
        out = subprocess.run(
            [AMD_SEARCHER, g["index"], os.path.join(g["dir"], "queries.bin"),
             str(g["meta"]["k"]), str(mc), outfile],
            capture_output=True, text=True, timeout=600)
        assert out.returncode == 0, out.stderr
        got = open(outfile, "rb").read()
        ref = open(os.path.join(g["dir"], f"results_mc{mc}.bin"), "rb").read()
        assert got == ref, f"{name} mc={mc}: byte mismatch"


@pytest.mark.gpu
def test_kdt_virtual_dispatch(tmp_path):
    names = [n for n in golden_fixtures() if n.startswith("kdt")]
    if not names:
        pytest.skip("no KDT fixtures")
    g = load_golden(names[0])
    for mc in g["meta"]["maxchecks"]:
        outfile = str(tmp_path / f"out_{mc}.bin")
        out = subprocess.run(
            [AMD_SEARCHER, g["index"], os.path.join(g["dir"], "queries.bin"),
             str(g["meta"]["k"]), str(mc), outfile],
            capture_output=True, text=True, timeout=600)
        assert out.returncode == 0, out.stderr
        got = open(outfile, "rb").read()
        ref = open(os.path.join(g["dir"], f"results_mc{mc}.bin"), "rb").read()
        assert got == ref
