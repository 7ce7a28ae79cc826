"""Distance unit tests — the reference DistanceTest.cpp:36-50 analog:
the backend's distance must match a plain scalar reference within 1e-5
relative for float, and exactly for int8 (integer distances are exact in
the reference — SURVEY.md §8a row a4)."""
import ctypes

import numpy as np
import pytest

from oracle import pyoracle


def scalar_l2(x, y):
    d = x.astype(np.float64) - y.astype(np.float64)
    return float((d * d).sum())


def scalar_cos(x, y, base):
    return base * base - float((x.astype(np.float64) * y.astype(np.float64)).sum())


@pytest.mark.parametrize("dim", [2, 3, 10, 16, 31, 32, 48, 100, 128, 200, 256, 768])
@pytest.mark.parametrize("dm", [0, 1])
def test_f32_distance_close(dim, dm):
    rng = np.random.default_rng(dim * 10 + dm)
    lib = pyoracle.load_library()
    for _ in range(20):
        x = rng.random(dim, dtype=np.float32)
        y = rng.random(dim, dtype=np.float32)
        got = lib.orc_distance(0, dm, x.ctypes.data_as(ctypes.c_void_p),
                               y.ctypes.data_as(ctypes.c_void_p), dim)
        want = scalar_l2(x, y) if dm == 0 else 1.0 - float(
            (x.astype(np.float64) * y.astype(np.float64)).sum())
        assert got == pytest.approx(want, rel=1e-5, abs=1e-6)


@pytest.mark.parametrize("dim", [2, 4, 10, 100, 128, 256])
@pytest.mark.parametrize("dm", [0, 1])
def test_i8_distance_exact(dim, dm):
    rng = np.random.default_rng(dim * 10 + dm)
    lib = pyoracle.load_library()
    for _ in range(20):
        x = rng.integers(-128, 128, dim).astype(np.int8)
        y = rng.integers(-128, 128, dim).astype(np.int8)
        got = lib.orc_distance(1, dm, x.ctypes.data_as(ctypes.c_void_p),
                               y.ctypes.data_as(ctypes.c_void_p), dim)
        want = scalar_l2(x, y) if dm == 0 else scalar_cos(x, y, 127)
        assert got == want  # exact integers
