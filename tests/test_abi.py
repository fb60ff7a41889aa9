"""CPU-side ABI checks: the HIP engine builds for gfx950, loads, exports every
symbol include/denormalized_amd.h declares, its pure host logic matches the
oracle, and compute entry points fail loudly without a GPU."""
import ctypes
import os
import re

import numpy as np
import pytest

import __graft_entry__ as graft
from oracle import pyoracle

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HEADER = os.path.join(REPO, "include", "denormalized_amd.h")


@pytest.fixture(scope="module")
def built():
    graft.build()
    from denormalized_amd import _lib
    return _lib.lib()


def test_exports_every_header_symbol(built):
    src = open(HEADER).read()
    src = re.sub(r"/\*.*?\*/", "", src, flags=re.S)
    fns = re.findall(r"\b(dz_\w+)\s*\(", src)
    fns = sorted(set(f for f in fns if not f.startswith("dz_window_desc")))
    assert len(fns) >= 15
    for f in fns:
        assert hasattr(built, f), f"missing export {f}"


def test_windows_for_range_matches_oracle(built):
    rng = np.random.default_rng(3)
    ws = np.empty(65536, np.int64)
    we = np.empty(65536, np.int64)
    for _ in range(300):
        len_ms = int(rng.choice([500, 1000, 1500, 2000, 5000, 60_000]))
        slide = int(rng.choice([0, 100, 250, 500, 1000]))
        if slide > len_ms:
            continue
        mn = int(rng.integers(60_000, 10_000_000))
        mx = mn + int(rng.integers(0, 50_000))
        n = built.dz_debug_windows_for_range(
            mn, mx, len_ms, slide,
            ws.ctypes.data_as(ctypes.c_void_p), we.ctypes.data_as(ctypes.c_void_p),
            65536)
        ows, owe = pyoracle.windows_for_range(mn, mx, len_ms, slide)
        assert n == len(ows)
        assert np.array_equal(ws[:n], ows) and np.array_equal(we[:n], owe)


def test_windows_for_range_negative_and_epoch_straddle(built):
    # pre-epoch and epoch-straddling ranges: the engine and the oracle both
    # restate the reference's as_secs() truncation (toward zero), so the
    # snap differs from floor for negative ts — pin that they AGREE there
    ws = np.empty(65536, np.int64)
    we = np.empty(65536, np.int64)
    cases = [
        (-10_000, -1, 1000, 0), (-10_000, 5_000, 1000, 0),
        (-7_500, -2_200, 1500, 0), (-10_000, 2_000, 1000, 250),
        (-5_000, 5_000, 2000, 500), (-999, 999, 500, 100),
        (-60_001, -59_000, 1000, 0), (0, 0, 1000, 0),
    ]
    for mn, mx, len_ms, slide in cases:
        n = built.dz_debug_windows_for_range(
            mn, mx, len_ms, slide,
            ws.ctypes.data_as(ctypes.c_void_p),
            we.ctypes.data_as(ctypes.c_void_p), 65536)
        ows, owe = pyoracle.windows_for_range(mn, mx, len_ms, slide)
        assert n == len(ows), (mn, mx, len_ms, slide)
        assert np.array_equal(ws[:n], ows) and np.array_equal(we[:n], owe), \
            (mn, mx, len_ms, slide)


def test_create_fails_loudly_without_gpu(built):
    try:
        import torch
        if torch.cuda.is_available():
            pytest.skip("GPU present")
    except Exception:
        pass
    from denormalized_amd import WindowOp
    with pytest.raises(RuntimeError, match="no HIP device|create failed"):
        WindowOp(length_ms=1000)


def test_join_and_decoder_fail_loudly_without_gpu(built):
    try:
        import torch
        if torch.cuda.is_available():
            pytest.skip("GPU present")
    except Exception:
        pass
    from denormalized_amd import JoinOp, JsonDecoder
    with pytest.raises(RuntimeError, match="no HIP device|create failed"):
        JoinOp(device=0)
    with pytest.raises(RuntimeError, match="no HIP device|create failed"):
        JsonDecoder(device=0)


def test_version(built):
    assert b"gfx950" in built.dz_version()
