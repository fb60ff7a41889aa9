"""ctypes wrapper over oracle/liboracle.so — TEST INFRASTRUCTURE ONLY.

Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may use
this module (oracle.c header states the policy). The product path never does.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "liboracle.so")


def build():
    """Compile liboracle.so if missing or stale."""
    src = os.path.join(_DIR, "oracle.c")
    if not os.path.exists(_SO) or os.path.getmtime(_SO) < os.path.getmtime(src):
        subprocess.check_call(["make", "-C", _DIR, "-s"])
    return _SO


_lib = None


def lib():
    global _lib
    if _lib is None:
        build()
        _lib = ctypes.CDLL(_SO)
        L = _lib
        i64 = ctypes.c_int64
        u64 = ctypes.c_uint64
        p = ctypes.c_void_p
        L.orc_create.restype = p
        L.orc_create.argtypes = [i64, i64]
        L.orc_destroy.argtypes = [p]
        L.orc_push.argtypes = [p, i64, p, p, p, p]
        L.orc_finish.argtypes = [p]
        L.orc_out_rows.restype = i64
        L.orc_out_rows.argtypes = [p]
        L.orc_out_fetch.argtypes = [p] + [p] * 9
        L.orc_open_frames.restype = i64
        L.orc_open_frames.argtypes = [p]
        L.orc_watermark.restype = i64
        L.orc_watermark.argtypes = [p]
        L.orc_windows_for_range.restype = i64
        L.orc_windows_for_range.argtypes = [i64, i64, i64, i64, p, p, i64]
        L.orc_gen.argtypes = [u64, i64, i64, i64, i64, i64, p, p, p]
    return _lib


def _ptr(a):
    return a.ctypes.data_as(ctypes.c_void_p) if a is not None else None


def windows_for_range(min_ts, max_ts, len_ms, slide_ms=0, cap=65536):
    L = lib()
    ws = np.empty(cap, np.int64)
    we = np.empty(cap, np.int64)
    n = L.orc_windows_for_range(min_ts, max_ts, len_ms, slide_ms, _ptr(ws), _ptr(we), cap)
    assert n <= cap
    return ws[:n].copy(), we[:n].copy()


def gen(seed, t0_ms, start_row, nrows, nkeys, rows_per_ms):
    L = lib()
    ts = np.empty(nrows, np.int64)
    kid = np.empty(nrows, np.int64)
    val = np.empty(nrows, np.float64)
    L.orc_gen(seed, t0_ms, start_row, nrows, nkeys, rows_per_ms,
              _ptr(ts), _ptr(kid), _ptr(val))
    return ts, kid, val


class Oracle:
    """CPU restatement of the reference grouped streaming window aggregate."""

    def __init__(self, window_length_ms, slide_ms=0):
        self._L = lib()
        self._h = self._L.orc_create(window_length_ms, slide_ms)
        if not self._h:
            raise ValueError("bad window config")

    def push(self, ts_ms, keys, vals, val_valid=None):
        ts_ms = np.ascontiguousarray(ts_ms, np.int64)
        keys = np.ascontiguousarray(keys, np.int64)
        vals = np.ascontiguousarray(vals, np.float64)
        n = len(ts_ms)
        assert len(keys) == n and len(vals) == n
        vv = None
        if val_valid is not None:
            vv = np.ascontiguousarray(val_valid, np.uint8)
            assert len(vv) == n
        self._L.orc_push(self._h, n, _ptr(ts_ms), _ptr(keys), _ptr(vals), _ptr(vv))

    def finish(self):
        self._L.orc_finish(self._h)

    def fetch(self):
        """Returns dict of output columns (and clears the queue)."""
        n = self._L.orc_out_rows(self._h)
        out = {
            "key": np.empty(n, np.int64),
            "count": np.empty(n, np.int64),
            "min": np.empty(n, np.float64),
            "max": np.empty(n, np.float64),
            "avg": np.empty(n, np.float64),
            "sum": np.empty(n, np.float64),
            "valid": np.empty(n, np.uint8),
            "window_start": np.empty(n, np.int64),
            "window_end": np.empty(n, np.int64),
        }
        self._L.orc_out_fetch(
            self._h, _ptr(out["key"]), _ptr(out["count"]), _ptr(out["min"]),
            _ptr(out["max"]), _ptr(out["avg"]), _ptr(out["sum"]),
            _ptr(out["valid"]), _ptr(out["window_start"]), _ptr(out["window_end"]))
        return out

    @property
    def open_frames(self):
        return self._L.orc_open_frames(self._h)

    @property
    def watermark(self):
        return self._L.orc_watermark(self._h)

    def close(self):
        if self._h:
            self._L.orc_destroy(self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class JoinOracle:
    """Inner equi-join restatement (oracle.c orc_join_*): build side
    trip_id -> driver_id, probe rows emit in row order, unmatched rows
    buffer and re-emit in original order when their build row arrives."""

    def __init__(self):
        self._L = lib()
        L = self._L
        import ctypes
        p = ctypes.c_void_p
        i64 = ctypes.c_int64
        if not getattr(L, "_join_boot", False):
            L.orc_join_create.restype = p
            L.orc_join_destroy.argtypes = [p]
            L.orc_join_push_build.argtypes = [p, i64, p, p]
            L.orc_join_push_probe.argtypes = [p, i64, p, p, p]
            L.orc_join_out_rows.restype = i64
            L.orc_join_out_rows.argtypes = [p]
            L.orc_join_unmatched.restype = i64
            L.orc_join_unmatched.argtypes = [p]
            L.orc_join_out_fetch.argtypes = [p, p, p, p]
            L._join_boot = True
        self._h = L.orc_join_create()

    def push_build(self, trips, drivers):
        import numpy as np
        trips = np.ascontiguousarray(trips, np.int64)
        drivers = np.ascontiguousarray(drivers, np.int64)
        self._L.orc_join_push_build(self._h, len(trips), _ptr(trips),
                                    _ptr(drivers))

    def push_probe(self, ts, trips, vals):
        import numpy as np
        ts = np.ascontiguousarray(ts, np.int64)
        trips = np.ascontiguousarray(trips, np.int64)
        vals = np.ascontiguousarray(vals, np.float64)
        self._L.orc_join_push_probe(self._h, len(ts), _ptr(ts), _ptr(trips),
                                    _ptr(vals))

    @property
    def unmatched(self):
        return self._L.orc_join_unmatched(self._h)

    def fetch(self):
        import numpy as np
        n = self._L.orc_join_out_rows(self._h)
        ts = np.empty(n, np.int64)
        drv = np.empty(n, np.int64)
        val = np.empty(n, np.float64)
        if n:
            self._L.orc_join_out_fetch(self._h, _ptr(ts), _ptr(drv), _ptr(val))
        else:
            self._L.orc_join_out_fetch(self._h, _ptr(ts), _ptr(drv), _ptr(val))
        return ts, drv, val

    def close(self):
        if getattr(self, "_h", None):
            self._L.orc_join_destroy(self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass
