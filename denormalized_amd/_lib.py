"""ctypes binding over the C ABI (include/denormalized_amd.h).

The HIP engine is REQUIRED for compute: load failures and missing-GPU errors
raise immediately — there is no CPU fallback in the product path.
"""
import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
# DZ_ENGINE_SO: A/B hook — point at an alternatively-built engine .so
_SO = os.environ.get("DZ_ENGINE_SO") or os.path.join(_DIR, "_dzengine.so")

DZ_OK = 0

WINDOW_TUMBLING = 0
WINDOW_SLIDING = 1
AGG_COUNT, AGG_MIN, AGG_MAX, AGG_SUM, AGG_AVG = 0, 1, 2, 3, 4
KEY_UTF8, KEY_INT64, KEY_DENSE_INT64 = 0, 1, 2
CMP = {"<": 0, "<=": 1, ">": 2, ">=": 3, "==": 4, "!=": 5}
AGG_BY_NAME = {"count": AGG_COUNT, "min": AGG_MIN, "max": AGG_MAX,
               "sum": AGG_SUM, "avg": AGG_AVG, "average": AGG_AVG}


class DzAggDesc(ctypes.Structure):
    _fields_ = [("op", ctypes.c_int), ("input_col", ctypes.c_int32)]


class DzWindowDesc(ctypes.Structure):
    _fields_ = [
        ("window_type", ctypes.c_int),
        ("length_ms", ctypes.c_int64),
        ("slide_ms", ctypes.c_int64),
        ("ts_col", ctypes.c_int32),
        ("group_col", ctypes.c_int32),
        ("key_kind", ctypes.c_int),
        ("aggs", ctypes.POINTER(DzAggDesc)),
        ("n_aggs", ctypes.c_int32),
        ("n_keys_hint", ctypes.c_int64),
        ("device", ctypes.c_int32),
        ("max_open_windows", ctypes.c_int32),
    ]


class DzColumn(ctypes.Structure):
    _fields_ = [
        ("len", ctypes.c_int64),
        ("validity", ctypes.c_void_p),
        ("offsets", ctypes.c_void_p),
        ("data", ctypes.c_void_p),
    ]


class DzBatch(ctypes.Structure):
    _fields_ = [("n_rows", ctypes.c_int64), ("n_cols", ctypes.c_int32),
                ("cols", ctypes.POINTER(DzColumn))]


class DzOutBatch(ctypes.Structure):
    _fields_ = [
        ("n_rows", ctypes.c_int64),
        ("key_i64", ctypes.POINTER(ctypes.c_int64)),
        ("key_offsets", ctypes.POINTER(ctypes.c_int32)),
        # c_void_p, NOT c_char_p: key bytes are arbitrary (may contain NUL);
        # a c_char_p attribute read returns a NUL-truncated copy, and
        # string_at on that copy could read out of bounds.
        ("key_data", ctypes.c_void_p),
        ("agg_cols", ctypes.POINTER(ctypes.c_void_p)),
        ("agg_valid", ctypes.POINTER(ctypes.c_uint8)),
        ("window_start_ms", ctypes.POINTER(ctypes.c_int64)),
        ("window_end_ms", ctypes.POINTER(ctypes.c_int64)),
    ]


class DzKernelStat(ctypes.Structure):
    _fields_ = [("name", ctypes.c_char * 32), ("launches", ctypes.c_uint64),
                ("total_ms", ctypes.c_double), ("bytes_per_launch_alg", ctypes.c_double)]


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_SO):
            raise RuntimeError(
                f"denormalized_amd HIP engine missing: {_SO}. "
                "Run __graft_entry__.build() (hipcc --offload-arch=gfx950).")
        L = ctypes.CDLL(_SO)
        p = ctypes.c_void_p
        i64 = ctypes.c_int64
        L.dz_window_op_create.restype = p
        L.dz_window_op_create.argtypes = [ctypes.POINTER(DzWindowDesc)]
        L.dz_window_op_push.argtypes = [p, ctypes.POINTER(DzBatch)]
        L.dz_window_op_push_device.argtypes = [p, i64, p, p, p]
        try:
            L.dz_window_op_push_device_borrowed.argtypes = [p, i64, p, p, p]
        except AttributeError:  # older engine build (A/B hook)
            pass
        L.dz_window_op_push_device_utf8.argtypes = [p, i64, p, p, p, p]
        L.dz_generate_utf8.argtypes = [ctypes.c_int32, ctypes.c_uint64, i64,
                                       i64, i64, p, p, p]
        L.dz_window_op_poll.argtypes = [p, ctypes.POINTER(ctypes.POINTER(DzOutBatch))]
        L.dz_window_op_finish.argtypes = [p]
        L.dz_window_op_drain.argtypes = [p]
        L.dz_window_op_destroy.argtypes = [p]
        L.dz_last_error.restype = ctypes.c_char_p
        L.dz_last_error.argtypes = [p]
        L.dz_window_op_advance_watermark.argtypes = [p, i64]
        L.dz_window_op_watermark.restype = i64
        L.dz_window_op_watermark.argtypes = [p]
        L.dz_window_op_open_windows.restype = i64
        L.dz_window_op_open_windows.argtypes = [p]
        L.dz_window_op_set_filter.argtypes = [p, ctypes.c_int32, ctypes.c_int32,
                                              ctypes.c_double]
        L.dz_window_op_kernel_stats.argtypes = [p, ctypes.POINTER(DzKernelStat),
                                                ctypes.c_int32,
                                                ctypes.POINTER(ctypes.c_int32)]
        L.dz_generate.argtypes = [ctypes.c_int32, ctypes.c_uint64, i64, i64, i64,
                                  i64, i64, p, p, p, p]
        L.dz_device_malloc.argtypes = [ctypes.c_int32, ctypes.c_size_t,
                                       ctypes.POINTER(p)]
        L.dz_device_free.argtypes = [p]
        L.dz_device_synchronize.argtypes = [ctypes.c_int32]
        L.dz_memcpy_d2h.argtypes = [p, p, ctypes.c_size_t]
        L.dz_memcpy_h2d.argtypes = [p, p, ctypes.c_size_t]
        L.dz_join_op_create.restype = p
        L.dz_join_op_create.argtypes = [ctypes.c_int32, i64]
        L.dz_join_op_destroy.argtypes = [p]
        L.dz_join_last_error.restype = ctypes.c_char_p
        L.dz_join_last_error.argtypes = [p]
        L.dz_join_op_push_build.argtypes = [p, i64, p, p]
        L.dz_join_op_push_probe.argtypes = [p, i64, p, p, p]
        L.dz_join_op_matches.argtypes = [p, ctypes.POINTER(i64),
                                         ctypes.POINTER(p), ctypes.POINTER(p),
                                         ctypes.POINTER(p)]
        L.dz_join_op_unmatched.restype = i64
        L.dz_join_op_unmatched.argtypes = [p]
        L.dz_json_decoder_create.restype = p
        L.dz_json_decoder_create.argtypes = [ctypes.c_int32, ctypes.c_char_p,
                                             ctypes.c_char_p, ctypes.c_char_p]
        L.dz_json_decoder_destroy.argtypes = [p]
        L.dz_json_decoder_last_error.restype = ctypes.c_char_p
        L.dz_json_decoder_last_error.argtypes = [p]
        L.dz_json_decode.argtypes = [p, p, i64]
        L.dz_json_decoder_batch.argtypes = [p, ctypes.POINTER(i64),
                                            ctypes.POINTER(p),
                                            ctypes.POINTER(p),
                                            ctypes.POINTER(p),
                                            ctypes.POINTER(p)]
        L.dz_generate_json.argtypes = [ctypes.c_int32, ctypes.c_uint64, i64,
                                       i64, i64, i64, i64, p, p, p]
        L.dz_debug_windows_for_range.restype = i64
        L.dz_debug_windows_for_range.argtypes = [i64, i64, i64, i64, p, p, i64]
        L.dz_version.restype = ctypes.c_char_p
        _lib = L
    return _lib


def _err(L, h=None):
    m = L.dz_last_error(h)
    return m.decode() if m else "unknown error"


def _np_ptr(a):
    return a.ctypes.data_as(ctypes.c_void_p) if a is not None else None


class WindowOp:
    """Python handle over dz_window_op (one partition's stream)."""

    def __init__(self, length_ms, slide_ms=0, aggs=(("count", 0), ("min", 0),
                 ("max", 0), ("avg", 0)), key_kind=KEY_UTF8, n_keys_hint=1024,
                 device=0, max_open_windows=0, ts_col=0, group_col=1,
                 value_col=2, no_group=False):
        self._L = lib()
        self.no_group = no_group
        if no_group:
            group_col = -1
            value_col = 1
        n = len(aggs)
        self._agg_arr = (DzAggDesc * n)()
        self.agg_names = []
        for i, (name, _col) in enumerate(aggs):
            self._agg_arr[i].op = AGG_BY_NAME[name] if isinstance(name, str) else name
            self._agg_arr[i].input_col = value_col
            self.agg_names.append(name if isinstance(name, str) else str(name))
        d = DzWindowDesc(
            window_type=WINDOW_SLIDING if slide_ms else WINDOW_TUMBLING,
            length_ms=length_ms, slide_ms=slide_ms or 0, ts_col=ts_col,
            group_col=group_col, key_kind=key_kind,
            aggs=ctypes.cast(self._agg_arr, ctypes.POINTER(DzAggDesc)),
            n_aggs=n, n_keys_hint=n_keys_hint, device=device,
            max_open_windows=max_open_windows)
        self.key_kind = key_kind
        self._h = self._L.dz_window_op_create(ctypes.byref(d))
        if not self._h:
            raise RuntimeError(f"dz_window_op_create failed: {_err(self._L)}")

    def _check(self, st, what):
        if st != DZ_OK:
            raise RuntimeError(f"{what} failed: {_err(self._L, self._h)}")

    def push(self, ts_ms, keys, vals, val_valid_bitmap=None):
        """Host-batch push (the drop-in boundary). keys: np.int64 array, or a
        list of str/bytes for utf8. val_valid_bitmap: LSB-first bitmap bytes."""
        ts_ms = np.ascontiguousarray(ts_ms, np.int64)
        vals = np.ascontiguousarray(vals, np.float64)
        n = len(ts_ms)
        keep = []
        if self.no_group:
            cols = (DzColumn * 2)()
            cols[0] = DzColumn(n, None, None, _np_ptr(ts_ms).value if n else None)
            bm = None
            if val_valid_bitmap is not None:
                bm = np.ascontiguousarray(val_valid_bitmap, np.uint8)
                keep.append(bm)
            cols[1] = DzColumn(n, _np_ptr(bm).value if bm is not None else None,
                               None, _np_ptr(vals).value if n else None)
            batch = DzBatch(n, 2, ctypes.cast(cols, ctypes.POINTER(DzColumn)))
            self._check(self._L.dz_window_op_push(self._h, ctypes.byref(batch)),
                        "push")
            return
        cols = (DzColumn * 3)()
        cols[0] = DzColumn(n, None, None, _np_ptr(ts_ms).value if n else None)
        if self.key_kind == KEY_UTF8:
            data = b"".join(k.encode() if isinstance(k, str) else bytes(k) for k in keys)
            offs = np.zeros(n + 1, np.int32)
            pos = 0
            for i, k in enumerate(keys):
                pos += len(k.encode() if isinstance(k, str) else bytes(k))
                offs[i + 1] = pos
            buf = np.frombuffer(data, np.uint8) if data else np.zeros(1, np.uint8)
            keep += [offs, buf]
            cols[1] = DzColumn(n, None, _np_ptr(offs).value, _np_ptr(buf).value)
        else:
            karr = np.ascontiguousarray(keys, np.int64)
            keep.append(karr)
            cols[1] = DzColumn(n, None, None, _np_ptr(karr).value if n else None)
        bm = None
        if val_valid_bitmap is not None:
            bm = np.ascontiguousarray(val_valid_bitmap, np.uint8)
            keep.append(bm)
        cols[2] = DzColumn(n, _np_ptr(bm).value if bm is not None else None, None,
                           _np_ptr(vals).value if n else None)
        batch = DzBatch(n, 3, ctypes.cast(cols, ctypes.POINTER(DzColumn)))
        self._check(self._L.dz_window_op_push(self._h, ctypes.byref(batch)), "push")

    def push_device(self, n, d_ts, d_kid32, d_vals, borrowed=False):
        """borrowed=True: zero-copy — the op reads these buffers until the
        NEXT call on the op; they must stay valid and unmodified until then."""
        fn = (self._L.dz_window_op_push_device_borrowed if borrowed
              else self._L.dz_window_op_push_device)
        self._check(fn(self._h, n, d_ts, d_kid32, d_vals), "push_device")

    def push_device_utf8(self, n, d_ts, d_key_offsets, d_key_data, d_vals):
        """Raw utf8 keys interned ON DEVICE (GroupValues::intern analog).
        Borrowed semantics: all four buffers stay valid until the next call
        on the op. Requires key_kind KEY_UTF8."""
        self._check(self._L.dz_window_op_push_device_utf8(
            self._h, n, d_ts, d_key_offsets, d_key_data, d_vals),
            "push_device_utf8")

    def poll(self, copy=True):
        """Returns a dict of numpy arrays for one emitted batch, or None.
        copy=False returns zero-copy views of the op-owned buffers (valid
        until the next poll on this handle — the C contract's ownership
        rule); copy=True materialises."""
        outp = ctypes.POINTER(DzOutBatch)()
        self._check(self._L.dz_window_op_poll(self._h, ctypes.byref(outp)), "poll")
        if not outp:
            return None

        def mk(p, n, ctype):
            if not n:
                return np.zeros(0, np.int64 if ctype == ctypes.c_int64 else np.float64)
            a = np.ctypeslib.as_array(ctypes.cast(p, ctypes.POINTER(ctype)), (n,))
            return a.copy() if copy else a
        ob = outp.contents
        n = ob.n_rows
        res = {"n_rows": n}
        if self.no_group:
            pass  # global aggregate: no group column in the output schema
        elif self.key_kind == KEY_UTF8:
            if copy:
                offs = (np.ctypeslib.as_array(ob.key_offsets, (n + 1,)).copy()
                        if n else np.zeros(1, np.int32))
                total = int(offs[-1]) if n else 0
                data = ctypes.string_at(ob.key_data, total) if total else b""
                res["key"] = [data[offs[i]:offs[i + 1]].decode()
                              for i in range(n)]
            else:
                # zero-copy: raw Arrow views (decoding every key into Python
                # strings costs more than the whole GPU step)
                offs = (np.ctypeslib.as_array(ob.key_offsets, (n + 1,))
                        if n else np.zeros(1, np.int32))
                total = int(offs[-1]) if n else 0
                res["key_offsets"] = offs
                res["key_data"] = (np.ctypeslib.as_array(
                    ctypes.cast(ob.key_data, ctypes.POINTER(ctypes.c_uint8)),
                    (total,)) if total else np.zeros(0, np.uint8))
        else:
            res["key"] = mk(ob.key_i64, n, ctypes.c_int64)
        for i, name in enumerate(self.agg_names):
            pt = ob.agg_cols[i]
            if AGG_BY_NAME.get(name, name) == AGG_COUNT or name == "count":
                res[name] = mk(pt, n, ctypes.c_int64)
            else:
                res[name] = mk(pt, n, ctypes.c_double)
        if n:
            v = np.ctypeslib.as_array(ob.agg_valid, (n,))
            res["valid"] = v.copy() if copy else v
        else:
            res["valid"] = np.zeros(0, np.uint8)
        res["window_start"] = mk(ob.window_start_ms, n, ctypes.c_int64)
        res["window_end"] = mk(ob.window_end_ms, n, ctypes.c_int64)
        return res

    def drain(self):
        """Wait until every triggered window's batch is pollable (emission is
        pipelined on a worker thread)."""
        self._check(self._L.dz_window_op_drain(self._h), "drain")

    def poll_all(self, drain=True, copy=True):
        """Collect every pollable batch as owned arrays. The C ownership
        contract invalidates a batch's buffers at the next poll on the
        handle, so a LIST of live views is impossible: poll_all always
        materialises (the copy parameter is kept for signature
        compatibility but no longer returns aliased views). Zero-copy
        consumption goes through poll(copy=False) or poll_iter(), reading
        each batch before advancing."""
        del copy
        if drain:
            self.drain()
        out = []
        while True:
            b = self.poll(copy=True)
            if b is None:
                return out
            out.append(b)

    def poll_iter(self, copy=False):
        """Generator form of poll_all: yields one batch at a time; with
        copy=False the yielded views are valid only until the next
        iteration (exactly the single-poll contract)."""
        while True:
            b = self.poll(copy=copy)
            if b is None:
                return
            yield b

    def finish(self):
        self._check(self._L.dz_window_op_finish(self._h), "finish")

    def set_filter(self, agg_name_or_idx, cmp, literal):
        idx = (self.agg_names.index(agg_name_or_idx)
               if isinstance(agg_name_or_idx, str) else agg_name_or_idx)
        self._check(self._L.dz_window_op_set_filter(
            self._h, idx, CMP[cmp] if isinstance(cmp, str) else cmp,
            float(literal)), "set_filter")

    def advance_watermark(self, wm_ms):
        self._check(self._L.dz_window_op_advance_watermark(self._h, wm_ms),
                    "advance_watermark")

    @property
    def watermark(self):
        return self._L.dz_window_op_watermark(self._h)

    @property
    def open_windows(self):
        return self._L.dz_window_op_open_windows(self._h)

    def kernel_stats(self):
        # 32 slots: the stat map currently holds up to 16 entries (6 device
        # kernels + 10 host timers) — headroom so a new timer can never
        # silently push the roofline's dominant kernel out of the window
        arr = (DzKernelStat * 32)()
        nout = ctypes.c_int32()
        self._check(self._L.dz_window_op_kernel_stats(self._h, arr, 32,
                                                      ctypes.byref(nout)), "stats")
        return {arr[i].name.decode(): {
            "launches": arr[i].launches, "total_ms": arr[i].total_ms,
            "bytes_per_launch_alg": arr[i].bytes_per_launch_alg,
        } for i in range(nout.value)}

    def close(self):
        if getattr(self, "_h", None):
            self._L.dz_window_op_destroy(self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class JsonDecoder:
    """Device JSON decode (from_topic's decode stage): newline-delimited
    records -> (ts, utf8 key column, value) device columns shaped for
    WindowOp.push_device_utf8. See include/denormalized_amd.h for the
    documented subset."""

    def __init__(self, device=0, ts_field="occurred_at_ms",
                 key_field="sensor_name", val_field="reading"):
        self._L = lib()
        self._h = self._L.dz_json_decoder_create(
            device, ts_field.encode(), key_field.encode(), val_field.encode())
        if not self._h:
            m = self._L.dz_json_decoder_last_error(None)
            raise RuntimeError(f"dz_json_decoder_create failed: "
                               f"{m.decode() if m else 'unknown'}")

    def _check(self, st, what):
        if st != DZ_OK:
            m = self._L.dz_json_decoder_last_error(self._h)
            raise RuntimeError(f"{what} failed: "
                               f"{m.decode() if m else 'unknown'}")

    def decode(self, d_bytes, n_bytes):
        self._check(self._L.dz_json_decode(self._h, d_bytes, n_bytes),
                    "json decode")
        return self.batch()

    def batch(self):
        """(n, d_ts, d_key_offsets, d_key_data, d_vals) of the LAST decode
        — device pointers, valid until the second-next decode."""
        n = ctypes.c_int64()
        ts = ctypes.c_void_p()
        ko = ctypes.c_void_p()
        kd = ctypes.c_void_p()
        v = ctypes.c_void_p()
        self._check(self._L.dz_json_decoder_batch(
            self._h, ctypes.byref(n), ctypes.byref(ts), ctypes.byref(ko),
            ctypes.byref(kd), ctypes.byref(v)), "batch")
        return n.value, ts, ko, kd, v

    def close(self):
        if getattr(self, "_h", None):
            self._L.dz_json_decoder_destroy(self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class JoinOp:
    """Stream inner equi-join on trip_id (BASELINE cfg5): build side
    (trip_id -> driver_id), probe side (ts, trip_id, value). Matches are
    device-resident columns shaped for WindowOp.push_device(borrowed=True)
    — see include/denormalized_amd.h for the emission discipline."""

    def __init__(self, device=0, n_trips_hint=1 << 20):
        self._L = lib()
        self._h = self._L.dz_join_op_create(device, n_trips_hint)
        if not self._h:
            m = self._L.dz_join_last_error(None)
            raise RuntimeError(f"dz_join_op_create failed: "
                               f"{m.decode() if m else 'unknown'}")

    def _check(self, st, what):
        if st != DZ_OK:
            m = self._L.dz_join_last_error(self._h)
            raise RuntimeError(f"{what} failed: "
                               f"{m.decode() if m else 'unknown'}")

    def push_build(self, n, d_trips, d_drivers):
        self._check(self._L.dz_join_op_push_build(self._h, n, d_trips,
                                                  d_drivers), "push_build")

    def push_probe(self, n, d_ts, d_trips, d_vals):
        self._check(self._L.dz_join_op_push_probe(self._h, n, d_ts, d_trips,
                                                  d_vals), "push_probe")

    def matches(self):
        """(n, d_ts, d_kid32, d_vals) of the LAST push — device pointers,
        valid until the second-next push on this op."""
        n = ctypes.c_int64()
        ts = ctypes.c_void_p()
        kid = ctypes.c_void_p()
        val = ctypes.c_void_p()
        self._check(self._L.dz_join_op_matches(
            self._h, ctypes.byref(n), ctypes.byref(ts), ctypes.byref(kid),
            ctypes.byref(val)), "matches")
        return n.value, ts, kid, val

    @property
    def unmatched(self):
        return self._L.dz_join_op_unmatched(self._h)

    def close(self):
        if getattr(self, "_h", None):
            self._L.dz_join_op_destroy(self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class DeviceArray:
    """Raw device allocation (no torch types cross the C ABI)."""

    def __init__(self, device, nbytes):
        self._L = lib()
        self.device = device
        self.nbytes = nbytes
        p = ctypes.c_void_p()
        if self._L.dz_device_malloc(device, nbytes, ctypes.byref(p)) != DZ_OK:
            raise RuntimeError("device malloc failed")
        self.ptr = p

    def to_host(self, dtype, count):
        out = np.empty(count, dtype)
        if self._L.dz_memcpy_d2h(_np_ptr(out), self.ptr, out.nbytes) != DZ_OK:
            raise RuntimeError("d2h failed")
        return out

    def from_host(self, arr):
        arr = np.ascontiguousarray(arr)
        if self._L.dz_memcpy_h2d(self.ptr, _np_ptr(arr), arr.nbytes) != DZ_OK:
            raise RuntimeError("h2d failed")

    def free(self):
        if getattr(self, "ptr", None):
            self._L.dz_device_free(self.ptr)
            self.ptr = None

    def __del__(self):
        try:
            self.free()
        except Exception:
            pass


def generate(device, seed, t0_ms, start_row, n_rows, n_keys, rows_per_ms,
             d_ts=None, d_keys=None, d_kid32=None, d_vals=None):
    L = lib()
    if L.dz_generate(device, seed, t0_ms, start_row, n_rows, n_keys, rows_per_ms,
                     d_ts, d_keys, d_kid32, d_vals) != DZ_OK:
        raise RuntimeError(f"dz_generate failed: {_err(L)}")


def generate_json(device, seed, t0_ms, start_row, n_rows, n_keys,
                  rows_per_ms, d_lens=None, d_offsets=None, d_data=None):
    L = lib()
    if L.dz_generate_json(device, seed, t0_ms, start_row, n_rows, n_keys,
                          rows_per_ms, d_lens, d_offsets, d_data) != DZ_OK:
        raise RuntimeError(f"dz_generate_json failed: {_err(L)}")


def generate_utf8(device, seed, start_row, n_rows, n_keys, d_lens=None,
                  d_offsets=None, d_key_data=None):
    """Synthetic "sensor_{k}" key column: lens pass (host cumsums into
    offsets) and/or fill pass."""
    L = lib()
    if L.dz_generate_utf8(device, seed, start_row, n_rows, n_keys, d_lens,
                          d_offsets, d_key_data) != DZ_OK:
        raise RuntimeError(f"dz_generate_utf8 failed: {_err(L)}")


def synchronize(device=0):
    L = lib()
    if L.dz_device_synchronize(device) != DZ_OK:
        raise RuntimeError("device synchronize failed")
