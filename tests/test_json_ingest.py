"""JSON ingest: the device decoder vs Python's json module on
identical bytes, the device generator vs a Python reimplementation of its
format, and the full on-wire pipeline (bytes -> decode -> device intern ->
window) vs the oracle fed by host-parsed rows. GPU-only (the decoder is
device code; the reference's serde_json behavior is restated through
Python's parser, which is correctly-rounding like strtod)."""
import json

import numpy as np
import pytest

import __graft_entry__ as graft
from oracle import pyoracle

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def built():
    graft.build()


def to_dev(b):
    from denormalized_amd import DeviceArray
    a = np.frombuffer(b, np.uint8) if isinstance(b, bytes) else b
    d = DeviceArray(0, max(1, a.nbytes))
    d.from_host(a)
    return d


def decode_all(js_bytes, **kw):
    from denormalized_amd import JsonDecoder, _lib
    import ctypes
    dec = JsonDecoder(device=0, **kw)
    d = to_dev(js_bytes)
    n, pts, pko, pkd, pv = dec.decode(d.ptr, len(js_bytes))
    L = _lib.lib()
    ts = np.empty(n, np.int64)
    ko = np.empty(n + 1, np.int32)
    v = np.empty(n, np.float64)
    L.dz_memcpy_d2h(ts.ctypes.data_as(ctypes.c_void_p), pts, n * 8)
    L.dz_memcpy_d2h(ko.ctypes.data_as(ctypes.c_void_p), pko, (n + 1) * 4)
    kd = np.empty(max(1, int(ko[-1])), np.uint8)
    L.dz_memcpy_d2h(kd.ctypes.data_as(ctypes.c_void_p), pkd, int(ko[-1]))
    L.dz_memcpy_d2h(v.ctypes.data_as(ctypes.c_void_p), pv, n * 8)
    keys = [kd.tobytes()[ko[i]:ko[i + 1]].decode() for i in range(n)]
    dec.close()
    d.free()
    return ts, keys, v


def test_decode_matches_python_json():
    rng = np.random.default_rng(31)
    rows = []
    for i in range(5000):
        rec = {"occurred_at_ms": int(1_000_000 + i // 3),
               "sensor_name": f"sensor_{int(rng.integers(0, 50))}",
               "reading": round(float(rng.uniform(-20, 115)), 6)}
        # field order varies; unrelated fields (incl. nested) are skipped
        if i % 3 == 0:
            rec = {"meta": {"nonsense": "MMMM", "deep": [1, {"x": "y"}]},
                   **rec, "extra": True, "nil": None}
        if i % 5 == 0:
            rec = dict(reversed(list(rec.items())))
        rows.append(rec)
    js = "\n".join(json.dumps(r, separators=(",", ":")) for r in rows)
    if len(rows) % 2 == 0:
        js += "\n"  # trailing newline form
    ts, keys, v = decode_all(js.encode())
    assert len(ts) == len(rows)
    for i, r in enumerate(rows):
        assert ts[i] == r["occurred_at_ms"]
        assert keys[i] == r["sensor_name"]
        # bitwise: the exact fast path equals Python's correctly-rounded parse
        assert np.float64(v[i]).view(np.int64) == \
            np.float64(r["reading"]).view(np.int64)


def test_decode_whitespace_and_exponents():
    js = ('{ "occurred_at_ms" : 5 , "sensor_name" : "a" , "reading" : 1.5e2 }\n'
          '{"reading":-0.25,"sensor_name":"bb","occurred_at_ms":6}')
    ts, keys, v = decode_all(js.encode())
    assert list(ts) == [5, 6]
    assert keys == ["a", "bb"]
    assert list(v) == [150.0, -0.25]


def test_decode_rejects_out_of_subset():
    for bad in (
        '{"occurred_at_ms":1,"sensor_name":"a","reading":0.12345678901234567}',
        '{"occurred_at_ms":1,"sensor_name":"a\\n","reading":1.0}',
        '{"occurred_at_ms":1,"sensor_name":"a"}',
        'not json at all',
    ):
        with pytest.raises(RuntimeError):
            decode_all(bad.encode())


def test_device_json_generator_matches_python():
    from denormalized_amd import DeviceArray, generate_json, synchronize
    n, nkeys, seed = 3000, 41, 55
    d_lens = DeviceArray(0, n * 4)
    generate_json(0, seed, 1_000_000, 70, n, nkeys, 10, d_lens=d_lens.ptr)
    synchronize(0)
    lens = d_lens.to_host(np.int32, n)
    offs = np.zeros(n + 1, np.int64)
    np.cumsum(lens, out=offs[1:])
    d_offs = to_dev(offs)
    d_data = DeviceArray(0, int(offs[-1]))
    generate_json(0, seed, 1_000_000, 70, n, nkeys, 10, d_offsets=d_offs.ptr,
                  d_data=d_data.ptr)
    synchronize(0)
    data = d_data.to_host(np.uint8, int(offs[-1])).tobytes()
    ts, kid, val = pyoracle.gen(seed, 1_000_000, 70, n, nkeys, 10)
    for i in range(n):
        line = data[offs[i]:offs[i + 1]]
        assert line.endswith(b"}\n")
        r = json.loads(line)
        assert r["occurred_at_ms"] == ts[i]
        assert r["sensor_name"] == f"sensor_{kid[i]}"
        assert abs(r["reading"] - val[i]) < 5e-7  # printed fixed-6
    for a in (d_lens, d_offs, d_data):
        a.free()


def test_onwire_pipeline_matches_oracle():
    """bytes -> device decode -> device intern -> window vs the oracle fed
    the host-parsed rows: the from_topic analog end to end, bit-exact."""
    from denormalized_amd import JsonDecoder, WindowOp, _lib
    rng = np.random.default_rng(77)
    nkeys = 120
    dec = JsonDecoder(device=0)
    op = WindowOp(length_ms=1000, key_kind=_lib.KEY_UTF8, n_keys_hint=nkeys)
    o = pyoracle.Oracle(1000, 0)
    outs, keep = [], []
    for b in range(3):
        n = 30_000
        ts = 1_000_000 + np.arange(b * n, (b + 1) * n) // 30
        kid = rng.integers(0, nkeys, n)
        val = np.round(rng.uniform(0, 115, n), 6)
        js = "\n".join(
            json.dumps({"occurred_at_ms": int(ts[i]),
                        "sensor_name": f"sensor_{kid[i]}",
                        "reading": float(val[i])}, separators=(",", ":"))
            for i in range(n)) + "\n"
        d = to_dev(js.encode())
        keep.append(d)
        nr, pts, pko, pkd, pv = dec.decode(d.ptr, len(js))
        assert nr == n
        op.push_device_utf8(nr, pts, pko, pkd, pv)
        outs += op.poll_all()
        # oracle side: host-parsed rows (identical doubles: both parses are
        # correctly rounded within the subset)
        parsed = [json.loads(l) for l in js.splitlines()]
        o.push(np.array([r["occurred_at_ms"] for r in parsed], np.int64),
               kid,  # sensor_{kid} maps back to kid for the oracle
               np.array([r["reading"] for r in parsed]))
    op.finish()
    o.finish()
    outs += op.poll_all()
    exp = o.fetch()
    from tests.test_gpu_parity import assert_parity
    assert len(exp["key"]) > 0
    assert_parity(outs, exp, utf8_keys=[f"sensor_{i}" for i in range(nkeys)])
    dec.close()
    op.close()
    o.close()
    for d in keep:
        d.free()
