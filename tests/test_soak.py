"""Cross-operator soak: multiple operators (window ops, join, JSON decoder)
live on one device simultaneously, interleaved pushes — stresses stream pools,
emission workers and the intern/dictionary paths against each other. Plus
seeded fuzz arms for the join interleavings and the JSON decoder."""
import json

import numpy as np
import pytest

import __graft_entry__ as graft
from oracle import pyoracle
from oracle.pyoracle import JoinOracle

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def built():
    graft.build()


def _dev(a, keep):
    from denormalized_amd import DeviceArray
    a = np.ascontiguousarray(a)
    d = DeviceArray(0, max(1, a.nbytes))
    d.from_host(a)
    keep.append(d)
    return d


@pytest.mark.timeout(240)
def test_concurrent_operators_soak():
    from denormalized_amd import JoinOp, JsonDecoder, WindowOp, _lib
    rng = np.random.default_rng(11)
    keep = []

    # pipeline A: utf8 window (device intern)
    opA = WindowOp(length_ms=1000, key_kind=_lib.KEY_UTF8, n_keys_hint=300)
    oraA = pyoracle.Oracle(1000, 0)
    # pipeline B: sliding dense window
    opB = WindowOp(length_ms=500, slide_ms=250, key_kind=_lib.KEY_DENSE_INT64,
                   n_keys_hint=512)
    oraB = pyoracle.Oracle(500, 250)
    # pipeline C: join -> window
    jo = JoinOp(device=0, n_trips_hint=1000)
    joraC = JoinOracle()
    opC = WindowOp(length_ms=1000, key_kind=_lib.KEY_DENSE_INT64,
                   n_keys_hint=64)
    oraC = pyoracle.Oracle(1000, 0)
    trips = rng.permutation(1000)
    drivers = rng.integers(0, 64, 1000)
    jo.push_build(1000, _dev(trips.astype(np.int64), keep).ptr,
                  _dev(drivers.astype(np.int64), keep).ptr)
    joraC.push_build(trips, drivers)
    # pipeline D: JSON decoder -> utf8 window
    dec = JsonDecoder(device=0)
    opD = WindowOp(length_ms=1000, key_kind=_lib.KEY_UTF8, n_keys_hint=80)
    oraD = pyoracle.Oracle(1000, 0)

    outsA, outsB, outsC, outsD = [], [], [], []
    n = 25_000
    for step in range(4):
        ts = (1_000_000 + np.arange(step * n, (step + 1) * n) // 25).astype(np.int64)
        # A: utf8 device intern
        kidA = rng.integers(0, 300, n)
        vA = rng.uniform(0, 115, n)
        names = [f"sensor_{k}".encode() for k in kidA]
        offs = np.zeros(n + 1, np.int32)
        np.cumsum([len(x) for x in names], out=offs[1:])
        data = np.frombuffer(b"".join(names), np.uint8)
        opA.push_device_utf8(n, _dev(ts, keep).ptr, _dev(offs, keep).ptr,
                             _dev(data, keep).ptr, _dev(vA, keep).ptr)
        oraA.push(ts, kidA, vA)
        # B: host push dense sliding
        kidB = rng.integers(0, 512, n)
        vB = rng.uniform(-10, 120, n)
        opB.push(ts, kidB, vB)
        oraB.push(ts, kidB, vB)
        # C: join probe -> window
        trC = rng.integers(0, 1200, n)  # some unmatched
        vC = rng.uniform(0, 115, n)
        jo.push_probe(n, _dev(ts, keep).ptr,
                      _dev(trC.astype(np.int64), keep).ptr,
                      _dev(vC, keep).ptr)
        joraC.push_probe(ts, trC, vC)
        nm, pts, pkid, pval = jo.matches()
        ets, edrv, ev = joraC.fetch()
        assert nm == len(ets)
        if nm:
            opC.push_device(nm, pts, pkid, pval)
            oraC.push(ets, edrv, ev)
        # D: JSON decode -> utf8 window
        kidD = rng.integers(0, 80, n // 5)
        vD = np.round(rng.uniform(0, 115, n // 5), 6)
        js = "\n".join(json.dumps({"occurred_at_ms": int(ts[i * 5]),
                                   "sensor_name": f"sensor_{kidD[i]}",
                                   "reading": float(vD[i])},
                                  separators=(",", ":"))
                       for i in range(n // 5)) + "\n"
        jb = js.encode()
        nr, dts, dko, dkd, dv = dec.decode(_dev(np.frombuffer(jb, np.uint8),
                                                keep).ptr, len(jb))
        assert nr == n // 5
        opD.push_device_utf8(nr, dts, dko, dkd, dv)
        oraD.push(ts[::5][:n // 5], kidD, vD)
        # drain opportunistically
        outsA += opA.poll_all()
        outsB += opB.poll_all()
        outsC += opC.poll_all()
        outsD += opD.poll_all()

    for op, ora, outs, utf8keys in (
            (opA, oraA, outsA, [f"sensor_{i}" for i in range(300)]),
            (opB, oraB, outsB, None),
            (opC, oraC, outsC, None),
            (opD, oraD, outsD, [f"sensor_{i}" for i in range(80)])):
        op.finish()
        outs += op.poll_all()
        ora.finish()
        exp = ora.fetch()
        from tests.test_gpu_parity import assert_parity
        assert len(exp["key"]) > 0
        assert_parity(outs, exp, utf8_keys=utf8keys)
        op.close()
        ora.close()
    jo.close()
    joraC.close()
    dec.close()
    for d in keep:
        d.free()


@pytest.mark.timeout(240)
@pytest.mark.parametrize("seed", [21, 22])
def test_join_fuzz_interleavings(seed):
    from denormalized_amd import JoinOp
    from tests.test_join import random_interleaving
    keep = []
    jo = JoinOp(device=0, n_trips_hint=512)
    oj = JoinOracle()
    for op in random_interleaving(seed, nops=20):
        if op[0] == "build":
            jo.push_build(len(op[1]), _dev(op[1].astype(np.int64), keep).ptr,
                          _dev(op[2].astype(np.int64), keep).ptr)
            oj.push_build(op[1], op[2])
        else:
            jo.push_probe(len(op[1]), _dev(op[1].astype(np.int64), keep).ptr,
                          _dev(op[2].astype(np.int64), keep).ptr,
                          _dev(op[3], keep).ptr)
            oj.push_probe(op[1], op[2], op[3])
        nm, pts, pkid, pval = jo.matches()
        ets, edrv, ev = oj.fetch()
        assert nm == len(ets)
        assert jo.unmatched == oj.unmatched
        if nm:
            import ctypes
            from denormalized_amd import _lib
            L = _lib.lib()
            h = np.empty(nm, np.int32)
            L.dz_memcpy_d2h(h.ctypes.data_as(ctypes.c_void_p), pkid, nm * 4)
            assert np.array_equal(h.astype(np.int64), edrv)
    jo.close()
    oj.close()
    for d in keep:
        d.free()


@pytest.mark.timeout(240)
def test_json_decoder_fuzz():
    from tests.test_json_ingest import decode_all
    rng = np.random.default_rng(33)
    rows = []
    for i in range(4000):
        rec = {"occurred_at_ms": int(rng.integers(0, 2**40)),
               "sensor_name": "k" * int(rng.integers(1, 30)) + str(i % 97),
               "reading": round(float(rng.uniform(-1e6, 1e6)), 6)}
        pool = [True, False, None, 12, -3.5, "strv",
                {"nested": [1, 2, {"y": "z"}]}, [0, {"a": "b"}, []]]
        for e in range(int(rng.integers(0, 3))):
            rec[f"x{e}"] = pool[int(rng.integers(0, len(pool)))]
        items = list(rec.items())
        rng.shuffle(items)
        rows.append(dict(items))
    js = "\n".join(json.dumps(r, separators=(",", ":")) for r in rows)
    ts, keys, v = decode_all(js.encode())
    assert len(ts) == len(rows)
    for i, r in enumerate(rows):
        assert ts[i] == r["occurred_at_ms"]
        assert keys[i] == r["sensor_name"]
        assert np.float64(v[i]).view(np.int64) == \
            np.float64(r["reading"]).view(np.int64)
