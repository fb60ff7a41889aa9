"""Stream join (BASELINE cfg5): CPU tests pin the oracle's join restatement
against an independent Python-dict restatement; GPU tests check the device
join (and the full join->window pipeline) against the oracle bit-exactly.
Emission discipline per include/denormalized_amd.h §Stream join."""
import numpy as np
import pytest

from oracle.pyoracle import JoinOracle, Oracle, gen


class PyJoin:
    """Independent restatement: dict build side, ordered probe/buffer."""

    def __init__(self):
        self.tab = {}
        self.un = []  # (ts, trip, val) in arrival order
        self.out = []

    def push_build(self, trips, drivers):
        for t, d in zip(trips, drivers):
            self.tab[int(t)] = int(d)
        still = []
        for (ts, tr, v) in self.un:
            if tr in self.tab:
                self.out.append((ts, self.tab[tr], v))
            else:
                still.append((ts, tr, v))
        self.un = still

    def push_probe(self, ts, trips, vals):
        for a, t, v in zip(ts, trips, vals):
            t = int(t)
            if t in self.tab:
                self.out.append((int(a), self.tab[t], float(v)))
            else:
                self.un.append((int(a), t, float(v)))


def random_interleaving(seed, nops=14):
    rng = np.random.default_rng(seed)
    trips = rng.permutation(400)
    drivers = rng.integers(0, 37, 400)
    built = 0
    ops = []
    for _ in range(nops):
        if rng.random() < 0.45 and built < 400:
            k = int(rng.integers(1, 120))
            k = min(k, 400 - built)
            ops.append(("build", trips[built:built + k],
                        drivers[built:built + k]))
            built += k
        else:
            n = int(rng.integers(1, 3000))
            ts = rng.integers(1_000_000, 1_050_000, n)
            tr = rng.integers(0, 500, n)  # some trips never exist
            v = rng.uniform(0, 115, n)
            ops.append(("probe", ts, tr, v))
    if built < 400:
        ops.append(("build", trips[built:], drivers[built:]))
    return ops


@pytest.mark.parametrize("seed", [1, 2, 3])
def test_join_oracle_matches_pyjoin(seed):
    j = JoinOracle()
    p = PyJoin()
    got = []
    for op in random_interleaving(seed):
        if op[0] == "build":
            j.push_build(op[1], op[2])
            p.push_build(op[1], op[2])
        else:
            j.push_probe(op[1], op[2], op[3])
            p.push_probe(op[1], op[2], op[3])
        ts, drv, val = j.fetch()
        got += list(zip(ts.tolist(), drv.tolist(), val.tolist()))
    assert j.unmatched == len(p.un)
    assert got == p.out
    j.close()


def test_join_oracle_hand_vectors():
    j = JoinOracle()
    j.push_probe([5, 6], [100, 200], [1.5, 2.5])
    assert j.unmatched == 2
    j.push_build([200], [9])
    ts, drv, val = j.fetch()
    assert ts.tolist() == [6] and drv.tolist() == [9] and val.tolist() == [2.5]
    j.push_build([200], [10])  # duplicate overwrites (dimension update)
    j.push_probe([7], [200], [3.5])
    _, drv, _ = j.fetch()
    assert drv.tolist() == [10]
    assert j.unmatched == 1  # trip 100 never arrived
    j.close()


# ------------------------------------------------------------------ GPU

@pytest.mark.gpu
def test_device_join_matches_oracle():
    import __graft_entry__ as graft
    graft.build()
    from denormalized_amd import DeviceArray, JoinOp

    jo = JoinOp(device=0, n_trips_hint=512)
    oj = JoinOracle()
    keep = []

    def dev(a, dt):
        a = np.ascontiguousarray(a, dt)
        d = DeviceArray(0, a.nbytes)
        d.from_host(a)
        keep.append(d)
        return d

    for op in random_interleaving(7):
        if op[0] == "build":
            dtr, ddr = dev(op[1], np.int64), dev(op[2], np.int64)
            jo.push_build(len(op[1]), dtr.ptr, ddr.ptr)
            oj.push_build(op[1], op[2])
        else:
            dts, dtr, dv = (dev(op[1], np.int64), dev(op[2], np.int64),
                            dev(op[3], np.float64))
            jo.push_probe(len(op[1]), dts.ptr, dtr.ptr, dv.ptr)
            oj.push_probe(op[1], op[2], op[3])
        n, pts, pkid, pval = jo.matches()
        ets, edrv, eval_ = oj.fetch()
        assert n == len(ets)
        if n:
            gts = DeviceArray.__new__(DeviceArray)  # raw d2h reads
            from denormalized_amd import _lib
            L = _lib.lib()
            hts = np.empty(n, np.int64)
            hkid = np.empty(n, np.int32)
            hval = np.empty(n, np.float64)
            L.dz_memcpy_d2h(hts.ctypes.data_as(__import__("ctypes").c_void_p),
                            pts, n * 8)
            L.dz_memcpy_d2h(hkid.ctypes.data_as(__import__("ctypes").c_void_p),
                            pkid, n * 4)
            L.dz_memcpy_d2h(hval.ctypes.data_as(__import__("ctypes").c_void_p),
                            pval, n * 8)
            assert np.array_equal(hts, ets)
            assert np.array_equal(hkid.astype(np.int64), edrv)
            assert np.array_equal(hval.view(np.int64), eval_.view(np.int64))
        assert jo.unmatched == oj.unmatched
    jo.close()
    oj.close()
    for d in keep:
        d.free()


@pytest.mark.gpu
def test_join_window_pipeline_matches_oracle():
    """cfg5 shape end to end ON DEVICE: probe batches join on trip_id, the
    matched (ts, driver, reading) columns feed the window op ZERO-COPY
    (borrowed push), 1s tumbling group-by driver; the oracle runs the same
    pipeline (join restatement -> window restatement). Bit-exact."""
    import __graft_entry__ as graft
    graft.build()
    from denormalized_amd import DeviceArray, JoinOp, WindowOp, _lib

    rng = np.random.default_rng(99)
    n_trips, n_drivers = 2_000, 61
    trips = rng.permutation(n_trips)
    drivers = rng.integers(0, n_drivers, n_trips)

    jo = JoinOp(device=0, n_trips_hint=n_trips)
    oj = JoinOracle()
    wop = WindowOp(length_ms=1000, key_kind=_lib.KEY_DENSE_INT64,
                   n_keys_hint=n_drivers, device=0)
    wor = Oracle(1000, 0)
    keep = []

    def dev(a, dt):
        a = np.ascontiguousarray(a, dt)
        d = DeviceArray(0, max(1, a.nbytes))
        d.from_host(a)
        keep.append(d)
        return d

    # half the trips known up front; the rest arrive mid-stream (late matches
    # exercise the re-probe -> late-data window path)
    half = n_trips // 2
    jo.push_build(half, dev(trips[:half], np.int64).ptr,
                  dev(drivers[:half], np.int64).ptr)
    oj.push_build(trips[:half], drivers[:half])

    outs = []
    steps = 5
    for s in range(steps + 1):
        if s < steps:
            n = 40_000
            ts = (1_000_000 + np.arange(s * n, (s + 1) * n) // 40)
            tr = rng.integers(0, n_trips, n)
            v = rng.uniform(0, 115, n)
            jo.push_probe(n, dev(ts, np.int64).ptr, dev(tr, np.int64).ptr,
                          dev(v, np.float64).ptr)
            oj.push_probe(ts, tr, v)
        else:
            jo.push_build(n_trips - half, dev(trips[half:], np.int64).ptr,
                          dev(drivers[half:], np.int64).ptr)
            oj.push_build(trips[half:], drivers[half:])
        nm, pts, pkid, pval = jo.matches()
        ets, edrv, eval_ = oj.fetch()
        assert nm == len(ets)
        if nm:
            wop.push_device(nm, pts, pkid, pval)  # staging push: the join's
            # buffers recycle at its second-next push
            wor.push(ets, edrv, eval_)
        outs += wop.poll_all()
    wop.finish()
    wor.finish()
    outs += wop.poll_all()
    exp = wor.fetch()
    from tests.test_gpu_parity import assert_parity
    assert len(exp["key"]) > 0
    assert_parity(outs, exp)
    jo.close()
    oj.close()
    wop.close()
    wor.close()
    for d in keep:
        d.free()
