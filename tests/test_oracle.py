"""Oracle correctness: hand-computed SQL-standard golden vectors, the one
reference-pinned fixture (avg state finalize, utils/serialization.rs:534-557),
reference quirk semantics (snap granularity, watermark timing, late data),
and randomized cross-check against an independent pure-Python restatement.
All CPU-only."""
import json
import os

import numpy as np
import pytest

from oracle import pyoracle
from tests.pyref import PyRef, windows_for_range as py_windows

GOLDEN_DIR = os.path.join(os.path.dirname(__file__), "golden")


def run_oracle(len_ms, slide_ms, batches, finish=True):
    o = pyoracle.Oracle(len_ms, slide_ms)
    for ts, k, v, *vv in batches:
        o.push(np.array(ts, np.int64), np.array(k, np.int64),
               np.array(v, np.float64), vv[0] if vv else None)
    if finish:
        o.finish()
    out = o.fetch()
    o.close()
    return out


def rows(out):
    return [(int(out["key"][i]), int(out["count"][i]), float(out["min"][i]),
             float(out["max"][i]), float(out["avg"][i]), float(out["sum"][i]),
             int(out["valid"][i]), int(out["window_start"][i]),
             int(out["window_end"][i])) for i in range(len(out["key"]))]


# ---------------------------------------------------------------- golden

def test_tumbling_hand_computed():
    out = run_oracle(1000, 0, [
        ([5000, 5100, 5900], [1, 2, 1], [10.0, 20.0, 30.0]),
        ([6100], [3], [5.0]),
    ])
    assert rows(out) == [
        (1, 2, 10.0, 30.0, 20.0, 40.0, 1, 5000, 6000),
        (2, 1, 20.0, 20.0, 20.0, 20.0, 1, 5000, 6000),
        (3, 1, 5.0, 5.0, 5.0, 5.0, 1, 6000, 7000),
    ]


def test_sliding_hand_computed():
    # len=2000 slide=1000: row coverage per streaming_window.rs:1062-1075
    out = run_oracle(2000, 1000, [([10500, 11500], [7, 7], [1.0, 2.0])])
    assert rows(out) == [
        (7, 1, 1.0, 1.0, 1.0, 1.0, 1, 9000, 11000),
        (7, 2, 1.0, 2.0, 1.5, 3.0, 1, 10000, 12000),
        (7, 1, 2.0, 2.0, 2.0, 2.0, 1, 11000, 13000),
    ]


def test_avg_state_reference_fixture():
    # The one hot-path-adjacent golden vector in the reference's own tests:
    # utils/serialization.rs:534-557 pins avg state = [sum f64, count u64],
    # finalize sum/count; 112.0 over 2 values -> 56.0.
    out = run_oracle(1000, 0, [([0, 1], [9, 9], [100.0, 12.0])])
    assert rows(out) == [(9, 2, 12.0, 100.0, 56.0, 112.0, 1, 0, 1000)]


def test_snap_whole_second_quirk():
    # streaming_window.rs:1088-1094: len 1500ms -> len_secs=1, snap to the
    # whole second, windows advance by 1500ms from there.
    ws, we = pyoracle.windows_for_range(5200, 5200, 1500, 0)
    assert list(ws) == [5000] and list(we) == [6500]
    # 5-second window (the simple_aggregation example length)
    ws, we = pyoracle.windows_for_range(12_345, 12_345, 5000, 0)
    assert list(ws) == [10_000] and list(we) == [15_000]


def test_subsecond_ms_generalization():
    # Reference divides by zero for len<1000ms; we define the ms
    # generalization: start = ts - ts % len.
    ws, we = pyoracle.windows_for_range(1234, 1234, 500, 0)
    assert list(ws) == [1000] and list(we) == [1500]
    ws, we = pyoracle.windows_for_range(1234, 1334, 500, 100)
    # sliding 500/100: every window [s, s+500) with s multiple of 100
    # overlapping [1234,1334]: s from snap(734)=700 ... step 100, skip
    # windows with min > end or max < start
    exp = [(s, s + 500) for s in range(700, 1400, 100)
           if not (1234 > s + 500 or 1334 < s)]
    assert list(zip(ws, we)) == exp


def test_tumbling_boundary_row_goes_to_next_window():
    # ts exactly at a window boundary belongs to the window starting there
    # ([start, end) routing, grouped_window_agg_stream.rs:568-600), and the
    # range list includes a window starting at max_ts (loop is <=).
    out = run_oracle(1000, 0, [([1000, 2000], [1, 1], [1.0, 2.0])])
    assert rows(out) == [
        (1, 1, 1.0, 1.0, 1.0, 1.0, 1, 1000, 2000),
        (1, 1, 2.0, 2.0, 2.0, 2.0, 1, 2000, 3000),
    ]


# ------------------------------------------------------ watermark / trigger

def test_emission_timing_watermark_is_max_of_batch_mins():
    o = pyoracle.Oracle(1000, 0)
    o.push(np.array([1000, 1999], np.int64), np.array([1, 1], np.int64),
           np.array([1.0, 2.0], np.float64))
    assert len(o.fetch()["key"]) == 0  # wm=1000 < 2000: nothing emitted
    o.push(np.array([1999], np.int64), np.array([1], np.int64),
           np.array([3.0], np.float64))
    assert len(o.fetch()["key"]) == 0  # wm=1999 < 2000: still open
    o.push(np.array([2000], np.int64), np.array([2], np.int64),
           np.array([4.0], np.float64))
    out = o.fetch()  # wm=2000 >= 2000: [1000,2000) emits now
    assert rows(out) == [(1, 3, 1.0, 3.0, 2.0, 6.0, 1, 1000, 2000)]
    o.close()


def test_watermark_never_regresses():
    o = pyoracle.Oracle(1000, 0)
    o.push(np.array([5000], np.int64), np.array([1], np.int64),
           np.array([1.0], np.float64))
    assert o.watermark == 5000
    o.push(np.array([3000], np.int64), np.array([1], np.int64),
           np.array([1.0], np.float64))
    assert o.watermark == 5000  # process_watermark keeps the max (:255-266)
    o.close()


def test_late_data_recreates_and_reemits_window():
    # ensure_window_frames_for_ranges has no seen-windows guard (:276-313):
    # a late row re-creates the closed frame and it re-emits immediately.
    o = pyoracle.Oracle(1000, 0)
    o.push(np.array([1000], np.int64), np.array([1], np.int64),
           np.array([1.0], np.float64))
    o.push(np.array([2500], np.int64), np.array([1], np.int64),
           np.array([9.0], np.float64))
    assert rows(o.fetch()) == [(1, 1, 1.0, 1.0, 1.0, 1.0, 1, 1000, 2000)]
    o.push(np.array([1500], np.int64), np.array([2], np.int64),
           np.array([5.0], np.float64))  # late row
    assert rows(o.fetch()) == [(2, 1, 5.0, 5.0, 5.0, 5.0, 1, 1000, 2000)]
    o.close()


# ------------------------------------------------------------- edge cases

def test_null_values_and_all_null_group():
    # count counts non-null; min/max/avg/sum NULL for an all-null group but
    # the group still emits (it was interned).
    out = run_oracle(1000, 0, [
        ([0, 1, 2], [1, 1, 2], [5.0, 7.0, 99.0], np.array([1, 0, 0], np.uint8)),
    ])
    assert rows(out) == [
        (1, 1, 5.0, 5.0, 5.0, 5.0, 1, 0, 1000),
        (2, 0, 0.0, 0.0, 0.0, 0.0, 0, 0, 1000),
    ]


def test_empty_batch_is_noop():
    o = pyoracle.Oracle(1000, 0)
    o.push(np.array([], np.int64), np.array([], np.int64),
           np.array([], np.float64))
    assert o.watermark == np.iinfo(np.int64).min  # unset
    o.close()


def test_insertion_order_emission():
    out = run_oracle(1000, 0, [
        ([0, 0, 0, 0], [42, 7, 42, 1], [1.0, 2.0, 3.0, 4.0]),
    ])
    assert [r[0] for r in rows(out)] == [42, 7, 1]  # first-seen order


def test_negative_values_minmax():
    out = run_oracle(1000, 0, [([0, 0, 0], [1, 1, 1], [-5.0, -1.0, -9.0])])
    assert rows(out) == [(1, 3, -9.0, -1.0, -5.0, -15.0, 1, 0, 1000)]


# ------------------------------------------------- cross-check vs pyref

@pytest.mark.parametrize("seed", [0, 1, 2, 3])
@pytest.mark.parametrize("len_ms,slide_ms", [
    (1000, 0), (5000, 0), (1500, 0), (500, 0), (2000, 1000), (500, 100),
    (3000, 500),
])
def test_randomized_crosscheck(seed, len_ms, slide_ms):
    rng = np.random.default_rng(seed * 1000 + len_ms + slide_ms)
    o = pyoracle.Oracle(len_ms, slide_ms)
    p = PyRef(len_ms, slide_ms)
    t = 10_000
    for _ in range(rng.integers(3, 8)):
        n = int(rng.integers(1, 200))
        # mostly-monotonic timestamps with occasional late rows
        ts = t + np.cumsum(rng.integers(0, 50, n))
        late = rng.random(n) < 0.05
        ts = np.where(late, np.maximum(0, ts - int(rng.integers(0, 3000))), ts)
        t = int(ts.max())
        k = rng.integers(0, 20, n).astype(np.int64)
        v = np.round(rng.uniform(-10, 115, n), 3)
        valid = (rng.random(n) > 0.1).astype(np.uint8)
        o.push(ts.astype(np.int64), k, v, valid)
        p.push(list(ts), list(k), list(v), list(valid))
    o.finish()
    p.finish()
    got = rows(o.fetch())
    exp = [(int(a), int(b), float(c), float(d), float(e), float(f), int(g),
            int(h), int(i)) for a, b, c, d, e, f, g, h, i in p.out]
    assert got == exp
    o.close()


def test_windows_for_range_crosscheck():
    rng = np.random.default_rng(7)
    for _ in range(200):
        len_ms = int(rng.choice([500, 1000, 1500, 2000, 5000]))
        slide = int(rng.choice([0, 100, 500, 1000]))
        if slide > len_ms:
            continue
        mn = int(rng.integers(10_000, 100_000))
        mx = mn + int(rng.integers(0, 20_000))
        ws, we = pyoracle.windows_for_range(mn, mx, len_ms, slide)
        assert list(zip(ws, we)) == py_windows(mn, mx, len_ms, slide)


# --------------------------------------------------------- generator spec

def test_generator_deterministic_and_bounded():
    ts, kid, val = pyoracle.gen(42, 1_000_000, 0, 10_000, 100, 1000)
    ts2, kid2, val2 = pyoracle.gen(42, 1_000_000, 0, 10_000, 100, 1000)
    assert np.array_equal(ts, ts2) and np.array_equal(kid, kid2)
    assert np.array_equal(val, val2)
    assert ts.min() == 1_000_000 and np.all(np.diff(ts) >= 0)
    assert kid.min() >= 0 and kid.max() < 100
    assert val.min() >= 0.0 and val.max() < 115.0
    # chunked generation is position-independent
    a = pyoracle.gen(42, 1_000_000, 5000, 100, 100, 1000)
    assert np.array_equal(a[0], ts[5000:5100])
    assert np.array_equal(a[1], kid[5000:5100])
    assert np.array_equal(a[2], val[5000:5100])


# ------------------------------------------------------- committed fixtures

def test_golden_fixtures():
    path = os.path.join(GOLDEN_DIR, "oracle_cases.json")
    with open(path) as f:
        cases = json.load(f)
    assert len(cases) >= 6
    for case in cases:
        out = run_oracle(case["len_ms"], case["slide_ms"],
                         [(b["ts"], b["keys"], b["vals"]) for b in case["batches"]])
        got = rows(out)
        exp = [tuple(r) for r in case["expected"]]
        assert [tuple(g) for g in got] == exp, case["name"]


def test_bench_generator_distribution_crosscheck():
    # oracle vs independent restatement on the BENCH's generator distribution
    # (1M rows, cfg2-shaped), batched like the bench pushes
    ts, kid, val = pyoracle.gen(42, 1_000_000, 0, 1_000_000, 10_000, 1000)
    o = pyoracle.Oracle(1000, 0)
    p = PyRef(1000, 0)
    step = 250_000
    for lo in range(0, 1_000_000, step):
        sl = slice(lo, lo + step)
        o.push(ts[sl], kid[sl], val[sl])
        p.push(ts[sl].tolist(), kid[sl].tolist(), val[sl].tolist())
    o.finish()
    p.finish()
    got = o.fetch()
    assert len(got["key"]) == len(p.out) > 9000
    for i in (0, 1, len(p.out) // 2, len(p.out) - 1):
        row = p.out[i]
        assert int(got["key"][i]) == row[0]
        assert int(got["count"][i]) == row[1]
        assert float(got["avg"][i]) == row[4]
    # full-array equality
    import numpy as np
    assert np.array_equal(got["key"], np.array([r[0] for r in p.out]))
    assert np.array_equal(got["count"], np.array([r[1] for r in p.out]))
    assert np.array_equal(got["sum"], np.array([r[5] for r in p.out]))
    o.close()
