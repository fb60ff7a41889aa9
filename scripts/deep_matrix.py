import sys
sys.path.insert(0, "/root/repo")
import numpy as np
from tests.test_gpu_parity import run_both, assert_parity

import os
rng = np.random.default_rng(int(os.environ.get("DM_SEED", "777")))
fails = 0
for case in range(100):
    len_ms = int(rng.choice([250, 333, 500, 1000, 1500, 2500, 5000]))
    slide_ms = int(rng.choice([0, 0, 0, 100, 125, 250, 500, len_ms, len_ms * 2]))
    nkeys = int(rng.choice([1, 2, 7, 100, 513, 3000, 70_000, 200_000]))
    nb = int(rng.integers(1, 6))
    rows = int(rng.integers(50, 60_000))
    with_nulls = bool(rng.random() < 0.35)
    hop = slide_ms if slide_ms else len_ms
    t = 1_000_000
    batches, valids = [], []
    for _ in range(nb):
        ts = (t + np.cumsum(rng.integers(0, int(rng.integers(1, 40)), rows))).astype(np.int64)
        span = int(ts.max()) - t
        cap = hop * 2000
        if span > cap:
            ts = (t + (ts - t) * cap // span).astype(np.int64)
        t = int(ts.max())
        k = rng.integers(0, nkeys, rows)
        v = rng.uniform(-50, 200, rows)
        batches.append((ts, k, v))
        valids.append((rng.random(rows) > 0.15) if with_nulls else None)
    print(f"case {case}: len={len_ms} slide={slide_ms} keys={nkeys} "
          f"rows={rows}x{nb} nulls={with_nulls}", flush=True)
    try:
        outs, exp = run_both(len_ms, slide_ms, batches,
                             n_keys_hint=min(nkeys, 128), valids=valids)
        assert_parity(outs, exp)
    except Exception as e:
        fails += 1
        print(f"FAIL case {case}: len={len_ms} slide={slide_ms} keys={nkeys} "
              f"rows={rows}x{nb} nulls={with_nulls}: {repr(e)[:200]}")
print(f"deep matrix: {100 - fails}/100 passed")
