"""Generates tests/golden/oracle_cases.json from the independent pure-Python
restatement (tests/pyref.py). Committed alongside the fixtures so the vectors
are reproducible. Run: python tests/golden/gen.py
"""
import json
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))
from tests.pyref import PyRef  # noqa: E402


def make_case(name, len_ms, slide_ms, batches):
    p = PyRef(len_ms, slide_ms)
    for ts, k, v in batches:
        p.push(ts, k, v)
    p.finish()
    return {
        "name": name, "len_ms": len_ms, "slide_ms": slide_ms,
        "batches": [{"ts": ts, "keys": k, "vals": v} for ts, k, v in batches],
        "expected": [list(r) for r in p.out],
    }


def main():
    rng = np.random.default_rng(20260915)
    cases = []
    # deterministic random small cases across window shapes
    shapes = [(1000, 0), (5000, 0), (1500, 0), (500, 0), (2000, 1000),
              (500, 100), (4000, 2000), (3000, 1000)]
    for idx, (len_ms, slide_ms) in enumerate(shapes):
        batches = []
        t = 50_000
        for _ in range(4):
            n = int(rng.integers(5, 60))
            ts = (t + np.cumsum(rng.integers(0, 40, n))).astype(int)
            t = int(ts.max())
            k = rng.integers(0, 8, n)
            v = np.round(rng.uniform(0, 115, n), 2)
            batches.append((ts.tolist(), k.tolist(), v.tolist()))
        cases.append(make_case(f"rand_{idx}_{len_ms}_{slide_ms}",
                               len_ms, slide_ms, batches))
    out = os.path.join(os.path.dirname(os.path.abspath(__file__)), "oracle_cases.json")
    with open(out, "w") as f:
        json.dump(cases, f)
    print(f"wrote {out}: {len(cases)} cases")


if __name__ == "__main__":
    main()
