"""Multi-shard helpers for the windowed-aggregate path.

Grouped pipelines shard by key (each rank owns disjoint groups — no merge
needed). GLOBAL aggregates (no GROUP BY) follow the reference's
Partial/Final two-stage plan (planner/streaming_window.rs:133-153,
FullWindowAggStream streaming_window.rs:640-1051): each shard emits partial
{count,min,max,sum} rows per window; merge_global_partials is the Final
stage, combining shards in rank order (deterministic; the f64 sum order is
rank-major, matching the reference's partition-merge structure rather than
single-stream row order — see DESIGN.md)."""
import numpy as np


def merge_global_partials(per_rank_batches):
    """per_rank_batches: list (rank order) of lists of emitted batches from a
    no-group WindowOp (each batch: one window, one row, with 'count'/'min'/
    'max'/'sum' or 'avg' columns). Returns merged rows keyed by window."""
    acc = {}  # (wstart, wend) -> [count, min, max, sum, any_valid]
    order = []
    for batches in per_rank_batches:
        for b in batches:
            for i in range(b["n_rows"]):
                key = (int(b["window_start"][i]), int(b["window_end"][i]))
                cnt = int(b["count"][i])
                valid = bool(b["valid"][i])
                if key not in acc:
                    acc[key] = [0, None, None, 0.0]
                    order.append(key)
                a = acc[key]
                a[0] += cnt
                if valid:
                    mn, mx, sm = float(b["min"][i]), float(b["max"][i]), float(b["sum"][i]) if "sum" in b else cnt * float(b["avg"][i])
                    a[1] = mn if a[1] is None or mn < a[1] else a[1]
                    a[2] = mx if a[2] is None or mx > a[2] else a[2]
                    a[3] += sm
    out = []
    for key in sorted(order):
        cnt, mn, mx, sm = acc[key]
        valid = cnt > 0
        out.append({
            "window_start": key[0], "window_end": key[1], "count": cnt,
            "min": mn if valid else 0.0, "max": mx if valid else 0.0,
            "sum": sm if valid else 0.0,
            "avg": (sm / cnt) if valid else 0.0, "valid": valid,
        })
    return out
