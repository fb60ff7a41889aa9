"""denormalized_amd — MI355X-native implementation of Denormalized's
streaming windowed-aggregate + filter hot path.

The public surface mirrors the reference's DataStream pipeline
(crates/core/src/datastream.rs and the py-denormalized bindings,
py-denormalized/src/datastream.rs:145-270):

    ctx = Context(device=0)
    ds = (ctx.from_batches(batches)          # from_topic analog for in-memory sources
            .window(["sensor_name"],         # datastream.rs:178-196
                    [("count", "reading"), ("min", "reading"),
                     ("max", "reading"), ("avg", "reading")],
                    1000, None)              # length ms, slide ms (None = tumbling)
            .filter("max", ">", 113.0))      # datastream.rs:94-105
    for batch in ds.run():                   # print_stream/sink analog
        ...

All per-row compute runs in hand-written CDNA4 HIP kernels behind the C ABI
(include/denormalized_amd.h); there is no CPU fallback.
"""
from . import _lib
from ._lib import (WindowOp, JoinOp, JsonDecoder, DeviceArray,  # noqa: F401
                   generate, generate_utf8, generate_json, synchronize)

__version__ = "0.1"


class DataStream:
    """Mirror of the reference DataStream fluent API (datastream.rs)."""

    def __init__(self, ctx, batches, ts_col, key_col, key_kind):
        self._ctx = ctx
        self._batches = batches
        self._ts_col = ts_col
        self._key_col = key_col
        self._key_kind = key_kind
        self._window = None
        self._filters = []

    def window(self, group_cols, aggs, length_ms, slide_ms=None):
        """datastream.rs:178-196 (StreamingLogicalPlanBuilder::streaming_window,
        logical_plan/mod.rs:27-60). group_cols: [key column name]; aggs:
        [(op_name, value_col_name), ...]."""
        if len(group_cols) > 1:
            raise ValueError("hot-path shape: at most one group column")
        if self._window is not None:
            raise ValueError("window() already applied")
        vcols = {c for _, c in aggs}
        if len(vcols) != 1:
            raise ValueError("hot-path shape: one aggregate input column")
        ds = self._copy()
        ds._window = {
            "group_col": group_cols[0] if group_cols else None,
            "aggs": list(aggs),
            "value_col": vcols.pop(),
            "length_ms": int(length_ms),
            "slide_ms": int(slide_ms) if slide_ms else 0,
        }
        return ds

    def filter(self, col, cmp, literal):
        """datastream.rs:94-105 — a FilterExec over the window's output;
        pushed into the operator's emission (same semantics: NULL rows drop)."""
        ds = self._copy()
        ds._filters = self._filters + [(col, cmp, float(literal))]
        return ds

    def _copy(self):
        ds = DataStream(self._ctx, self._batches, self._ts_col, self._key_col,
                        self._key_kind)
        ds._window = self._window
        ds._filters = list(self._filters)
        return ds

    def _make_op(self, n_keys_hint=1024):
        w = self._window
        if w is None:
            raise ValueError("no window() in pipeline (hot path is window+filter)")
        op = WindowOp(length_ms=w["length_ms"], slide_ms=w["slide_ms"],
                      aggs=[(name, 0) for name, _ in w["aggs"]],
                      key_kind=self._key_kind, device=self._ctx.device,
                      n_keys_hint=n_keys_hint,
                      no_group=w["group_col"] is None)
        for (col, cmp, lit) in self._filters:
            names = [name for name, _ in w["aggs"]]
            if col not in names:
                raise ValueError(f"filter column {col!r} not in aggregate outputs")
            op.set_filter(col, cmp, lit)
        return op

    def run(self, n_keys_hint=1024):
        """Consume the source, yielding emitted window batches (the
        print_stream/sink_python analog, datastream.rs:311-374)."""
        op = self._make_op(n_keys_hint)
        w = self._window
        try:
            for b in self._batches:
                keys = b[w["group_col"]] if w["group_col"] is not None else None
                op.push(b[self._ts_col], keys, b[w["value_col"]],
                        b.get("_validity"))
                for out in op.poll_all():
                    yield out
            op.finish()
            for out in op.poll_all():
                yield out
        finally:
            op.close()

    def collect(self, n_keys_hint=1024):
        return list(self.run(n_keys_hint))

    def sink_python(self, callback, n_keys_hint=1024):
        """Feed each emitted batch to `callback` (the PyDataStream.sink_python
        analog, py-denormalized/src/datastream.rs:232-270)."""
        for out in self.run(n_keys_hint):
            callback(out)

    def print_stream(self):
        for out in self.run():
            n = out["n_rows"]
            for i in range(n):
                row = {k: (v[i] if hasattr(v, "__getitem__") else v)
                       for k, v in out.items() if k != "n_rows"}
                print(row)


class Context:
    """Mirror of the reference Context (crates/core/src/context.rs:25-72)."""

    def __init__(self, device=0):
        self.device = device

    def from_batches(self, batches, ts_col="occurred_at_ms",
                     key_col="sensor_name", key_kind=None):
        """from_topic analog for an in-memory source (context.rs:65-72 registers
        the Kafka TableProvider; here any iterable of column dicts). Each batch:
        {ts_col: int64 ms array, key_col: list[str] or int64 array,
        value cols: float64 arrays}."""
        if key_kind is None:
            first = batches[0] if isinstance(batches, (list, tuple)) and batches else None
            if first is not None and len(first[key_col]) and isinstance(first[key_col][0], (str, bytes)):
                key_kind = _lib.KEY_UTF8
            else:
                key_kind = _lib.KEY_INT64
        return DataStream(self, batches, ts_col, key_col, key_kind)
