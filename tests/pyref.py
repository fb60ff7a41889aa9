"""Independent pure-Python restatement of the reference semantics.

Deliberately written WITHOUT looking at oracle/oracle.c's structure — a second,
independent restatement of the same reference code paths, used to cross-check
the C oracle on small cases. Citations as in oracle/oracle.c (the authoritative
list): grouped_window_agg_stream.rs:326-420/:501-605/:220-266,
streaming_window.rs:1053-1094, continuous/mod.rs:42-89, time.rs:31-57.
"""


def snap(ts_ms, len_ms):
    # streaming_window.rs:1088-1094 (whole-second truncation);
    # ms generalization for sub-second lengths
    len_s = len_ms // 1000
    if len_s == 0:
        return ts_ms - (ts_ms % len_ms)
    return (ts_ms // 1000) // len_s * len_s * 1000


def windows_for_range(mn, mx, len_ms, slide_ms=0):
    out = []
    if slide_ms > 0:
        cur = snap(mn - len_ms, len_ms)
        while cur <= mx:
            end = cur + len_ms
            if not (mn > end or mx < cur):
                out.append((cur, end))
            cur += slide_ms
    else:
        cur = snap(mn, len_ms)
        while cur <= mx:
            out.append((cur, cur + len_ms))
            cur += len_ms
    return out


class PyRef:
    def __init__(self, len_ms, slide_ms=0):
        self.len_ms = len_ms
        self.slide_ms = slide_ms
        self.frames = {}  # start -> (end, {key: [cnt, min, max, sum]}, key insertion order list)
        self.watermark = None
        self.out = []  # rows: (key, cnt, min, max, avg, sum, valid, wstart, wend)

    def push(self, ts, keys, vals, valid=None):
        n = len(ts)
        if n == 0:
            return
        mn, mx = min(ts), max(ts)
        for (ws, we) in windows_for_range(mn, mx, self.len_ms, self.slide_ms):
            if ws not in self.frames:
                self.frames[ws] = (we, {}, [])
            _, table, order = self.frames[ws]
            for i in range(n):
                t = ts[i]
                if t < ws or t >= we:
                    continue
                k = keys[i]
                if k not in table:
                    table[k] = [0, None, None, 0.0]
                    order.append(k)
                if valid is None or valid[i]:
                    a = table[k]
                    v = vals[i]
                    a[0] += 1
                    if a[1] is None:
                        a[1] = v
                        a[2] = v
                    else:
                        if v < a[1]:
                            a[1] = v
                        if v > a[2]:
                            a[2] = v
                    a[3] += v
        if self.watermark is None or self.watermark <= mn:
            self.watermark = mn
        self._trigger()

    def _trigger(self):
        if self.watermark is None:
            return
        for ws in sorted(self.frames):
            we, table, order = self.frames[ws]
            if self.watermark >= we:
                for k in order:
                    cnt, lo, hi, s = table[k]
                    ok = cnt > 0
                    self.out.append((
                        k, cnt,
                        lo if ok else 0.0, hi if ok else 0.0,
                        (s / cnt) if ok else 0.0, s if ok else 0.0,
                        1 if ok else 0, ws, we))
                del self.frames[ws]

    def finish(self):
        mx = self.watermark or 0
        for ws, (we, _, _) in self.frames.items():
            mx = max(mx, we)
        self.watermark = mx
        self._trigger()
