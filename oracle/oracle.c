/* oracle/oracle.c — CPU restatement of Denormalized's grouped streaming
 * window aggregate (the hot path named by BASELINE.json `north_star`).
 *
 * TEST INFRASTRUCTURE ONLY. This file is the parity ORACLE: only tests/,
 * __graft_entry__.smoke() and bench.py's `cpu_baseline` leg may call it.
 * The product path (denormalized_amd/) must never import, link or route
 * through this code; it exists to CHECK the HIP path, and to be timed as
 * the CPU baseline.
 *
 * PARITY PINNING STATUS: partially pinned. The reference's windowed-aggregate
 * operator has no tests of its own (its only tests are commented out:
 * crates/core/src/physical_optimizer/coalesce_before_streaming_window_aggregate.rs:97-240),
 * and the aggregate arithmetic lives in an unvendored DataFusion fork
 * (probably-nothing-labs/arrow-datafusion @ d812edc..., Cargo.toml:37) that
 * cannot be compiled in this environment (no Rust toolchain, no network).
 * What IS pinned by the reference's own tests: the avg accumulator state
 * layout and finalization [sum f64, count u64] -> sum/count
 * (crates/core/src/utils/serialization.rs:534-557, value 112.0/2 = 56.0),
 * replicated in tests/test_oracle.py. Everything else is pinned by this
 * restatement cross-checked against hand-computed SQL-standard vectors and
 * an independent pure-Python restatement (tests/golden/gen.py). NaN ordering
 * for min/max is therefore "parity unpinned": we define first-non-null
 * initialisation + strict `<` / `>` updates (NaN never replaces a non-NaN
 * value once one is seen) and document it in DESIGN.md.
 *
 * Semantics restated from (file:line into /root/reference):
 *  - poll loop: crates/core/src/physical_plan/continuous/grouped_window_agg_stream.rs:326-420
 *  - window ranges: crates/core/src/physical_plan/continuous/streaming_window.rs:1053-1086
 *  - snap_to_window_start (whole-second granularity): streaming_window.rs:1088-1094
 *    (sub-second windows divide by zero in the reference; we use the ms
 *     generalization start = ts - ts % len_ms)
 *  - row routing [start,end): grouped_window_agg_stream.rs:548-605
 *  - group intern + accumulate: grouped_window_agg_stream.rs:501-537
 *    (per-frame GroupValues hash table, insertion-order group ids; DF fork)
 *  - watermark: physical_plan/utils/time.rs:31-57 (per-batch min/max of
 *    canonical_timestamp) + process_watermark :255-266 (running max of
 *    batch minimums, never decreasing)
 *  - trigger/emit: grouped_window_agg_stream.rs:220-253 (frames in ascending
 *    window-start order with watermark >= window_end emit and are removed;
 *    group rows in insertion order), window columns appended:
 *    continuous/mod.rs:64-89
 *  - accumulators (DF fork, unvendored — see pinning note): count = number of
 *    non-null values (i64); min/max null-skipping f64; avg = [sum f64,
 *    count u64] finalized sum/count at emit.
 *
 * Build: gcc -O2 -std=c11 -shared -fPIC oracle.c -o liboracle.so  (see Makefile)
 */

#include <stdint.h>
#include <stdlib.h>
#include <string.h>
#include <stdio.h>

/* ------------------------------------------------------------------ */
/* Window math                                                         */
/* ------------------------------------------------------------------ */

/* streaming_window.rs:1088-1094. The reference converts both the timestamp
 * and the window length to WHOLE SECONDS before snapping; a window length
 * of e.g. 1500 ms snaps to 1-second multiples (len_secs = 1) while windows
 * still advance by the full 1500 ms. Sub-second lengths (len_secs == 0)
 * divide by zero in the reference; we generalize to ms granularity. */
static int64_t snap_to_window_start(int64_t ts_ms, int64_t len_ms) {
    int64_t len_secs = len_ms / 1000;
    if (len_secs == 0) {
        return ts_ms - (ts_ms % len_ms); /* ms generalization */
    }
    int64_t ts_secs = ts_ms / 1000; /* as_secs() truncation */
    return (ts_secs / len_secs) * len_secs * 1000;
}

/* streaming_window.rs:1053-1086. Returns number of (start,end) ranges
 * written (up to cap). slide_ms == 0 means tumbling. */
int64_t orc_windows_for_range(int64_t min_ts, int64_t max_ts,
                              int64_t len_ms, int64_t slide_ms,
                              int64_t* starts, int64_t* ends, int64_t cap) {
    int64_t n = 0;
    if (slide_ms > 0) { /* Sliding(window_length, slide), :1062-1075 */
        int64_t cur = snap_to_window_start(min_ts - len_ms, len_ms);
        while (cur <= max_ts) {
            int64_t cur_end = cur + len_ms;
            if (min_ts > cur_end || max_ts < cur) { cur += slide_ms; continue; }
            if (n < cap) { starts[n] = cur; ends[n] = cur_end; }
            n++;
            cur += slide_ms;
        }
    } else { /* Tumbling(window_length), :1076-1082 */
        int64_t cur = snap_to_window_start(min_ts, len_ms);
        while (cur <= max_ts) {
            int64_t cur_end = cur + len_ms;
            if (n < cap) { starts[n] = cur; ends[n] = cur_end; }
            n++;
            cur = cur_end;
        }
    }
    return n;
}

/* ------------------------------------------------------------------ */
/* Per-frame state: insertion-order group table + accumulators         */
/* (restates the DF fork's GroupValues + GroupsAccumulators, see header)*/
/* ------------------------------------------------------------------ */

typedef struct {
    /* open-address hash: slot -> group index + 1 (0 = empty) */
    uint32_t* slots;
    uint64_t  mask;      /* table size - 1 (power of two) */
    /* group arrays, insertion order (GroupValues emits insertion order) */
    int64_t* keys;
    int64_t* cnt;       /* count of non-null values */
    double*  vmin;
    double*  vmax;
    double*  sum;
    uint8_t* has_val;   /* group has seen at least one non-null value */
    int64_t  ngroups;
    int64_t  cap;
} GroupTable;

typedef struct {
    int64_t start_ms, end_ms;
    GroupTable gt;
} Frame;

typedef struct OrcOp {
    int64_t len_ms, slide_ms;
    /* open frames, kept sorted by start_ms ascending (BTreeMap order) */
    Frame* frames;
    int64_t nframes, frames_cap;
    /* watermark */
    int64_t watermark_ms;
    int     has_watermark;
    /* emitted output rows (appended at trigger, fetched by caller) */
    int64_t* o_keys; int64_t* o_cnt; double* o_min; double* o_max;
    double* o_avg; double* o_sum; uint8_t* o_valid; /* min/max/avg/sum validity */
    int64_t* o_wstart; int64_t* o_wend;
    int64_t o_n, o_cap;
    char err[256];
} OrcOp;

static uint64_t hash_key(int64_t k) {
    /* splitmix64 finalizer — internal detail, any hash works */
    uint64_t x = (uint64_t)k;
    x += 0x9e3779b97f4a7c15ULL;
    x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ULL;
    x = (x ^ (x >> 27)) * 0x94d049bb133111ebULL;
    return x ^ (x >> 31);
}

static void gt_init(GroupTable* gt) {
    gt->mask = 63;
    gt->slots = (uint32_t*)calloc(gt->mask + 1, sizeof(uint32_t));
    gt->cap = 48;
    gt->keys = (int64_t*)malloc(gt->cap * sizeof(int64_t));
    gt->cnt = (int64_t*)malloc(gt->cap * sizeof(int64_t));
    gt->vmin = (double*)malloc(gt->cap * sizeof(double));
    gt->vmax = (double*)malloc(gt->cap * sizeof(double));
    gt->sum = (double*)malloc(gt->cap * sizeof(double));
    gt->has_val = (uint8_t*)malloc(gt->cap);
    gt->ngroups = 0;
}

static void gt_free(GroupTable* gt) {
    free(gt->slots); free(gt->keys); free(gt->cnt); free(gt->vmin);
    free(gt->vmax); free(gt->sum); free(gt->has_val);
    memset(gt, 0, sizeof(*gt));
}

static void gt_grow_table(GroupTable* gt) {
    uint64_t nmask = gt->mask * 2 + 1;
    uint32_t* ns = (uint32_t*)calloc(nmask + 1, sizeof(uint32_t));
    for (int64_t g = 0; g < gt->ngroups; g++) {
        uint64_t h = hash_key(gt->keys[g]) & nmask;
        while (ns[h]) h = (h + 1) & nmask;
        ns[h] = (uint32_t)(g + 1);
    }
    free(gt->slots);
    gt->slots = ns;
    gt->mask = nmask;
}

static int64_t gt_intern(GroupTable* gt, int64_t key) {
    if ((uint64_t)gt->ngroups * 10 >= (gt->mask + 1) * 7) gt_grow_table(gt);
    uint64_t h = hash_key(key) & gt->mask;
    for (;;) {
        uint32_t s = gt->slots[h];
        if (s == 0) break;
        if (gt->keys[s - 1] == key) return (int64_t)(s - 1);
        h = (h + 1) & gt->mask;
    }
    if (gt->ngroups == gt->cap) {
        gt->cap *= 2;
        gt->keys = (int64_t*)realloc(gt->keys, gt->cap * sizeof(int64_t));
        gt->cnt = (int64_t*)realloc(gt->cnt, gt->cap * sizeof(int64_t));
        gt->vmin = (double*)realloc(gt->vmin, gt->cap * sizeof(double));
        gt->vmax = (double*)realloc(gt->vmax, gt->cap * sizeof(double));
        gt->sum = (double*)realloc(gt->sum, gt->cap * sizeof(double));
        gt->has_val = (uint8_t*)realloc(gt->has_val, gt->cap);
    }
    int64_t g = gt->ngroups++;
    gt->keys[g] = key;
    gt->cnt[g] = 0;
    gt->vmin[g] = 0.0;
    gt->vmax[g] = 0.0;
    gt->sum[g] = 0.0;
    gt->has_val[g] = 0;
    gt->slots[h] = (uint32_t)(g + 1);
    return g;
}

/* ------------------------------------------------------------------ */
/* Op lifecycle                                                        */
/* ------------------------------------------------------------------ */

OrcOp* orc_create(int64_t window_length_ms, int64_t slide_ms) {
    if (window_length_ms <= 0) return NULL;
    OrcOp* op = (OrcOp*)calloc(1, sizeof(OrcOp));
    op->len_ms = window_length_ms;
    op->slide_ms = slide_ms;
    return op;
}

void orc_destroy(OrcOp* op) {
    if (!op) return;
    for (int64_t i = 0; i < op->nframes; i++) gt_free(&op->frames[i].gt);
    free(op->frames);
    free(op->o_keys); free(op->o_cnt); free(op->o_min); free(op->o_max);
    free(op->o_avg); free(op->o_sum); free(op->o_valid);
    free(op->o_wstart); free(op->o_wend);
    free(op);
}

static Frame* find_or_create_frame(OrcOp* op, int64_t start, int64_t end) {
    /* binary search by start */
    int64_t lo = 0, hi = op->nframes;
    while (lo < hi) {
        int64_t mid = (lo + hi) / 2;
        if (op->frames[mid].start_ms < start) lo = mid + 1; else hi = mid;
    }
    if (lo < op->nframes && op->frames[lo].start_ms == start)
        return &op->frames[lo];
    /* insert at lo (ensure_window_frames_for_ranges, :276-313 — frames are
     * re-created even for windows that already closed: late data re-emits) */
    if (op->nframes == op->frames_cap) {
        op->frames_cap = op->frames_cap ? op->frames_cap * 2 : 16;
        op->frames = (Frame*)realloc(op->frames, op->frames_cap * sizeof(Frame));
    }
    memmove(&op->frames[lo + 1], &op->frames[lo],
            (op->nframes - lo) * sizeof(Frame));
    op->nframes++;
    Frame* f = &op->frames[lo];
    f->start_ms = start;
    f->end_ms = end;
    gt_init(&f->gt);
    return f;
}

static void out_reserve(OrcOp* op, int64_t add) {
    if (op->o_n + add <= op->o_cap) return;
    int64_t nc = op->o_cap ? op->o_cap : 1024;
    while (nc < op->o_n + add) nc *= 2;
    op->o_keys = (int64_t*)realloc(op->o_keys, nc * sizeof(int64_t));
    op->o_cnt = (int64_t*)realloc(op->o_cnt, nc * sizeof(int64_t));
    op->o_min = (double*)realloc(op->o_min, nc * sizeof(double));
    op->o_max = (double*)realloc(op->o_max, nc * sizeof(double));
    op->o_avg = (double*)realloc(op->o_avg, nc * sizeof(double));
    op->o_sum = (double*)realloc(op->o_sum, nc * sizeof(double));
    op->o_valid = (uint8_t*)realloc(op->o_valid, nc);
    op->o_wstart = (int64_t*)realloc(op->o_wstart, nc * sizeof(int64_t));
    op->o_wend = (int64_t*)realloc(op->o_wend, nc * sizeof(int64_t));
    op->o_cap = nc;
}

/* trigger_windows: grouped_window_agg_stream.rs:220-253. Emits every frame
 * whose end <= watermark, ascending start order, groups in insertion order. */
static void trigger_windows(OrcOp* op) {
    if (!op->has_watermark) return;
    int64_t w = 0; /* compaction write cursor over frames kept open */
    for (int64_t i = 0; i < op->nframes; i++) {
        Frame* f = &op->frames[i];
        if (op->watermark_ms >= f->end_ms) {
            GroupTable* gt = &f->gt;
            out_reserve(op, gt->ngroups);
            for (int64_t g = 0; g < gt->ngroups; g++) {
                int64_t o = op->o_n++;
                op->o_keys[o] = gt->keys[g];
                op->o_cnt[o] = gt->cnt[g];
                int valid = gt->cnt[g] > 0; /* all-null group: min/max/avg NULL */
                op->o_valid[o] = (uint8_t)valid;
                op->o_min[o] = valid ? gt->vmin[g] : 0.0;
                op->o_max[o] = valid ? gt->vmax[g] : 0.0;
                op->o_sum[o] = valid ? gt->sum[g] : 0.0;
                /* avg finalize = sum/count: utils/serialization.rs:534-557 */
                op->o_avg[o] = valid ? gt->sum[g] / (double)gt->cnt[g] : 0.0;
                op->o_wstart[o] = f->start_ms;
                op->o_wend[o] = f->end_ms;
            }
            gt_free(gt);
        } else {
            op->frames[w++] = *f;
        }
    }
    op->nframes = w;
}

/* One input batch == one poll of the stream (poll_next_inner :326-420).
 * ts_ms must be non-null (the reference's canonical_timestamp always is:
 * kafka_stream_read.rs:254-268 computes it for every row). val_valid may be
 * NULL meaning all values valid. */
void orc_push(OrcOp* op, int64_t nrows, const int64_t* ts_ms,
              const int64_t* keys, const double* vals,
              const uint8_t* val_valid) {
    if (nrows <= 0) return;
    /* RecordBatchWatermark::try_from — min/max of the batch (time.rs:31-57) */
    int64_t mn = ts_ms[0], mx = ts_ms[0];
    for (int64_t i = 1; i < nrows; i++) {
        if (ts_ms[i] < mn) mn = ts_ms[i];
        if (ts_ms[i] > mx) mx = ts_ms[i];
    }
    /* get_windows_for_watermark + ensure frames + per-frame push */
    int64_t cap = 4096;
    int64_t* ws = (int64_t*)malloc(cap * sizeof(int64_t));
    int64_t* we = (int64_t*)malloc(cap * sizeof(int64_t));
    int64_t nw = orc_windows_for_range(mn, mx, op->len_ms, op->slide_ms, ws, we, cap);
    if (nw > cap) { /* enormous span; regrow */
        free(ws); free(we);
        ws = (int64_t*)malloc(nw * sizeof(int64_t));
        we = (int64_t*)malloc(nw * sizeof(int64_t));
        orc_windows_for_range(mn, mx, op->len_ms, op->slide_ms, ws, we, nw);
    }
    for (int64_t r = 0; r < nw; r++) {
        Frame* f = find_or_create_frame(op, ws[r], we[r]);
        /* frame.push: [start,end) ts routing (:548-605) then
         * group_aggregate_batch (:501-537), rows in batch order */
        GroupTable* gt = &f->gt;
        for (int64_t i = 0; i < nrows; i++) {
            int64_t t = ts_ms[i];
            if (t < f->start_ms || t >= f->end_ms) continue;
            int64_t g = gt_intern(gt, keys[i]);
            int valid = val_valid ? val_valid[i] : 1;
            if (valid) {
                double v = vals[i];
                gt->cnt[g] += 1;
                if (!gt->has_val[g]) {
                    gt->has_val[g] = 1;
                    gt->vmin[g] = v;
                    gt->vmax[g] = v;
                } else {
                    if (v < gt->vmin[g]) gt->vmin[g] = v;
                    if (v > gt->vmax[g]) gt->vmax[g] = v;
                }
                gt->sum[g] += v; /* row-order f64 fold — parity-critical */
            }
        }
    }
    free(ws); free(we);
    /* process_watermark (:255-266): running max of batch minimums */
    if (!op->has_watermark || op->watermark_ms <= mn) {
        op->watermark_ms = mn;
        op->has_watermark = 1;
    }
    trigger_windows(op);
}

/* Close every remaining frame (extension for finite runs: equivalent to the
 * watermark advancing past every open window end). */
void orc_finish(OrcOp* op) {
    int64_t mx = op->has_watermark ? op->watermark_ms : 0;
    for (int64_t i = 0; i < op->nframes; i++)
        if (op->frames[i].end_ms > mx) mx = op->frames[i].end_ms;
    op->watermark_ms = mx;
    op->has_watermark = 1;
    trigger_windows(op);
}

int64_t orc_out_rows(OrcOp* op) { return op->o_n; }

/* Copies all pending output rows into caller buffers and clears the queue.
 * Column order mirrors the reference output schema (group key, aggregates
 * in declaration order, window_start_time, window_end_time —
 * streaming_window.rs:1096-1134 + continuous/mod.rs:42-62). */
void orc_out_fetch(OrcOp* op, int64_t* keys, int64_t* cnt, double* vmin,
                   double* vmax, double* avg, double* sum, uint8_t* valid,
                   int64_t* wstart, int64_t* wend) {
    int64_t n = op->o_n;
    if (keys) memcpy(keys, op->o_keys, n * sizeof(int64_t));
    if (cnt) memcpy(cnt, op->o_cnt, n * sizeof(int64_t));
    if (vmin) memcpy(vmin, op->o_min, n * sizeof(double));
    if (vmax) memcpy(vmax, op->o_max, n * sizeof(double));
    if (avg) memcpy(avg, op->o_avg, n * sizeof(double));
    if (sum) memcpy(sum, op->o_sum, n * sizeof(double));
    if (valid) memcpy(valid, op->o_valid, n);
    if (wstart) memcpy(wstart, op->o_wstart, n * sizeof(int64_t));
    if (wend) memcpy(wend, op->o_wend, n * sizeof(int64_t));
    op->o_n = 0;
}

int64_t orc_open_frames(OrcOp* op) { return op->nframes; }
int64_t orc_watermark(OrcOp* op) { return op->has_watermark ? op->watermark_ms : INT64_MIN; }

/* ------------------------------------------------------------------ */
/* Deterministic synthetic sensor stream (shared spec with the GPU      */
/* generator; mirrors examples/examples/emit_measurements.rs:30-67 +    */
/* examples/src/lib.rs:3-17: keys "sensor_{i}", reading = U[0,115) f64, */
/* occurred_at_ms monotonic).                                           */
/*                                                                      */
/* Spec: row i (global index) has                */
/*   ts_ms  = t0 + i / rows_per_ms                                      */
/*   r      = splitmix64(seed ^ (0x9e3779b97f4a7c15 * (i+1)))           */
/*   kid    = r % nkeys                                                 */
/*   val    = ((splitmix64(r) >> 11) * 2^-53) * 115.0                   */
/* ------------------------------------------------------------------ */

static uint64_t splitmix64(uint64_t x) {
    x += 0x9e3779b97f4a7c15ULL;
    x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ULL;
    x = (x ^ (x >> 27)) * 0x94d049bb133111ebULL;
    return x ^ (x >> 31);
}

void orc_gen(uint64_t seed, int64_t t0_ms, int64_t start_row, int64_t nrows,
             int64_t nkeys, int64_t rows_per_ms,
             int64_t* ts_ms, int64_t* kid, double* val) {
    for (int64_t i = 0; i < nrows; i++) {
        int64_t gi = start_row + i;
        uint64_t r = splitmix64(seed ^ (0x9e3779b97f4a7c15ULL * (uint64_t)(gi + 1)));
        if (ts_ms) ts_ms[i] = t0_ms + gi / rows_per_ms;
        if (kid) kid[i] = (int64_t)(r % (uint64_t)nkeys);
        if (val) val[i] = (double)(splitmix64(r) >> 11) * (1.0 / 9007199254740992.0) * 115.0;
    }
}

/* ------------------------------------------------------------------ */
/* Stream join restatement (BASELINE cfg5: rideshare-style two-topic   */
/* inner equi-join on trip_id feeding the windowed group-by).          */
/*                                                                     */
/* Reference anchors: DataStream::join/join_on lowers to DataFusion's  */
/* standard inner hash join (crates/core/src/datastream.rs:126-175;    */
/* the arithmetic lives in the unvendored DataFusion fork — parity     */
/* pinning as for the aggregates, see the header note). The STREAMING  */
/* emission discipline is defined HERE (the reference's only join      */
/* example joins two closed-window outputs, examples/stream_join.rs):  */
/*  - build side (trips): (trip_id -> driver_id), at most one row per  */
/*    trip_id (later duplicates overwrite — dimension-table update);   */
/*  - probe side (events): each pushed batch emits its matching rows   */
/*    IN ROW ORDER; rows with no match yet are buffered IN ROW ORDER;  */
/*  - each build push re-probes the buffer and emits newly matched     */
/*    rows in their original buffered order (late-data semantics of    */
/*    the downstream window op apply);                                 */
/*  - inner join: rows never matched are dropped.                      */
/* The GPU join implements the SAME discipline, so join->window        */
/* pipelines compare bit-exactly.                                      */
/* ------------------------------------------------------------------ */

typedef struct {
    int64_t cap, mask, n;   /* open-address trip table, power-of-two */
    int64_t* trips;         /* INT64_MIN = empty */
    int64_t* drivers;
    /* unmatched probe rows, in arrival order */
    int64_t un_cap, un_n;
    int64_t* u_ts; int64_t* u_trip; double* u_val;
    /* matched output, cleared by fetch */
    int64_t o_cap, o_n;
    int64_t* o_ts; int64_t* o_driver; double* o_val;
} OrcJoin;

static void oj_out(OrcJoin* j, int64_t ts, int64_t drv, double v) {
    if (j->o_n == j->o_cap) {
        j->o_cap = j->o_cap ? j->o_cap * 2 : 1024;
        j->o_ts = (int64_t*)realloc(j->o_ts, j->o_cap * 8);
        j->o_driver = (int64_t*)realloc(j->o_driver, j->o_cap * 8);
        j->o_val = (double*)realloc(j->o_val, j->o_cap * 8);
    }
    j->o_ts[j->o_n] = ts; j->o_driver[j->o_n] = drv; j->o_val[j->o_n] = v;
    j->o_n++;
}

static int64_t oj_find(OrcJoin* j, int64_t trip) {
    uint64_t h = (uint64_t)trip * 0x9e3779b97f4a7c15ULL;
    int64_t s = (int64_t)(h & (uint64_t)j->mask);
    while (j->trips[s] != INT64_MIN) {
        if (j->trips[s] == trip) return s;
        s = (s + 1) & j->mask;
    }
    return ~s; /* not found; ~slot = insertion point */
}

OrcJoin* orc_join_create(void) {
    OrcJoin* j = (OrcJoin*)calloc(1, sizeof(OrcJoin));
    j->cap = 1024; j->mask = j->cap - 1;
    j->trips = (int64_t*)malloc(j->cap * 8);
    j->drivers = (int64_t*)malloc(j->cap * 8);
    for (int64_t i = 0; i < j->cap; i++) j->trips[i] = INT64_MIN;
    return j;
}

void orc_join_destroy(OrcJoin* j) {
    if (!j) return;
    free(j->trips); free(j->drivers);
    free(j->u_ts); free(j->u_trip); free(j->u_val);
    free(j->o_ts); free(j->o_driver); free(j->o_val);
    free(j);
}

void orc_join_push_build(OrcJoin* j, int64_t n, const int64_t* trips,
                         const int64_t* drivers) {
    for (int64_t i = 0; i < n; i++) {
        if (j->n * 2 >= j->cap) { /* grow + rehash */
            int64_t oc = j->cap;
            int64_t* ot = j->trips; int64_t* od = j->drivers;
            j->cap *= 2; j->mask = j->cap - 1; j->n = 0;
            j->trips = (int64_t*)malloc(j->cap * 8);
            j->drivers = (int64_t*)malloc(j->cap * 8);
            for (int64_t s = 0; s < j->cap; s++) j->trips[s] = INT64_MIN;
            for (int64_t s = 0; s < oc; s++)
                if (ot[s] != INT64_MIN) {
                    int64_t ns = oj_find(j, ot[s]);
                    ns = ~ns;
                    j->trips[ns] = ot[s]; j->drivers[ns] = od[s]; j->n++;
                }
            free(ot); free(od);
        }
        int64_t s = oj_find(j, trips[i]);
        if (s < 0) { s = ~s; j->trips[s] = trips[i]; j->n++; }
        j->drivers[s] = drivers[i]; /* later duplicate overwrites */
    }
    /* re-probe the unmatched buffer in original order */
    int64_t w = 0;
    for (int64_t i = 0; i < j->un_n; i++) {
        int64_t s = oj_find(j, j->u_trip[i]);
        if (s >= 0) {
            oj_out(j, j->u_ts[i], j->drivers[s], j->u_val[i]);
        } else {
            j->u_ts[w] = j->u_ts[i]; j->u_trip[w] = j->u_trip[i];
            j->u_val[w] = j->u_val[i]; w++;
        }
    }
    j->un_n = w;
}

void orc_join_push_probe(OrcJoin* j, int64_t n, const int64_t* ts,
                         const int64_t* trips, const double* vals) {
    for (int64_t i = 0; i < n; i++) {
        int64_t s = oj_find(j, trips[i]);
        if (s >= 0) {
            oj_out(j, ts[i], j->drivers[s], vals[i]);
        } else {
            if (j->un_n == j->un_cap) {
                j->un_cap = j->un_cap ? j->un_cap * 2 : 1024;
                j->u_ts = (int64_t*)realloc(j->u_ts, j->un_cap * 8);
                j->u_trip = (int64_t*)realloc(j->u_trip, j->un_cap * 8);
                j->u_val = (double*)realloc(j->u_val, j->un_cap * 8);
            }
            j->u_ts[j->un_n] = ts[i]; j->u_trip[j->un_n] = trips[i];
            j->u_val[j->un_n] = vals[i]; j->un_n++;
        }
    }
}

int64_t orc_join_out_rows(OrcJoin* j) { return j->o_n; }
int64_t orc_join_unmatched(OrcJoin* j) { return j->un_n; }

void orc_join_out_fetch(OrcJoin* j, int64_t* ts, int64_t* drivers,
                        double* vals) {
    memcpy(ts, j->o_ts, j->o_n * 8);
    memcpy(drivers, j->o_driver, j->o_n * 8);
    memcpy(vals, j->o_val, j->o_n * 8);
    j->o_n = 0;
}
