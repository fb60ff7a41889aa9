/* denormalized_amd.h — C ABI of the MI355X-native streaming windowed-aggregate
 * operator (the drop-in replacement for Denormalized's hot path).
 *
 * Boundary contract. Each entry point cites the reference
 * interface it replaces (file:line into /root/reference, the public
 * probably-nothing-labs/denormalized tree @ 2024-12-18). The reference host is
 * Rust; a maintainer binds this library with a plain `extern "C"` FFI block —
 * see INTEGRATION.md for the Rust-side stub. No GPU-runtime types cross this
 * boundary: plain pointers and sizes only.
 */
#ifndef DENORMALIZED_AMD_H
#define DENORMALIZED_AMD_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ------------------------------------------------------------------ */
/* Status and descriptors                                              */
/* ------------------------------------------------------------------ */

typedef enum dz_status {
    DZ_OK = 0,
    DZ_ERR = 1,          /* inspect dz_last_error() */
} dz_status;

/* PhysicalStreamingWindowType (crates/core/src/physical_plan/continuous/
 * streaming_window.rs; Session is todo!() in the reference,
 * streaming_window.rs:1062). */
typedef enum dz_window_type {
    DZ_WINDOW_TUMBLING = 0,   /* Tumbling(length) */
    DZ_WINDOW_SLIDING  = 1,   /* Sliding(length, slide) */
} dz_window_type;

/* The aggregate set of the hot path (simple_aggregation.rs:46-52:
 * count/min/max/avg; sum is the avg partial and is also exposed). */
typedef enum dz_agg_op {
    DZ_AGG_COUNT = 0,
    DZ_AGG_MIN   = 1,
    DZ_AGG_MAX   = 2,
    DZ_AGG_SUM   = 3,
    DZ_AGG_AVG   = 4,
} dz_agg_op;

typedef enum dz_key_kind {
    DZ_KEY_UTF8        = 0,  /* dictionary-encoded host-side, first-seen ids */
    DZ_KEY_INT64       = 1,  /* arbitrary int64 keys, host dictionary        */
    DZ_KEY_DENSE_INT64 = 2,  /* keys are already dense ids in [0, n_keys)    */
} dz_key_kind;

typedef struct dz_agg_desc {
    dz_agg_op op;
    int32_t input_col;        /* column index in the pushed batch (f64) */
} dz_agg_desc;

/* Construction parameters. Mirrors what the planner hands StreamingWindowExec:
 * (mode, group-by, aggregate exprs, window type) —
 * crates/core/src/planner/streaming_window.rs:133-165 and
 * StreamingWindowExec::new (physical_plan/continuous/streaming_window.rs:201+).
 * This build supports the hot path's shape: one group column, aggregates over
 * one f64 value column, mode Single (grouped). */
typedef struct dz_window_desc {
    dz_window_type window_type;
    int64_t length_ms;
    int64_t slide_ms;          /* ignored for tumbling */
    int32_t ts_col;            /* Timestamp(ms) column index (the reference's
                                * _streaming_internal_metadata.canonical_timestamp,
                                * crates/common/src/lib.rs:5) */
    int32_t group_col;
    dz_key_kind key_kind;
    const dz_agg_desc* aggs;
    int32_t n_aggs;
    int64_t n_keys_hint;       /* initial key-capacity; grows as needed */
    int32_t device;            /* HIP device ordinal */
    int32_t max_open_windows;  /* 0 => 64 */
} dz_window_desc;

/* ------------------------------------------------------------------ */
/* Batches — Arrow C-data-interface-style buffers                      */
/* (the reference operator consumes/produces Arrow RecordBatches:      */
/*  grouped_window_agg_stream.rs:548-605; caller owns input, the op    */
/*  copies what it keeps; the op owns an output batch until the next   */
/*  poll on the same handle)                                           */
/* ------------------------------------------------------------------ */

typedef struct dz_column {
    int64_t len;
    const uint8_t* validity;   /* Arrow validity bitmap (LSB-first), NULL = all valid */
    const int32_t* offsets;    /* utf8 columns only: len+1 entries */
    const void* data;          /* i64 / f64 values, or utf8 bytes */
} dz_column;

typedef struct dz_batch {
    int64_t n_rows;
    int32_t n_cols;
    const dz_column* cols;
} dz_batch;

/* Emitted window batch. Column order mirrors the reference output schema:
 * group key, aggregates in declaration order, then window_start_time and
 * window_end_time (Timestamp(ms)) — streaming_window.rs:1096-1134 +
 * continuous/mod.rs:42-62. Rows are groups in insertion (first-seen) order
 * per window, windows in ascending start order (GroupValues emits insertion
 * order; trigger iterates the BTreeMap: grouped_window_agg_stream.rs:220-253). */
typedef struct dz_out_batch {
    int64_t n_rows;
    /* group key column (one of): */
    const int64_t* key_i64;        /* key kinds INT64 / DENSE_INT64 */
    const int32_t* key_offsets;    /* key kind UTF8 */
    const char*    key_data;
    /* aggregate columns, one per dz_window_desc.aggs entry:
     * COUNT -> const int64_t*, others -> const double* */
    const void* const* agg_cols;
    const uint8_t* agg_valid;      /* byte mask (1=valid) for min/max/sum/avg
                                    * (all-null groups); count is never null */
    const int64_t* window_start_ms;
    const int64_t* window_end_ms;
} dz_out_batch;

/* ------------------------------------------------------------------ */
/* Operator lifecycle                                                  */
/* ------------------------------------------------------------------ */

typedef struct dz_window_op dz_window_op;

/* ExecutionPlan construction + execute(partition) →
 * GroupedWindowAggStream::new (streaming_window.rs:421-482,
 * grouped_window_agg_stream.rs:111-214). One handle = one partition's
 * stream; single-consumer. Returns NULL on error (see dz_last_error(NULL)). */
dz_window_op* dz_window_op_create(const dz_window_desc* desc);

/* One input batch == one poll of the input stream
 * (poll_next_inner: grouped_window_agg_stream.rs:326-420): computes the batch
 * watermark, ensures window frames, routes + aggregates rows, advances the
 * shared watermark, triggers closed windows. */
dz_status dz_window_op_push(dz_window_op* op, const dz_batch* batch);

/* Device-resident push: same results as dz_window_op_push but the three
 * buffers already live in the op's device HBM (keys as dense int32 ids).
 * This is the bench's timed entry (inputs resident per measurement contract);
 * the host-buffer path above is the reference-shaped boundary.
 * Pipelined: the call stages the inputs (they may be freed once it returns)
 * and enqueues the batch's reduction, then returns; routing/aggregation and
 * any window closes complete at the NEXT call into the op (push/poll/
 * advance_watermark/finish/destroy all flush — poll opportunistically, the
 * rest unconditionally). Emitted batches and the local watermark may
 * therefore lag one device-pushed batch; results are identical. */
dz_status dz_window_op_push_device(dz_window_op* op, int64_t n_rows,
                                   const int64_t* d_ts_ms,
                                   const int32_t* d_key_ids,
                                   const double* d_vals);

/* Zero-copy variant of dz_window_op_push_device: the op reads the caller's
 * device buffers directly instead of staging a copy, so they must remain
 * valid AND unmodified until the NEXT call into the op (when the deferred
 * phase consumes them). Use when the caller owns a resident stream buffer
 * (the bench; an FFI caller holding the RecordBatch across its poll-loop
 * iteration); use the staging variant above when buffer lifetime past the
 * call cannot be guaranteed. */
dz_status dz_window_op_push_device_borrowed(dz_window_op* op, int64_t n_rows,
                                            const int64_t* d_ts_ms,
                                            const int32_t* d_key_ids,
                                            const double* d_vals);

/* Device-resident push with RAW UTF8 KEYS: the per-row string interning the
 * reference pays inside GroupValues::intern (grouped_window_agg_stream.rs:512)
 * runs ON DEVICE (open-address fingerprint table + device string pool), so
 * key materialization is inside the measured path. Borrowed semantics: the
 * offsets/data/ts/vals buffers stay valid and unmodified until the NEXT call
 * into the op. Requires key_kind DZ_KEY_UTF8. Key capacity is sized from
 * n_keys_hint at create (table 4x hint, pool 64 B/key average) and growth
 * past it fails loudly — size the hint for the stream's cardinality. */
dz_status dz_window_op_push_device_utf8(dz_window_op* op, int64_t n_rows,
                                        const int64_t* d_ts_ms,
                                        const int32_t* d_key_offsets,
                                        const char* d_key_data,
                                        const double* d_vals);

/* Retrieve emitted closed windows (the stream's output RecordBatch,
 * trigger_windows :220-253). *out = NULL when nothing is pending.
 * The returned batch stays valid until the next poll/destroy. */
dz_status dz_window_op_poll(dz_window_op* op, const dz_out_batch** out);

/* Close every remaining open window (extension for finite streams; the
 * reference stream is unbounded and only closes on watermark advance). */
dz_status dz_window_op_finish(dz_window_op* op);

/* Emission is pipelined off the push path (a worker thread builds output
 * batches while later pushes compute). Blocks until every window already
 * triggered is available to poll — the synchronous-poll behaviour of the
 * reference stream when a consumer needs it. finish() implies drain. */
dz_status dz_window_op_drain(dz_window_op* op);

void dz_window_op_destroy(dz_window_op* op);

/* DataFusionError-by-value analog (crates/common/src/error/mod.rs:13-44). */
const char* dz_last_error(dz_window_op* op);

/* ------------------------------------------------------------------ */
/* Watermark sharing (the Arc<Mutex<watermark>> shared across partitions:
 * streaming_window.rs:210, grouped_window_agg_stream.rs:255-266). In the
 * multi-GPU plan each shard merges the global watermark (e.g. an RCCL/gloo
 * all-reduce MAX) and injects it here before polling. */
dz_status dz_window_op_advance_watermark(dz_window_op* op, int64_t wm_ms);
int64_t dz_window_op_watermark(dz_window_op* op);   /* INT64_MIN if unset */
int64_t dz_window_op_open_windows(dz_window_op* op);

/* ------------------------------------------------------------------ */
/* Filter pushdown: the pipeline's `.filter(col("max") > lit(113))`
 * (datastream.rs:94-105) evaluates over the window's output; pushing it
 * into the operator's emission keeps the boundary drop-in (output batches
 * are already filtered). cmp: 0 '<', 1 '<=', 2 '>', 3 '>=', 4 '==', 5 '!='.
 * agg_idx indexes dz_window_desc.aggs. NULL aggregate values never pass. */
dz_status dz_window_op_set_filter(dz_window_op* op, int32_t agg_idx,
                                  int32_t cmp, double literal);

/* ------------------------------------------------------------------ */
/* Stream join (BASELINE cfg5): inner equi-join on trip_id feeding the
 * windowed group-by. The reference lowers DataStream::join to DataFusion's
 * inner hash join (crates/core/src/datastream.rs:126-175; its only join
 * example composes two streams, examples/examples/stream_join.rs). The
 * streaming discipline (restated in oracle/oracle.c orc_join_*): build side
 * = (trip_id -> driver_id) dimension table (later duplicates overwrite;
 * the winner among duplicates WITHIN one build batch is unspecified); each
 * probe batch emits its matching rows IN ROW ORDER; unmatched rows buffer
 * in row order and re-emit (original order) when their build row arrives;
 * rows never matched are dropped (inner join). Outputs are device-resident
 * (ts, driver id as a dense int32 key, value) columns shaped for a
 * zero-copy dz_window_op_push_device_borrowed — driver ids must fit int32.
 * Build-table capacity is sized from n_trips_hint (4x, pow2) and overflow
 * fails loudly. */

typedef struct dz_join_op dz_join_op;

dz_join_op* dz_join_op_create(int32_t device, int64_t n_trips_hint);
void dz_join_op_destroy(dz_join_op* op);
const char* dz_join_last_error(dz_join_op* op);

/* Device-resident build push (trip_id -> driver_id rows). Also re-probes
 * the unmatched buffer: newly matched rows become this call's matches. */
dz_status dz_join_op_push_build(dz_join_op* op, int64_t n_rows,
                                const int64_t* d_trip_ids,
                                const int64_t* d_driver_ids);

/* Device-resident probe push (event rows). Matching rows become this
 * call's matches; the rest join the unmatched buffer. */
dz_status dz_join_op_push_probe(dz_join_op* op, int64_t n_rows,
                                const int64_t* d_ts_ms,
                                const int64_t* d_trip_ids,
                                const double* d_vals);

/* The LAST push's matched rows (device pointers, valid until the
 * second-next push on this op — double-buffered so a window op's borrowed
 * push can consume them across one pipeline step). */
dz_status dz_join_op_matches(dz_join_op* op, int64_t* n_out,
                             const int64_t** d_ts_ms, const int32_t** d_kids,
                             const double** d_vals);
int64_t dz_join_op_unmatched(dz_join_op* op);

/* ------------------------------------------------------------------ */
/* JSON ingest (from_topic's decode stage): the reference decodes Kafka
 * payload bytes on the host with serde_json
 * (crates/core/src/formats/decoders/json.rs:23-46, driven by
 * kafka_stream_read.rs:165-296); here the decode runs ON DEVICE over
 * newline-delimited records, producing (int64 ts, utf8 key column, f64
 * value) device columns shaped for dz_window_op_push_device_utf8 — the
 * on-wire-bytes -> windowed-aggregate pipeline stays in HBM end to end.
 * Documented subset (anything else fails loudly): records are non-empty
 * '\n'-separated JSON objects; the three schema fields are top-level
 * (other fields, including nested objects/arrays, are skipped); no escape
 * sequences inside schema field names or the key string; numeric literals
 * within the exact Clinger fast path (<= 15 significant digits, |decimal
 * exponent| <= 22 — equal to strtod bit-for-bit there). */

typedef struct dz_json_decoder dz_json_decoder;

dz_json_decoder* dz_json_decoder_create(int32_t device, const char* ts_field,
                                        const char* key_field,
                                        const char* val_field);
void dz_json_decoder_destroy(dz_json_decoder* d);
const char* dz_json_decoder_last_error(dz_json_decoder* d);

/* Decode one device-resident byte buffer. Columns are retrieved with
 * dz_json_decoder_batch and stay valid until the SECOND-next decode
 * (double-buffered, covering a utf8 borrowed push across one step —
 * provided batch sizes do not grow, which reallocates). */
dz_status dz_json_decode(dz_json_decoder* d, const char* d_bytes,
                         int64_t n_bytes);
dz_status dz_json_decoder_batch(dz_json_decoder* d, int64_t* n_out,
                                const int64_t** d_ts_ms,
                                const int32_t** d_key_offsets,
                                const char** d_key_data,
                                const double** d_vals);

/* Synthetic on-wire JSON generator (bench/test input): one newline-
 * delimited record per row of the same seeded sensor stream, reading
 * printed as fixed 6-decimal (inside the decoder's exact subset).
 * Lens pass (host cumsums into int64 offsets) then fill pass. */
dz_status dz_generate_json(int32_t device, uint64_t seed, int64_t t0_ms,
                           int64_t start_row, int64_t n_rows, int64_t n_keys,
                           int64_t rows_per_ms, int32_t* d_lens,
                           const int64_t* d_offsets, char* d_data);

/* ------------------------------------------------------------------ */
/* Synthetic sensor stream generator, on device (bench/test input; spec in
 * DESIGN.md §Generator, bit-identical to oracle orc_gen; mirrors
 * examples/examples/emit_measurements.rs:30-67). Any output pointer may be
 * NULL to skip that column. d_key_ids is the dense int32 form of d_keys. */
dz_status dz_generate(int32_t device, uint64_t seed, int64_t t0_ms,
                      int64_t start_row, int64_t n_rows, int64_t n_keys,
                      int64_t rows_per_ms, int64_t* d_ts_ms,
                      int64_t* d_keys, int32_t* d_key_ids, double* d_vals);

/* Synthetic utf8 key column generator ("sensor_{k}", k from the same seeded
 * draw as dz_generate): pass 1 writes per-row byte lengths to d_lens (host
 * cumsums them into offsets — int32 offsets bound one batch to 2 GiB of key
 * bytes); pass 2 fills d_data at d_offsets. Either pointer pair may be NULL
 * to run one pass. */
dz_status dz_generate_utf8(int32_t device, uint64_t seed, int64_t start_row,
                           int64_t n_rows, int64_t n_keys, int32_t* d_lens,
                           const int32_t* d_offsets, char* d_key_data);

/* Device memory helpers for the bench path (thin wrappers so no HIP types
 * cross the boundary). */
dz_status dz_device_malloc(int32_t device, size_t bytes, void** out);
dz_status dz_device_free(void* ptr);
dz_status dz_device_synchronize(int32_t device);
dz_status dz_memcpy_d2h(void* dst_host, const void* src_dev, size_t bytes);
dz_status dz_memcpy_h2d(void* dst_dev, const void* src_host, size_t bytes);

/* ------------------------------------------------------------------ */
/* Kernel timing (HIP events recorded on the op's own stream; feeds the
 * bench's roofline line). total_ms covers all completed launches. */
typedef struct dz_kernel_stat {
    char name[32];
    uint64_t launches;
    double total_ms;
    double bytes_per_launch_alg;  /* algorithmic bytes of the LAST launch */
} dz_kernel_stat;

dz_status dz_window_op_kernel_stats(dz_window_op* op, dz_kernel_stat* out,
                                    int32_t cap, int32_t* n_out);

/* ------------------------------------------------------------------ */
/* Pure host logic exported for CPU-only tests (no GPU needed):
 * get_windows_for_watermark + snap_to_window_start
 * (streaming_window.rs:1053-1094; ms generalization for sub-second lengths
 * per SURVEY.md §7). slide_ms == 0 means tumbling. Returns the number of
 * ranges (writes up to cap). */
int64_t dz_debug_windows_for_range(int64_t min_ts, int64_t max_ts,
                                   int64_t len_ms, int64_t slide_ms,
                                   int64_t* starts, int64_t* ends, int64_t cap);

const char* dz_version(void);

#ifdef __cplusplus
}
#endif

#endif /* DENORMALIZED_AMD_H */
