/* window_op.cpp — MI355X-native engine behind the C ABI in
 * include/denormalized_amd.h. Host-side mirror of the reference's
 * GroupedWindowAggStream control flow (window frames, watermark, trigger:
 * crates/core/src/physical_plan/continuous/grouped_window_agg_stream.rs) with
 * all per-row work on the GPU (kernels.hip). The GPU is REQUIRED: every
 * compute entry point fails loudly when no HIP device is present — there is
 * no CPU fallback in the product path.
 */
#include "dz_internal.h"
#include "denormalized_amd.h"

#include <algorithm>
#include <cstdio>
#include <cstring>
#include <deque>
#include <map>
#include <string>
#include <unordered_map>
#include <atomic>
#include <condition_variable>
#include <memory>
#include <mutex>
#include <thread>
#include <vector>

namespace {

std::string g_err; /* create-time errors (no handle yet) */

int64_t snap_ms_host(int64_t ts, int64_t len_ms) {
    int64_t len_s = len_ms / 1000;
    if (len_s == 0) return ts - (ts % len_ms);
    return (ts / 1000) / len_s * len_s * 1000;
}

} // namespace

/* get_windows_for_watermark restated for the host (streaming_window.rs:
 * 1053-1086; ms generalization for sub-second lengths per SURVEY §7). */
extern "C" int64_t dz_debug_windows_for_range(int64_t min_ts, int64_t max_ts,
                                              int64_t len_ms, int64_t slide_ms,
                                              int64_t* starts, int64_t* ends,
                                              int64_t cap) {
    int64_t n = 0;
    if (slide_ms > 0) {
        int64_t cur = snap_ms_host(min_ts - len_ms, len_ms);
        while (cur <= max_ts) {
            int64_t end = cur + len_ms;
            if (!(min_ts > end || max_ts < cur)) {
                if (n < cap) { starts[n] = cur; ends[n] = end; }
                n++;
            }
            cur += slide_ms;
        }
    } else {
        int64_t cur = snap_ms_host(min_ts, len_ms);
        while (cur <= max_ts) {
            if (n < cap) { starts[n] = cur; ends[n] = cur + len_ms; }
            n++;
            cur += len_ms;
        }
    }
    return n;
}

extern "C" const char* dz_version(void) { return "denormalized-amd 0.1 (gfx950)"; }

/* ------------------------------------------------------------------ */

struct PendingEvent {
    std::string name;
    hipEvent_t a, b;
    double bytes;
};

/* Append-only dictionary store shared between the push thread (appends) and
 * the emission workers (read indices below their job's n_keys snapshot).
 * A deque here was an unsynchronized race: push_back mutates the deque's
 * internal block map, which a concurrent operator[] walks. This store never
 * mutates published state: chunks are fixed arrays whose pointers publish
 * once (release store), elements are written before the job hand-off (the
 * e_mtx pair gives the happens-before), and appends to a chunk touch only
 * indices no worker's snapshot covers. */
template <typename T>
struct ChunkedDict {
    static constexpr int CHUNK_LOG = 12; /* 4096 entries per chunk */
    static constexpr size_t CHUNK = 1u << CHUNK_LOG;
    static constexpr size_t MAX_CHUNKS = 8192; /* 33M entries = the key cap */
    std::atomic<T*> chunks[MAX_CHUNKS] = {};
    size_t n = 0; /* push thread only; workers use their job snapshot */
    void push_back(T v) {
        size_t c = n >> CHUNK_LOG;
        T* p = chunks[c].load(std::memory_order_relaxed);
        if (!p) {
            p = new T[CHUNK];
            chunks[c].store(p, std::memory_order_release);
        }
        p[n & (CHUNK - 1)] = std::move(v);
        n++;
    }
    const T& operator[](size_t i) const {
        return chunks[i >> CHUNK_LOG].load(std::memory_order_acquire)
                     [i & (CHUNK - 1)];
    }
    size_t size() const { return n; }
    ~ChunkedDict() {
        for (size_t c = 0; c <= (n >> CHUNK_LOG) && c < MAX_CHUNKS; c++)
            delete[] chunks[c].load(std::memory_order_relaxed);
    }
};

struct KStatAcc {
    uint64_t launches = 0;
    uint64_t timed = 0;   /* launches that carried HIP timing events */
    double ms = 0.0;      /* summed over the TIMED launches only */
    double last_bytes = 0.0;
};

struct GroupCtl {
    /* sibling coordination for one group-batched emission chain */
    std::atomic<int> claimed{0};
    std::atomic<bool> ready{false};
    uint32_t offs[17] = {};
    uint32_t gtot = 0;
    int span = -1;              /* pinned span index, or -1 (heap fallback) */
    char* heap = nullptr;       /* liveness fallback when no span was free */
    std::atomic<int> holders{0};   /* zero-copy slice OutBufs alive */
    std::atomic<bool> all_built{false};
};

struct OutBuf {
    /* zero-copy device-path emission: the batch's aggregate columns (and
     * the dense key column) are served as views into the pinned packed
     * span; the OutBuf holds the slab until the consumer moves to the next
     * batch (poll pop releases it). hold_slab == -1 => fully materialized. */
    const char* packed = nullptr;
    uint32_t packed_nt = 0;   /* column-section stride of the packed span */
    uint32_t packed_off = 0;  /* this batch's slice offset (group spans) */
    int hold_slab = -1;
    uint32_t slab_gen = 0;  /* pool generation the held slab belongs to */
    std::shared_ptr<GroupCtl> hold_group; /* group span held until poll
                                           * advances past this batch */
    std::vector<int64_t> key_i64;
    std::vector<int32_t> key_offsets;
    std::vector<char> key_data;
    std::vector<std::vector<int64_t>> agg_i64; /* per agg col (COUNT) */
    std::vector<std::vector<double>> agg_f64;  /* per agg col (others) */
    std::vector<uint8_t> agg_valid;
    std::vector<int64_t> wstart, wend;
    /* view plumbing */
    std::vector<const void*> agg_ptrs;
    dz_out_batch view;
};

struct dz_window_op {
    /* config */
    dz_window_type wtype;
    int64_t len_ms, slide_ms;
    int32_t ts_col, group_col;
    bool no_group = false; /* global aggregate (no GROUP BY): Partial/Final
                            * path of the reference, streaming_window.rs:640-1051;
                            * one group (dense key 0) per window */
    dz_key_kind key_kind;
    std::vector<dz_agg_desc> aggs;
    int device = 0;
    int32_t max_open = 64;
    std::string err;

    hipStream_t stream = nullptr;
    /* emission runs off the compute stream; the per-close device chains
     * (slabread -> sort -> gather) of different closes are independent, so
     * they round-robin over E_CSTREAMS streams (each with its own radix
     * scratch) instead of serializing on one */
    static constexpr int E_CSTREAMS = 4;
    hipStream_t copy_stream = nullptr;   /* == c_streams[0] */
    hipStream_t c_streams[E_CSTREAMS] = {};

    /* dictionary (host side; first-seen insertion order == emitted id order
     * is NOT assumed — emission sorts by first row; the dict only maps
     * key values <-> dense ids). ChunkedDict: append-only, safe for the
     * emission workers to read below their job's n_keys snapshot while the
     * push thread appends. */
    std::unordered_map<std::string, int32_t> dict_utf8;
    ChunkedDict<std::string> dict_strs;
    std::unordered_map<int64_t, int32_t> dict_i64;
    ChunkedDict<int64_t> dict_vals;
    int64_t n_keys = 0; /* dense key count seen so far */

    /* persistent group state: one slab per slot, 5 contiguous fields of kcap
     * 8-byte entries each ([slot][{cnt,first,min,max,sum}][kcap]) so a window
     * close is ONE D2H copy. Field base pointers stride 5*kcap per slot. */
    int64_t kcap = 0; /* multiple of NB */
    int32_t nslots = 0;
    uint64_t* s_base = nullptr; /* the slab allocation */
    uint64_t* s_cnt = nullptr;
    uint64_t* s_first = nullptr;
    double* s_min = nullptr;
    double* s_max = nullptr;
    double* s_sum = nullptr;
    struct FreeSlot {
        int32_t slot;
        hipEvent_t ev;     /* copy-done gate or null */
        bool ev_owned = true; /* false: a rotating frontier event — wait on
                               * reuse but never return it to the pool */
    };
    std::vector<FreeSlot> free_slots;
    struct OpenWin { int64_t end; int32_t slot;
                     uint64_t row_base; /* rows_seen at open: emission sort
                                         * keys are first - row_base */ };
    std::map<int64_t, OpenWin> open; /* by window start (BTreeMap order) */

    int64_t watermark = INT64_MIN;
    bool has_wm = false;
    uint32_t batch_seq = 0;
    uint64_t rows_seen = 0;  /* stream-lifetime row counter (first-seen ids) */

    /* scratch (d_ghist/d_scalars double-buffered: the deferred device-push
     * pipeline runs batch N's reduction on i_stream while batch N-1's scan
     * still reads the other buffer on the compute stream) */
    uint32_t* d_ghist[2] = {};
    uint32_t* d_gofs[2] = {};
    uint32_t* d_total[2] = {};
    uint32_t* d_base[2] = {};
    int C_cap = 0;
    uint4* d_grec = nullptr;  /* 16 B {val, rowidx, meta} records (scatter out) */
    uint32_t* d_b1offs = nullptr; /* two-level L1 segment layout [NB][256] */
    uint32_t* d_b1lens = nullptr;
    uint4* d_grec2 = nullptr;     /* two-level intermediate records */
    int64_t l2_cap = 0;
    int64_t rec_cap = 0;
    uint64_t* d_scalars[2] = {};
    uint64_t* h_scalars = nullptr; /* pinned, 2 x 3 (per pipeline buffer) */
    int32_t* d_slotmap = nullptr;   /* MAX_RANGES, allocated at create */
    int32_t* h_slotmap[2] = {};     /* pinned upload staging, per pipeline
                                     * buffer (host reuse must not outrun
                                     * the async H2D) */
    int32_t* d_zero_kid = nullptr; /* no_group: all rows in group 0 */
    int64_t zero_cap = 0;
    int32_t* d_resetlist = nullptr; /* MAX_RANGES, allocated at create */
    int32_t* h_resetlist[2] = {};
    uint32_t* d_dbg = nullptr; /* kernel bounds-guard cells (4 u32, zeroed) */
    bool dbg_checked_err = false;

    /* device utf8 intern (GroupValues::intern at device rate). Fixed-capacity
     * from n_keys_hint at create; capacity overflow flags d_dbg[3]. */
    uint4* d_itab = nullptr;        /* 16 B slots {fp, id, len|off<<6} */
    uint32_t* d_itab_row = nullptr; /* claiming row, valid within one claim */
    uint32_t i_pmask = 0;
    uint32_t* d_ioff = nullptr;     /* per-id {pool offset, byte length} */
    uint32_t* d_ilen = nullptr;
    char* d_ipool = nullptr;        /* first-seen key bytes */
    uint32_t* d_ictrs = nullptr;    /* [0]=next id, [1]=pool cursor */
    uint32_t i_idcap = 0, i_poolcap = 0;
    int32_t* d_ikid[2] = {};        /* interned dense ids, per pipeline buf */
    int64_t i_kid_cap[2] = {0, 0};
    int64_t mirror_keys = 0;        /* ids mirrored into dict_strs so far */

    /* input staging (host-batch path) */
    int64_t* d_ts = nullptr;
    int32_t* d_kid = nullptr;
    double* d_vals = nullptr;
    uint8_t* d_valbm = nullptr;
    int64_t in_cap = 0;
    char* h_stage = nullptr; /* pinned */
    size_t h_stage_cap = 0;

    /* deferred device-push pipeline: push_device enqueues input staging plus
     * the min/max(/hist) reduction on i_stream and returns; the window
     * decisions and partition/fold launches run at the NEXT call into the op
     * (or at a poll once the reduction is host-visible), so the host's
     * decision wait overlaps the previous step's device execution instead of
     * serializing behind it. */
    hipStream_t i_stream = nullptr;
    struct Pend {
        bool active = false, sliding = false, dense = false, deferred = false;
        int buf = 0;
        int C = 0;
        int64_t n = 0, chunk = 0;
        const int64_t* ts = nullptr;
        const int32_t* kid = nullptr;
        const double* vals = nullptr;
        const uint8_t* valbm = nullptr;
    } pend;
    int next_buf = 0;
    int64_t* d_sts[2] = {};   /* op-owned device input ring (device-push) */
    int32_t* d_skid[2] = {};
    double* d_svals[2] = {};
    int64_t s_in_cap[2] = {0, 0};
    hipEvent_t ev_ready[2] = {};    /* scalars D2H host-visible */
    hipEvent_t ev_staged[2] = {};   /* caller inputs copied into the ring */
    hipEvent_t ev_consumed[2] = {}; /* scatter that read buffer b completed */
    bool consumed_valid[2] = {false, false};

    /* async DEVICE-SIDE emission: at window close the copy stream runs
     * compact -> stable radix sort by first-seen row -> gather+filter
     * (kernels.hip launch_emission); a worker thread then copies exactly
     * nt sorted rows D2H and formats them off the push critical path. */
    /* 32 slabs: a slab is held from enqueue until the worker's host build
     * completes (~fold latency + build); with the deferred push pipeline the
     * push thread no longer idles in a stream sync each step, so trigger
     * bursts (≈8 closes/step at cfg2) need 2-3 steps of slack or the enqueue
     * path sleeps on e_cv waiting for a free slab. */
    static constexpr int E_POOL = 32;
    static constexpr int E_WORKERS = 4;
    struct DevEmit { /* per-slot device scratch, carved from one alloc */
        char* base = nullptr;
        uint64_t* ekeys; uint32_t* ekid;      /* touched groups (compact) */
        uint64_t* fkeys; uint32_t* fkid;      /* filter passers (packed)  */
        uint32_t* fiota;                      /* sort payload / sidx out  */
        uint64_t* skeys; uint32_t* okid;      /* sort ping-pong buffers   */
        uint32_t* counter;                    /* [0]=touched, [1]=passers */
        uint64_t* ocnt;
        double* omin; double* omax; double* osum; double* oavg;
        uint8_t* oflags;
        char* pout;  /* permuted, packed final-order columns (53 B/entry) */
    };
    DevEmit e_dev[E_POOL];
    uint64_t* e_slabs[E_POOL] = {}; /* pinned output staging per slot */
    uint32_t* e_pcnt = nullptr;     /* pinned per-slot nt counters */
    uint32_t* d_rhist[E_CSTREAMS] = {}; /* radix scratch, one per c_stream */
    uint32_t* d_roffs[E_CSTREAMS] = {};
    int64_t e_slab_kcap = 0;
    struct EmitJob {
        hipEvent_t ev;
        int slab;
        int64_t wstart, wend, n_keys, kcap;
        uint64_t ticket;
        bool device; /* device-sorted columns vs raw host slab */
        int cs = 0;  /* c_streams index this close's device chain runs on */
        /* host-path closes of one trigger group share ONE event (their slab
         * reads ride one gather + one D2H); the last sibling to finish
         * returns the event and the group buffer to their pools */
        std::shared_ptr<std::atomic<int>> grp_left;
        int gbuf = -1;  /* pinned group buffer index, or -1 (own slab) */
        int goff = 0;   /* this close's slab offset within the group buffer */
        bool grouped = false;      /* group-batched device emission */
        std::shared_ptr<GroupCtl> ctl;
    };
    /* pinned group buffers for batched host-path emission (gather kernel
     * packs up to EGROUP slots contiguously on device, one D2H lands here) */
    static constexpr int E_GBUFS = 4;
    uint64_t* e_gbufs[E_GBUFS] = {};     /* pinned, host view */
    uint64_t* e_gbufs_dev[E_GBUFS] = {}; /* device view of the same memory:
                                          * the gather kernel writes output
                                          * straight over the host link — a
                                          * hipMemcpyAsync D2H here performs
                                          * the transfer at ENQUEUE time on
                                          * the calling thread (measured),
                                          * a kernel store does not */
    std::vector<int> e_gfree;            /* guarded by e_mtx */
    int64_t e_gbuf_kcap = 0;
    /* zero-copy OutBufs hold pinned slabs across a pool regrow (keyspace
     * growth): old allocations go to a graveyard until destroy, and a
     * generation counter stops a stale hold from re-entering the new pool */
    uint32_t e_slab_gen = 0;             /* guarded by e_mtx */
    std::vector<uint64_t*> e_graveyard;
    /* slot-release frontier events for group-batched host emission: each
     * group records the next event of this rotation after its gather; the
     * freed slots' reset waits reference it WITHOUT taking ownership (a
     * later re-record only pushes the observed instant later — safe).
     * Rotating avoids re-recording a still-pending event, which the runtime
     * serves with an internal wait. */
    static constexpr int E_FRONTIERS = 4;
    hipEvent_t e_frontier[E_FRONTIERS] = {};
    int e_frontier_idx = 0;
    std::deque<EmitJob> e_jobs;     /* guarded by e_mtx */
    std::vector<int> e_free;        /* free slab indices, guarded by e_mtx */
    std::vector<hipEvent_t> e_ev_pool; /* events for emission jobs (e_mtx) */
    std::mutex e_mtx;
    std::condition_variable e_cv;   /* job added / slab freed / drained */
    std::vector<std::thread> e_workers;
    bool e_stop = false;
    int e_inflight = 0;
    uint64_t e_ticket_next = 0;      /* next ticket to assign (e_mtx) */
    uint64_t e_ticket_pop = 0;       /* next ticket poll may emit (out_mtx) */
    std::map<uint64_t, OutBuf> e_done; /* completed out-of-order (out_mtx) */
    std::atomic<uint32_t> e_nt_hint{0}; /* last close's passer count:
                                          * adaptive emission-path choice */
    /* GROUP-BATCHED device emission: one chain serves a whole trigger
     * group of closes (the per-close chain's ~25 host enqueues were the
     * push-thread wall on many-small-close workloads — cfg3 sliding).
     * Worst-case sized: gcount is capped at E_GELEMS/kcap closes, so the
     * fused gather can never overflow. */
    static constexpr int64_t E_GELEMS = 8 << 20;
    struct GDev {
        char* base = nullptr;
        uint64_t* gkeys; uint64_t* gskeys;
        uint32_t* gkid; uint32_t* gokid; uint32_t* giota;
        uint64_t* gcnt_col;
        double* gmin; double* gmax; double* gsum; double* gavg;
        uint8_t* gflags;
        uint32_t* ctr;    /* [0]=total, [1..16]=per-close counts */
        uint32_t* rhist; uint32_t* roffs;
        char* pout;       /* packed 53 B/elem, close-major sorted */
    };
    static constexpr int E_GD = 4; /* groups in flight (one per
                                    * emission stream) */
    GDev e_gd[E_GD];
    char* e_gpin[E_GD] = {};      /* pinned packed span pool (free-listed;
                                   * held by zero-copy slice consumers) */
    std::vector<int> e_gspan_free; /* guarded by e_mtx */
    uint32_t* e_gpcnt = nullptr;  /* pinned, E_GD x 32 counters */
    std::vector<int> e_gdfree;    /* guarded by e_mtx */
    bool e_gd_ready = false;
    std::atomic<uint64_t> e_build_ns{0};
    std::atomic<uint64_t> e_builds{0};
    std::atomic<uint64_t> e_zc_builds{0};   /* zero-copy vs materializing */
    std::atomic<uint64_t> e_copy_builds{0}; /* device-path builds */
    std::mutex out_mtx;             /* guards outq */

    /* filter pushdown */
    bool has_filter = false;
    int32_t f_idx = 0, f_cmp = 0;
    double f_lit = 0.0;

    /* output */
    std::deque<OutBuf> outq;
    OutBuf current;
    bool has_current = false;
    /* consumed OutBuf husks recycled to the build workers: vector capacity
     * is retained, so ~1M-row closes stop paying allocation + page faults
     * on every build (the reason 8 workers ran SLOWER than 4) */
    std::vector<OutBuf> ob_pool;  /* guarded by out_mtx */

    /* timing */
    std::vector<hipEvent_t> ev_pool;
    std::deque<PendingEvent> pending;
    std::map<std::string, KStatAcc> stats;
};

#define CHK(op, call)                                                      \
    do {                                                                   \
        hipError_t e_ = (call);                                            \
        if (e_ != hipSuccess) {                                            \
            (op)->err = std::string(#call) + ": " + hipGetErrorString(e_); \
            return DZ_ERR;                                                 \
        }                                                                  \
    } while (0)

static void emit_worker_main(dz_window_op* op);
static dz_status ensure_emission(dz_window_op* op);
static dz_status process_pending(dz_window_op* op);
static dz_status intern_sync_mirror(dz_window_op* op);
static void build_emission_slice(dz_window_op* op, int64_t wstart,
                                 int64_t wend, uint32_t total, uint32_t off,
                                 uint32_t n, const char* p, OutBuf* out);
static void build_emission_slice_views(dz_window_op* op, int64_t wstart,
                                       int64_t wend, uint32_t n, OutBuf* out);

static hipEvent_t get_event(dz_window_op* op) {
    if (!op->ev_pool.empty()) {
        hipEvent_t e = op->ev_pool.back();
        op->ev_pool.pop_back();
        return e;
    }
    hipEvent_t e;
    hipEventCreate(&e);
    return e;
}

static void drain_events(dz_window_op* op, bool wait) {
    while (!op->pending.empty()) {
        PendingEvent& p = op->pending.front();
        if (!wait && hipEventQuery(p.b) != hipSuccess) break;
        if (wait) hipEventSynchronize(p.b);
        float ms = 0;
        hipEventElapsedTime(&ms, p.a, p.b);
        KStatAcc& s = op->stats[p.name];
        s.ms += ms;
        op->ev_pool.push_back(p.a);
        op->ev_pool.push_back(p.b);
        op->pending.pop_front();
    }
}

#include <chrono>
struct HostTimer {
    dz_window_op* op;
    const char* name;
    std::chrono::steady_clock::time_point t0;
    HostTimer(dz_window_op* o, const char* n)
        : op(o), name(n), t0(std::chrono::steady_clock::now()) {}
    ~HostTimer();
};

template <typename F>
static void timed_on(dz_window_op* op, hipStream_t s, const char* name,
                     double bytes, F&& fn) {
    KStatAcc& st = op->stats[name];
    st.launches++;
    st.last_bytes = bytes;
    /* sample the HIP-event timing (2 records per kernel per step cost
     * ~40 us/step of enqueue latency on the push thread); the export
     * scales ms back up by launches/timed */
    if ((st.launches & 7) != 1) {
        fn();
        return;
    }
    st.timed++;
    hipEvent_t a = get_event(op), b = get_event(op);
    hipEventRecord(a, s);
    fn();
    hipEventRecord(b, s);
    op->pending.push_back({name, a, b, bytes});
}

template <typename F>
static void timed(dz_window_op* op, const char* name, double bytes, F&& fn) {
    timed_on(op, op->stream, name, bytes, std::forward<F>(fn));
}

HostTimer::~HostTimer() {
    double ms = std::chrono::duration<double, std::milli>(
                    std::chrono::steady_clock::now() - t0).count();
    KStatAcc& s = op->stats[name];
    s.launches++;
    s.ms += ms;
}

/* ------------------------------------------------------------------ */
/* state allocation / growth                                           */
/* ------------------------------------------------------------------ */

static dz_status state_alloc(dz_window_op* op, int64_t kcap_new, int32_t nslots_new) {
    /* allocate fresh arrays, memset to empty, copy any open-slot regions */
    kcap_new = std::max<int64_t>(kcap_new, dz::NB);
    kcap_new = (kcap_new + dz::NB - 1) / dz::NB * dz::NB;
    if (kcap_new / dz::NB > 65535) {
        op->err = "key capacity exceeds 65535*NB (~33M) — not supported yet";
        return DZ_ERR;
    }
    if (kcap_new == op->kcap && nslots_new <= op->nslots) return DZ_OK;
    nslots_new = std::max(nslots_new, op->nslots);
    size_t cells = (size_t)kcap_new * nslots_new * 5;
    uint64_t* n_base;
    CHK(op, hipMalloc(&n_base, cells * 8));
    uint64_t* n_cnt = n_base;
    uint64_t* n_first = n_base + kcap_new;
    double* n_min = (double*)(n_base + 2 * kcap_new);
    double* n_max = (double*)(n_base + 3 * kcap_new);
    double* n_sum = (double*)(n_base + 4 * kcap_new);
    size_t stride_new = (size_t)kcap_new * 5, stride_old = (size_t)op->kcap * 5;
    for (int32_t s = 0; s < nslots_new; s++) {
        CHK(op, hipMemsetAsync(n_cnt + s * stride_new, 0, kcap_new * 8, op->stream));
        CHK(op, hipMemsetAsync(n_first + s * stride_new, 0xFF, kcap_new * 8, op->stream));
        /* min/max/sum need no init: fold ignores them while cnt == 0 */
    }
    for (auto& kv : op->open) { /* preserve open-window state */
        int32_t s = kv.second.slot;
        int64_t span = std::min(op->kcap, kcap_new);
        for (int f = 0; f < 5; f++)
            CHK(op, hipMemcpyAsync(n_base + s * stride_new + (size_t)f * kcap_new,
                                   op->s_base + s * stride_old + (size_t)f * op->kcap,
                                   span * 8, hipMemcpyDeviceToDevice, op->stream));
    }
    CHK(op, hipStreamSynchronize(op->stream));
    for (int s = 0; s < dz_window_op::E_CSTREAMS; s++) /* emission reads s_base */
        if (op->c_streams[s])
            CHK(op, hipStreamSynchronize(op->c_streams[s]));
    hipFree(op->s_base);
    op->s_base = n_base;
    op->s_cnt = n_cnt; op->s_first = n_first; op->s_min = n_min;
    op->s_max = n_max; op->s_sum = n_sum;
    for (int32_t s = op->nslots; s < nslots_new; s++)
        op->free_slots.push_back({s, nullptr});
    op->kcap = kcap_new;
    op->nslots = nslots_new;
    return DZ_OK;
}


/* ------------------------------------------------------------------ */
/* create / destroy                                                    */
/* ------------------------------------------------------------------ */

extern "C" dz_window_op* dz_window_op_create(const dz_window_desc* desc) {
    g_err.clear();
    if (!desc) { g_err = "null desc"; return nullptr; }
    if (desc->length_ms <= 0) { g_err = "length_ms must be > 0"; return nullptr; }
    if (desc->window_type == DZ_WINDOW_SLIDING && desc->slide_ms <= 0) {
        g_err = "sliding window needs slide_ms > 0";
        return nullptr;
    }
    if (desc->n_aggs <= 0 || !desc->aggs) { g_err = "need at least one aggregate"; return nullptr; }
    int ndev = 0;
    if (hipGetDeviceCount(&ndev) != hipSuccess || ndev <= desc->device) {
        g_err = "no HIP device available (this operator has no CPU fallback)";
        return nullptr;
    }
    auto* op = new dz_window_op();
    op->wtype = desc->window_type;
    op->len_ms = desc->length_ms;
    op->slide_ms = desc->window_type == DZ_WINDOW_SLIDING ? desc->slide_ms : 0;
    op->ts_col = desc->ts_col;
    op->group_col = desc->group_col;
    op->no_group = desc->group_col < 0;
    op->key_kind = op->no_group ? DZ_KEY_DENSE_INT64 : desc->key_kind;
    op->aggs.assign(desc->aggs, desc->aggs + desc->n_aggs);
    op->device = desc->device;
    op->max_open = desc->max_open_windows > 0 ? desc->max_open_windows : 4096;
    bool cs_ok = true;
    if (hipSetDevice(op->device) != hipSuccess ||
        hipStreamCreate(&op->stream) != hipSuccess ||
        hipStreamCreate(&op->copy_stream) != hipSuccess) {
        g_err = "hip device/stream init failed";
        delete op;
        return nullptr;
    }
    op->c_streams[0] = op->copy_stream;
    for (int i = 1; i < dz_window_op::E_CSTREAMS; i++)
        cs_ok = cs_ok && hipStreamCreate(&op->c_streams[i]) == hipSuccess;
    if (!cs_ok) { /* degrade to one emission stream */
        for (int i = 1; i < dz_window_op::E_CSTREAMS; i++)
            op->c_streams[i] = op->copy_stream;
    }
    hipMalloc(&op->d_scalars[0], 3 * 8);
    hipMalloc(&op->d_scalars[1], 3 * 8);
    hipHostMalloc((void**)&op->h_scalars, 2 * 3 * 8);
    hipMalloc(&op->d_slotmap, (size_t)dz::MAX_RANGES * 4);
    hipMalloc(&op->d_resetlist, (size_t)dz::MAX_RANGES * 4);
    hipMalloc(&op->d_dbg, 16);
    hipMemset(op->d_dbg, 0, 16);
    for (int i = 0; i < 2; i++) {
        hipHostMalloc((void**)&op->h_slotmap[i], (size_t)dz::MAX_RANGES * 4);
        hipHostMalloc((void**)&op->h_resetlist[i], (size_t)dz::MAX_RANGES * 4);
    }
    {
        /* high-priority ingest stream: the next batch's reduction should
         * fill gaps ahead of the current batch's queued partition/fold, not
         * drain after it — its result gates the host's next decisions */
        int least = 0, greatest = 0;
        hipDeviceGetStreamPriorityRange(&least, &greatest);
        hipStreamCreateWithPriority(&op->i_stream, hipStreamNonBlocking,
                                    greatest);
    }
    for (int i = 0; i < 2; i++) {
        hipEventCreateWithFlags(&op->ev_ready[i], hipEventDisableTiming);
        hipEventCreateWithFlags(&op->ev_staged[i], hipEventDisableTiming);
        hipEventCreateWithFlags(&op->ev_consumed[i], hipEventDisableTiming);
    }
    int64_t hint = std::max<int64_t>(desc->n_keys_hint, 1);
    if (state_alloc(op, hint, 4) != DZ_OK) {
        g_err = op->err;
        hipStreamDestroy(op->stream);
        hipStreamDestroy(op->copy_stream);
        delete op;
        return nullptr;
    }
    if (ensure_emission(op) != DZ_OK) {
        g_err = op->err;
        /* continue: trigger retries; create-time prealloc is a fast-path */
        op->err.clear();
    }
    for (int i = 0; i < dz_window_op::E_WORKERS; i++)
        op->e_workers.emplace_back(emit_worker_main, op);
    return op;
}

extern "C" void dz_window_op_destroy(dz_window_op* op) {
    if (!op) return;
    hipSetDevice(op->device);
    process_pending(op); /* flush any deferred device push (best effort) */
    hipStreamSynchronize(op->stream);
    if (op->i_stream) hipStreamSynchronize(op->i_stream);
    {
        std::lock_guard<std::mutex> lk(op->e_mtx);
        op->e_stop = true;
    }
    op->e_cv.notify_all();
    for (auto& w : op->e_workers)
        if (w.joinable()) w.join();
    drain_events(op, true);
    for (auto e : op->ev_pool) hipEventDestroy(e);
    for (auto e : op->e_ev_pool) hipEventDestroy(e);
    for (auto& s : op->e_slabs)
        if (s) hipHostFree(s);
    for (auto* g : op->e_graveyard)
        hipHostFree(g);
    for (auto& g : op->e_gbufs)
        if (g) hipHostFree(g);
    for (auto& d : op->e_dev)
        if (d.base) hipFree(d.base);
    for (auto& g : op->e_gd)
        if (g.base) hipFree(g.base);
    for (auto& g : op->e_gpin)
        if (g) hipHostFree(g);
    if (op->e_gpcnt) hipHostFree(op->e_gpcnt);
    if (op->e_pcnt) hipHostFree(op->e_pcnt);
    for (int s = 0; s < dz_window_op::E_CSTREAMS; s++) {
        hipFree(op->d_rhist[s]);
        hipFree(op->d_roffs[s]);
    }
    hipFree(op->s_base);
    hipFree(op->d_ghist[0]); hipFree(op->d_ghist[1]);
    hipFree(op->d_gofs[0]); hipFree(op->d_gofs[1]);
    hipFree(op->d_total[0]); hipFree(op->d_total[1]);
    hipFree(op->d_base[0]); hipFree(op->d_base[1]);
    hipFree(op->d_grec);
    hipFree(op->d_b1offs); hipFree(op->d_b1lens);
    hipFree(op->d_grec2);
    hipFree(op->d_scalars[0]); hipFree(op->d_scalars[1]);
    hipFree(op->d_slotmap); hipFree(op->d_zero_kid);
    hipFree(op->d_resetlist); hipFree(op->d_dbg);
    hipFree(op->d_itab); hipFree(op->d_itab_row);
    hipFree(op->d_ioff); hipFree(op->d_ilen);
    hipFree(op->d_ipool); hipFree(op->d_ictrs);
    hipFree(op->d_ikid[0]); hipFree(op->d_ikid[1]);
    hipFree(op->d_ts); hipFree(op->d_kid); hipFree(op->d_vals); hipFree(op->d_valbm);
    for (int i = 0; i < 2; i++) {
        hipFree(op->d_sts[i]); hipFree(op->d_skid[i]); hipFree(op->d_svals[i]);
        if (op->ev_ready[i]) hipEventDestroy(op->ev_ready[i]);
        if (op->ev_staged[i]) hipEventDestroy(op->ev_staged[i]);
        if (op->ev_consumed[i]) hipEventDestroy(op->ev_consumed[i]);
    }
    if (op->h_scalars) hipHostFree(op->h_scalars);
    for (int i = 0; i < 2; i++) {
        if (op->h_slotmap[i]) hipHostFree(op->h_slotmap[i]);
        if (op->h_resetlist[i]) hipHostFree(op->h_resetlist[i]);
    }
    if (op->h_stage) hipHostFree(op->h_stage);
    hipStreamDestroy(op->stream);
    hipStreamDestroy(op->copy_stream);
    for (int s = 1; s < dz_window_op::E_CSTREAMS; s++)
        if (op->c_streams[s] && op->c_streams[s] != op->copy_stream)
            hipStreamDestroy(op->c_streams[s]);
    if (op->i_stream) hipStreamDestroy(op->i_stream);
    for (auto& fs : op->free_slots)
        if (fs.ev) hipEventDestroy(fs.ev);
    for (auto e : op->e_frontier)
        if (e) hipEventDestroy(e);
    delete op;
}

extern "C" const char* dz_last_error(dz_window_op* op) {
    if (!op) return g_err.empty() ? nullptr : g_err.c_str();
    return op->err.empty() ? nullptr : op->err.c_str();
}

/* ------------------------------------------------------------------ */
/* emission (trigger_windows, grouped_window_agg_stream.rs:220-253)    */
/* ------------------------------------------------------------------ */


static bool filter_pass(dz_window_op* op, int64_t row_cnt,
                        double vmin, double vmax, double vsum, bool valid) {
    if (!op->has_filter) return true;
    double v;
    dz_agg_op o = op->aggs[op->f_idx].op;
    if (o == DZ_AGG_COUNT) {
        v = (double)row_cnt;
    } else {
        if (!valid) return false; /* NULL never passes a comparison filter */
        switch (o) {
            case DZ_AGG_MIN: v = vmin; break;
            case DZ_AGG_MAX: v = vmax; break;
            case DZ_AGG_SUM: v = vsum; break;
            case DZ_AGG_AVG: v = vsum / (double)row_cnt; break;
            default: v = 0; break;
        }
    }
    switch (op->f_cmp) {
        case 0: return v < op->f_lit;
        case 1: return v <= op->f_lit;
        case 2: return v > op->f_lit;
        case 3: return v >= op->f_lit;
        case 4: return v == op->f_lit;
        case 5: return v != op->f_lit;
        default: return true;
    }
}

/* Build one emitted batch from a pinned copy of a slot slab
 * ([cnt][first][min][max][sum], each `kcap` 8-byte entries). Runs on the
 * emission worker thread: touches only immutable config, the deque-backed
 * dictionaries (indices < the job's n_keys snapshot) and the slab. */
static void build_emission_host(dz_window_op* op, int64_t wstart, int64_t wend,
                           int64_t K, int64_t kcap, const uint64_t* slab,
                           OutBuf* out) {
    const uint64_t* f_cnt = slab;
    const uint64_t* f_first = slab + kcap;
    const double* f_min = (const double*)(slab + 2 * kcap);
    const double* f_max = (const double*)(slab + 3 * kcap);
    const double* f_sum = (const double*)(slab + 4 * kcap);

    /* groups in first-seen (insertion) order: GroupValues emits insertion
     * order; sort touched keys by first-row sequence (first values are
     * distinct rows, so the order is exact). LSD radix (4 x 16-bit passes,
     * passes over all-equal digits skipped) — ~4x faster than std::sort at
     * the 10k-group scale this runs at per window close. */
    std::vector<std::pair<uint64_t, int32_t>> touched, scratch;
    touched.reserve(4096);
    for (int64_t k = 0; k < K; k++)
        if (f_first[k] != ~0ULL) touched.emplace_back(f_first[k], (int32_t)k);
    if (touched.size() > 1) {
        scratch.resize(touched.size());
        uint32_t hist[2048];
        for (int pass = 0; pass < 6; pass++) {
            int sh = pass * 11;
            memset(hist, 0, sizeof(hist));
            for (auto& p : touched) hist[(p.first >> sh) & 0x7FF]++;
            uint64_t d0 = (touched[0].first >> sh) & 0x7FF;
            if (hist[d0] == touched.size()) continue; /* all-equal digit */
            uint32_t run = 0;
            for (int d = 0; d < 2048; d++) { uint32_t t = hist[d]; hist[d] = run; run += t; }
            for (auto& p : touched) scratch[hist[(p.first >> sh) & 0x7FF]++] = p;
            touched.swap(scratch);
        }
    }

    /* filter pushdown (datastream.rs:94-105) — keep list */
    std::vector<int32_t> rows;
    rows.reserve(touched.size());
    for (auto& p : touched) {
        int32_t k = p.second;
        int64_t cnt = (int64_t)f_cnt[k];
        if (filter_pass(op, cnt, f_min[k], f_max[k], f_sum[k], cnt > 0))
            rows.push_back(k);
    }
    size_t n = rows.size(), na = op->aggs.size();

    OutBuf& ob = *out;
    ob.agg_i64.resize(na);
    ob.agg_f64.resize(na);
    if (op->key_kind == DZ_KEY_UTF8) {
        ob.key_offsets.resize(n + 1);
        ob.key_offsets[0] = 0;
        size_t total = 0;
        for (size_t i = 0; i < n; i++) total += op->dict_strs[rows[i]].size();
        ob.key_data.resize(total);
        size_t pos = 0;
        for (size_t i = 0; i < n; i++) {
            const std::string& s = op->dict_strs[rows[i]];
            memcpy(ob.key_data.data() + pos, s.data(), s.size());
            pos += s.size();
            ob.key_offsets[i + 1] = (int32_t)pos;
        }
    } else if (op->key_kind == DZ_KEY_INT64) {
        ob.key_i64.resize(n);
        for (size_t i = 0; i < n; i++) ob.key_i64[i] = op->dict_vals[rows[i]];
    } else {
        ob.key_i64.resize(n);
        for (size_t i = 0; i < n; i++) ob.key_i64[i] = rows[i];
    }
    for (size_t a = 0; a < na; a++) {
        switch (op->aggs[a].op) {
            case DZ_AGG_COUNT: {
                auto& col = ob.agg_i64[a];
                col.resize(n);
                for (size_t i = 0; i < n; i++) col[i] = (int64_t)f_cnt[rows[i]];
                break;
            }
            case DZ_AGG_MIN: {
                auto& col = ob.agg_f64[a];
                col.resize(n);
                for (size_t i = 0; i < n; i++)
                    col[i] = f_cnt[rows[i]] ? f_min[rows[i]] : 0.0;
                break;
            }
            case DZ_AGG_MAX: {
                auto& col = ob.agg_f64[a];
                col.resize(n);
                for (size_t i = 0; i < n; i++)
                    col[i] = f_cnt[rows[i]] ? f_max[rows[i]] : 0.0;
                break;
            }
            case DZ_AGG_SUM: {
                auto& col = ob.agg_f64[a];
                col.resize(n);
                for (size_t i = 0; i < n; i++)
                    col[i] = f_cnt[rows[i]] ? f_sum[rows[i]] : 0.0;
                break;
            }
            case DZ_AGG_AVG: {
                auto& col = ob.agg_f64[a];
                col.resize(n);
                for (size_t i = 0; i < n; i++) {
                    uint64_t c = f_cnt[rows[i]];
                    col[i] = c ? f_sum[rows[i]] / (double)c : 0.0;
                }
                break;
            }
        }
    }
    ob.agg_valid.resize(n);
    for (size_t i = 0; i < n; i++) ob.agg_valid[i] = f_cnt[rows[i]] > 0 ? 1 : 0;
    ob.wstart.assign(n, wstart);
    ob.wend.assign(n, wend);
    ob.view.n_rows = (int64_t)n;
}

/* Pinned emission slab layout for kcap entries (49 bytes per entry):
 * [kid u32][sidx u32][flags u8][cnt u64][min f64][max f64][sum f64][avg f64]
 * column sections; kid/cols are in COMPACT order, sidx is the sorted
 * (insertion-order) permutation of compact indices. kcap is a multiple of
 * NB=512, so every section stays 8-byte aligned. */
static constexpr int64_t SLAB_BYTES_PER_ENTRY = 56;

/* Build one emitted batch from the device-sorted, device-filtered columns
 * (insertion order already established by the GPU radix sort). Runs on the
 * emission worker pool: touches only immutable config, the deque-backed
 * dictionaries (indices < the job's n_keys snapshot) and the pinned slab. */
static void build_emission(dz_window_op* op, int64_t wstart, int64_t wend,
                           uint32_t nt, const uint64_t* slab, OutBuf* out,
                           int slab_idx = -1, bool zero_copy = false) {
    /* packed final-order columns from k_epermute (53 B/row): sequential
     * reads only — the per-row permutation gather already ran on device.
     * zero_copy: serve the aggregate columns (and dense keys + validity)
     * as VIEWS into the pinned span; the OutBuf keeps the slab until the
     * consumer's next poll. */
    const char* p = (const char*)slab;
    const int64_t* pkey = (const int64_t*)p;
    const uint64_t* pcnt = (const uint64_t*)(p + (size_t)nt * 8);
    const double* pmin = (const double*)(p + (size_t)nt * 16);
    const double* pmax = (const double*)(p + (size_t)nt * 24);
    const double* psum = (const double*)(p + (size_t)nt * 32);
    const double* pavg = (const double*)(p + (size_t)nt * 40);
    const uint32_t* pkid = (const uint32_t*)(p + (size_t)nt * 48);
    const uint8_t* pfl = (const uint8_t*)(p + (size_t)nt * 52);
    size_t n = nt;
    size_t na = op->aggs.size();
    OutBuf& ob = *out;
    ob.agg_i64.resize(na);
    ob.agg_f64.resize(na);
    if (zero_copy && n > 0) {
        ob.packed = p;
        ob.packed_nt = nt;
        ob.hold_slab = slab_idx;
    }
    if (op->no_group) {
        /* global aggregate: output schema has no group column
         * (create_schema with empty group exprs, streaming_window.rs:1096+) */
    } else if (op->key_kind == DZ_KEY_UTF8) {
        ob.key_offsets.resize(n + 1);
        ob.key_offsets[0] = 0;
        size_t total = 0;
        for (size_t i = 0; i < n; i++) total += op->dict_strs[pkid[i]].size();
        ob.key_data.resize(total);
        size_t pos = 0;
        for (size_t i = 0; i < n; i++) {
            const std::string& s = op->dict_strs[pkid[i]];
            memcpy(ob.key_data.data() + pos, s.data(), s.size());
            pos += s.size();
            ob.key_offsets[i + 1] = (int32_t)pos;
        }
    } else if (op->key_kind == DZ_KEY_INT64) {
        ob.key_i64.resize(n);
        for (size_t i = 0; i < n; i++) ob.key_i64[i] = op->dict_vals[pkid[i]];
    } else if (!ob.packed) {
        ob.key_i64.assign(pkey, pkey + n);
    }
    if (!ob.packed)
    for (size_t a = 0; a < na; a++) {
        switch (op->aggs[a].op) {
            case DZ_AGG_COUNT:
                ob.agg_i64[a].assign((const int64_t*)pcnt,
                                     (const int64_t*)pcnt + n);
                break;
            case DZ_AGG_MIN: ob.agg_f64[a].assign(pmin, pmin + n); break;
            case DZ_AGG_MAX: ob.agg_f64[a].assign(pmax, pmax + n); break;
            case DZ_AGG_SUM: ob.agg_f64[a].assign(psum, psum + n); break;
            case DZ_AGG_AVG: ob.agg_f64[a].assign(pavg, pavg + n); break;
        }
    }
    if (!ob.packed) ob.agg_valid.assign(pfl, pfl + n);
    ob.wstart.assign(n, wstart);
    ob.wend.assign(n, wend);
    ob.view.n_rows = (int64_t)n;
}

static OutBuf take_outbuf(dz_window_op* op) {
    std::lock_guard<std::mutex> lk(op->out_mtx);
    if (op->ob_pool.empty()) return OutBuf();
    OutBuf ob = std::move(op->ob_pool.back());
    op->ob_pool.pop_back();
    ob.packed = nullptr;
    ob.packed_nt = 0;
    ob.packed_off = 0;
    ob.hold_slab = -1;
    ob.hold_group.reset();
    ob.key_i64.clear();
    ob.key_offsets.clear();
    ob.key_data.clear();
    for (auto& c : ob.agg_i64) c.clear();
    for (auto& c : ob.agg_f64) c.clear();
    ob.agg_valid.clear();
    ob.wstart.clear();
    ob.wend.clear();
    return ob;
}

static void emit_drain(dz_window_op* op) {
    std::unique_lock<std::mutex> lk(op->e_mtx);
    op->e_cv.wait(lk, [&] { return op->e_inflight == 0; });
}

/* hipEventSynchronize pays ~0.5-1 ms blocked-wakeup latency per call; the
 * emission workers poll instead (hipEventQuery + yield): the events complete
 * in tens of microseconds once the copy stream reaches them.
 * TERMINAL errors (a device fault poisons every query with e.g.
 * hipErrorIllegalAddress) must break the loop — spinning on them forever
 * turns one bad kernel into a process hang; the next CHK'd hip call after
 * the spin reports the sticky error. */
static void event_spin(hipEvent_t ev) {
    hipError_t e;
    while ((e = hipEventQuery(ev)) != hipSuccess) {
        if (e != hipErrorNotReady) {
            fprintf(stderr, "dz: event_spin aborted: %s\n",
                    hipGetErrorString(e));
            return;
        }
        std::this_thread::yield();
    }
}

/* worker-side variant: a tight hipEventQuery loop from several worker
 * threads contends the runtime lock against the push thread's own queries
 * and launches; the workers can afford 20 µs of extra latency */
static void event_spin_relaxed(hipEvent_t ev) {
    hipError_t e;
    while ((e = hipEventQuery(ev)) != hipSuccess) {
        if (e != hipErrorNotReady) {
            fprintf(stderr, "dz: event_spin aborted: %s\n",
                    hipGetErrorString(e));
            return;
        }
        std::this_thread::sleep_for(std::chrono::microseconds(20));
    }
}

static void emit_worker_main(dz_window_op* op) {
    hipSetDevice(op->device);
    for (;;) {
        dz_window_op::EmitJob job;
        {
            std::unique_lock<std::mutex> lk(op->e_mtx);
            op->e_cv.wait(lk, [&] { return op->e_stop || !op->e_jobs.empty(); });
            if (op->e_jobs.empty()) {
                if (op->e_stop) return;
                continue;
            }
            job = op->e_jobs.front();
            op->e_jobs.pop_front();
        }
        event_spin_relaxed(job.ev); /* emission chain complete */
        auto t0 = std::chrono::steady_clock::now();
        OutBuf ob = take_outbuf(op);
        if (job.grouped) {
            /* group-batched close: the first sibling to claim grabs a
             * pinned span (heap fallback keeps liveness if consumers lag),
             * pulls the whole packed block D2H and publishes the per-close
             * offsets; every sibling then builds its own slice — ZERO-COPY
             * views into the span for large slices, the span held until
             * every viewing batch has been consumed */
            auto& ctl = *job.ctl;
            if (ctl.claimed.exchange(1) == 0) {
                const uint32_t* c = op->e_gpcnt + 32 * job.gbuf;
                uint32_t gtot = c[0];
                ctl.offs[0] = 0;
                for (int i = 0; i < 16; i++)
                    ctl.offs[i + 1] = ctl.offs[i] + c[1 + i];
                char* dst = nullptr;
                if (gtot) {
                    {
                        std::lock_guard<std::mutex> lk(op->e_mtx);
                        if (!op->e_gspan_free.empty()) {
                            ctl.span = op->e_gspan_free.back();
                            op->e_gspan_free.pop_back();
                        }
                    }
                    if (ctl.span >= 0) {
                        dst = op->e_gpin[ctl.span];
                    } else {
                        ctl.heap = (char*)malloc((size_t)gtot * 53);
                        dst = ctl.heap;
                    }
                    hipMemcpy(dst, op->e_gd[job.gbuf].pout,
                              (size_t)gtot * 53, hipMemcpyDeviceToHost);
                }
                ctl.gtot = gtot;
                ctl.ready.store(true, std::memory_order_release);
            } else {
                while (!ctl.ready.load(std::memory_order_acquire))
                    std::this_thread::sleep_for(std::chrono::microseconds(20));
            }
            const uint32_t off = ctl.offs[job.goff];
            const uint32_t cnt = ctl.offs[job.goff + 1] - off;
            op->e_nt_hint.store(cnt, std::memory_order_relaxed);
            const char* src = ctl.span >= 0 ? op->e_gpin[ctl.span] : ctl.heap;
            const bool zc = ctl.span >= 0 && cnt >= 16384;
            if (zc) {
                ctl.holders.fetch_add(1);
                ob.packed = src;
                ob.packed_nt = ctl.gtot;
                ob.packed_off = off;
                ob.hold_group = job.ctl;
                build_emission_slice_views(op, job.wstart, job.wend, cnt, &ob);
            } else {
                build_emission_slice(op, job.wstart, job.wend, ctl.gtot, off,
                                     cnt, src, &ob);
            }
        } else if (job.device) {
            /* copy exactly nt2 packed rows (the filter already ran on
             * device, so this is the final output volume, not the keyspace) */
            uint32_t nt = op->e_pcnt[job.slab];
            op->e_nt_hint.store(nt, std::memory_order_relaxed);
            if (nt > 0) {
                dz_window_op::DevEmit& d = op->e_dev[job.slab];
                /* the device already applied the sorted permutation and
                 * packed every final-order column: ONE contiguous pull */
                hipStream_t wcs = op->c_streams[job.cs];
                hipError_t e = hipMemcpyAsync(op->e_slabs[job.slab], d.pout,
                                              (size_t)nt * 53,
                                              hipMemcpyDeviceToHost, wcs);
                if (e == hipSuccess) e = hipEventRecord(job.ev, wcs);
                if (e != hipSuccess) {
                    /* surface on the op; emit an empty batch rather than
                     * formatting garbage */
                    op->err = std::string("emission copy failed: ") +
                              hipGetErrorString(e);
                    nt = 0;
                } else {
                    event_spin_relaxed(job.ev);
                }
            }
            bool zc = false;
            uint32_t gen = 0;
            if (nt >= 16384) {
                /* big outputs skip the materializing copies; LIVENESS: only
                 * while the slab pool stays deep — if consumers lag, fall
                 * back to copy+release so trigger_windows can always make
                 * progress */
                std::lock_guard<std::mutex> lk(op->e_mtx);
                zc = op->e_free.size() >= 8;
                gen = op->e_slab_gen;
            }
            build_emission(op, job.wstart, job.wend, nt,
                           op->e_slabs[job.slab], &ob, job.slab, zc);
            ob.slab_gen = gen;
            (zc ? op->e_zc_builds : op->e_copy_builds)++;
        } else {
            /* gbuf jobs: job.ev = gather done, and the gather wrote this
             * close's slab directly into the pinned group buffer */
            const uint64_t* slab = job.gbuf >= 0
                ? op->e_gbufs[job.gbuf] + (size_t)job.goff * job.kcap * 5
                : op->e_slabs[job.slab];
            build_emission_host(op, job.wstart, job.wend, job.n_keys, job.kcap,
                                slab, &ob);
        }
        op->e_build_ns += (uint64_t)std::chrono::duration_cast<std::chrono::nanoseconds>(
            std::chrono::steady_clock::now() - t0).count();
        op->e_builds++;
        const int ob_holds_slab = ob.hold_slab;
        {
            std::lock_guard<std::mutex> lk(op->out_mtx);
            op->e_done.emplace(job.ticket, std::move(ob));
            while (!op->e_done.empty() &&
                   op->e_done.begin()->first == op->e_ticket_pop) {
                op->outq.push_back(std::move(op->e_done.begin()->second));
                op->e_done.erase(op->e_done.begin());
                op->e_ticket_pop++;
            }
        }
        {
            std::lock_guard<std::mutex> lk(op->e_mtx);
            if (job.slab >= 0 && ob_holds_slab != job.slab)
                op->e_free.push_back(job.slab);
            /* group-shared resources go back to their pools only once every
             * sibling is past its spin (i.e. fully built) */
            if (!job.grp_left || job.grp_left->fetch_sub(1) == 1) {
                op->e_ev_pool.push_back(job.ev);
                if (job.grouped) {
                    op->e_gdfree.push_back(job.gbuf);
                    auto& ctl = *job.ctl;
                    ctl.all_built.store(true);
                    if (ctl.holders.load() == 0) { /* no zero-copy viewers */
                        if (ctl.span >= 0) {
                            op->e_gspan_free.push_back(ctl.span);
                            ctl.span = -1;
                        }
                        free(ctl.heap);
                        ctl.heap = nullptr;
                    }
                } else if (job.gbuf >= 0) {
                    op->e_gfree.push_back(job.gbuf);
                }
            }
            op->e_inflight--;
        }
        op->e_cv.notify_all();
    }
}

static dz_status ensure_emission(dz_window_op* op) {
    if (op->e_slab_kcap == op->kcap) return DZ_OK;
    {
        emit_drain(op);
        int64_t kc = op->kcap;
        int64_t nblk = (kc + dz::EMIT_RCHUNK - 1) / dz::EMIT_RCHUNK;
        for (int s = 0; s < dz_window_op::E_CSTREAMS; s++) {
            hipFree(op->d_rhist[s]);
            hipFree(op->d_roffs[s]);
            /* + tail: RSEG partial rows + digit bases for the parallel scan */
            CHK(op, hipMalloc(&op->d_rhist[s],
                              (size_t)(nblk + 17) * dz::EMIT_RBINS * 4));
            CHK(op, hipMalloc(&op->d_roffs[s],
                              (size_t)nblk * dz::EMIT_RBINS * 4));
        }
        if (!op->e_pcnt)
            CHK(op, hipHostMalloc((void**)&op->e_pcnt, dz_window_op::E_POOL * 4));
        for (int i = 0; i < dz_window_op::E_POOL; i++) {
            auto& d = op->e_dev[i];
            hipFree(d.base);
            /* 3x u64 keys + 5x u64/f64 columns + 4x u32 + flags + counters
             * + the 53 B/entry packed permuted-output block */
            CHK(op, hipMalloc(&d.base, (size_t)kc * 140 + 128));
            char* p = d.base;
            d.ekeys = (uint64_t*)p; p += kc * 8;
            d.fkeys = (uint64_t*)p; p += kc * 8;
            d.skeys = (uint64_t*)p; p += kc * 8;
            d.ocnt = (uint64_t*)p; p += kc * 8;
            d.omin = (double*)p; p += kc * 8;
            d.omax = (double*)p; p += kc * 8;
            d.osum = (double*)p; p += kc * 8;
            d.oavg = (double*)p; p += kc * 8;
            d.ekid = (uint32_t*)p; p += kc * 4;
            d.fkid = (uint32_t*)p; p += kc * 4;
            d.fiota = (uint32_t*)p; p += kc * 4;
            d.okid = (uint32_t*)p; p += kc * 4;
            d.oflags = (uint8_t*)p; p += kc;
            d.counter = (uint32_t*)p; p += 64;
            d.pout = p;
            if (op->e_slabs[i]) op->e_graveyard.push_back(op->e_slabs[i]);
            CHK(op, hipHostMalloc((void**)&op->e_slabs[i],
                                  (size_t)kc * SLAB_BYTES_PER_ENTRY + 16));
        }
        /* group buffers for batched host-path emission (small keyspaces
         * only: a group buffer is EGROUP whole slabs) */
        for (int i = 0; i < dz_window_op::E_GBUFS; i++)
            if (op->e_gbufs[i]) {
                hipHostFree(op->e_gbufs[i]);
                op->e_gbufs[i] = nullptr;
            }
        bool have_gbufs = kc <= 65536;
        if (have_gbufs) {
            size_t gbytes = (size_t)(dz_window_op::E_POOL / 2) * kc * 5 * 8;
            for (int i = 0; i < dz_window_op::E_GBUFS; i++) {
                CHK(op, hipHostMalloc((void**)&op->e_gbufs[i], gbytes,
                                      hipHostMallocMapped));
                CHK(op, hipHostGetDevicePointer((void**)&op->e_gbufs_dev[i],
                                                op->e_gbufs[i], 0));
            }
        }
        std::lock_guard<std::mutex> lk(op->e_mtx);
        op->e_slab_gen++;
        op->e_free.clear();
        for (int i = 0; i < dz_window_op::E_POOL; i++) op->e_free.push_back(i);
        op->e_gfree.clear();
        if (have_gbufs)
            for (int i = 0; i < dz_window_op::E_GBUFS; i++)
                op->e_gfree.push_back(i);
        op->e_slab_kcap = op->kcap;
    }
    return DZ_OK;
}

/* allocate the group-batched emission scratch (one-time) */
static dz_status ensure_gd(dz_window_op* op) {
    if (op->e_gd_ready) return DZ_OK;
    constexpr int64_t E = dz_window_op::E_GELEMS;
    const int64_t nblk = (E + dz::EMIT_RCHUNK - 1) / dz::EMIT_RCHUNK;
    for (int i = 0; i < dz_window_op::E_GD; i++) {
        auto& g = op->e_gd[i];
        size_t bytes = (size_t)E * (8 + 8 + 4 + 4 + 4 + 8 + 32 + 1 + 53) +
                       (size_t)(nblk + 17 + nblk) * dz::EMIT_RBINS * 4 + 256;
        CHK(op, hipMalloc(&g.base, bytes));
        char* p = g.base;
        g.gkeys = (uint64_t*)p; p += E * 8;
        g.gskeys = (uint64_t*)p; p += E * 8;
        g.gcnt_col = (uint64_t*)p; p += E * 8;
        g.gmin = (double*)p; p += E * 8;
        g.gmax = (double*)p; p += E * 8;
        g.gsum = (double*)p; p += E * 8;
        g.gavg = (double*)p; p += E * 8;
        g.gkid = (uint32_t*)p; p += E * 4;
        g.gokid = (uint32_t*)p; p += E * 4;
        g.giota = (uint32_t*)p; p += E * 4;
        g.rhist = (uint32_t*)p; p += (size_t)(nblk + 17) * dz::EMIT_RBINS * 4;
        g.roffs = (uint32_t*)p; p += (size_t)nblk * dz::EMIT_RBINS * 4;
        g.gflags = (uint8_t*)p; p += E;
        g.ctr = (uint32_t*)p; p += 128;
        g.pout = p;
        CHK(op, hipHostMalloc((void**)&op->e_gpin[i], (size_t)E * 53));
    }
    CHK(op, hipHostMalloc((void**)&op->e_gpcnt,
                          (size_t)dz_window_op::E_GD * 32 * 4));
    {
        std::lock_guard<std::mutex> lk(op->e_mtx);
        for (int i = 0; i < dz_window_op::E_GD; i++) {
            op->e_gdfree.push_back(i);
            op->e_gspan_free.push_back(i);
        }
    }
    op->e_gd_ready = true;
    return DZ_OK;
}

/* Build one close's slice of a group-batched packed span (close-major,
 * insertion-ordered): total = the group's element count (column-section
 * stride), [off, off+n) = this close's rows. Materialising build — the
 * group path runs in the many-SMALL-closes regime by construction. */
static void build_emission_slice(dz_window_op* op, int64_t wstart,
                                 int64_t wend, uint32_t total, uint32_t off,
                                 uint32_t n, const char* p, OutBuf* out) {
    const int64_t* pkey = (const int64_t*)p + off;
    const uint64_t* pcnt = (const uint64_t*)(p + (size_t)total * 8) + off;
    const double* pmin = (const double*)(p + (size_t)total * 16) + off;
    const double* pmax = (const double*)(p + (size_t)total * 24) + off;
    const double* psum = (const double*)(p + (size_t)total * 32) + off;
    const double* pavg = (const double*)(p + (size_t)total * 40) + off;
    const uint32_t* pkid = (const uint32_t*)(p + (size_t)total * 48) + off;
    const uint8_t* pfl = (const uint8_t*)(p + (size_t)total * 52) + off;
    size_t na = op->aggs.size();
    OutBuf& ob = *out;
    ob.agg_i64.resize(na);
    ob.agg_f64.resize(na);
    if (op->no_group) {
    } else if (op->key_kind == DZ_KEY_UTF8) {
        ob.key_offsets.resize(n + 1);
        ob.key_offsets[0] = 0;
        size_t totalb = 0;
        for (size_t i = 0; i < n; i++) totalb += op->dict_strs[pkid[i]].size();
        ob.key_data.resize(totalb);
        size_t pos = 0;
        for (size_t i = 0; i < n; i++) {
            const std::string& str = op->dict_strs[pkid[i]];
            memcpy(ob.key_data.data() + pos, str.data(), str.size());
            pos += str.size();
            ob.key_offsets[i + 1] = (int32_t)pos;
        }
    } else if (op->key_kind == DZ_KEY_INT64) {
        ob.key_i64.resize(n);
        for (size_t i = 0; i < n; i++) ob.key_i64[i] = op->dict_vals[pkid[i]];
    } else {
        ob.key_i64.assign(pkey, pkey + n);
    }
    for (size_t a = 0; a < na; a++) {
        switch (op->aggs[a].op) {
            case DZ_AGG_COUNT:
                ob.agg_i64[a].assign((const int64_t*)pcnt,
                                     (const int64_t*)pcnt + n);
                break;
            case DZ_AGG_MIN: ob.agg_f64[a].assign(pmin, pmin + n); break;
            case DZ_AGG_MAX: ob.agg_f64[a].assign(pmax, pmax + n); break;
            case DZ_AGG_SUM: ob.agg_f64[a].assign(psum, psum + n); break;
            case DZ_AGG_AVG: ob.agg_f64[a].assign(pavg, pavg + n); break;
        }
    }
    ob.agg_valid.assign(pfl, pfl + n);
    ob.wstart.assign(n, wstart);
    ob.wend.assign(n, wend);
    ob.view.n_rows = (int64_t)n;
}

/* Zero-copy slice build: the aggregate columns (and dense keys +
 * validity) are served by poll() as views into the group span at
 * packed_off; only dictionary-encoded keys materialize here. */
static void build_emission_slice_views(dz_window_op* op, int64_t wstart,
                                       int64_t wend, uint32_t n, OutBuf* out) {
    OutBuf& ob = *out;
    const char* p = ob.packed;
    const uint32_t total = ob.packed_nt;
    const uint32_t* pkid =
        (const uint32_t*)(p + (size_t)total * 48) + ob.packed_off;
    size_t na = op->aggs.size();
    ob.agg_i64.resize(na);
    ob.agg_f64.resize(na);
    if (op->no_group) {
    } else if (op->key_kind == DZ_KEY_UTF8) {
        ob.key_offsets.resize(n + 1);
        ob.key_offsets[0] = 0;
        size_t totalb = 0;
        for (size_t i = 0; i < n; i++) totalb += op->dict_strs[pkid[i]].size();
        ob.key_data.resize(totalb);
        size_t pos = 0;
        for (size_t i = 0; i < n; i++) {
            const std::string& str = op->dict_strs[pkid[i]];
            memcpy(ob.key_data.data() + pos, str.data(), str.size());
            pos += str.size();
            ob.key_offsets[i + 1] = (int32_t)pos;
        }
    } else if (op->key_kind == DZ_KEY_INT64) {
        ob.key_i64.resize(n);
        for (size_t i = 0; i < n; i++) ob.key_i64[i] = op->dict_vals[pkid[i]];
    } /* dense: the key column is a view (poll) */
    ob.wstart.assign(n, wstart);
    ob.wend.assign(n, wend);
    ob.view.n_rows = (int64_t)n;
}

/* trigger_windows (grouped_window_agg_stream.rs:220-253): closed windows are
 * copied D2H asynchronously and built by the worker thread off the push
 * critical path; dz_window_op_drain/finish wait for completion. */
static dz_status trigger_windows(dz_window_op* op) {
    if (!op->has_wm) return DZ_OK;
    struct Closed { int64_t start, end; int32_t slot; uint64_t base; };
    std::vector<Closed> closed;
    for (auto it = op->open.begin(); it != op->open.end();) {
        if (op->watermark >= it->second.end) {
            closed.push_back({it->first, it->second.end, it->second.slot,
                              it->second.row_base});
            it = op->open.erase(it);
        } else {
            ++it;
        }
    }
    if (closed.empty()) return DZ_OK;
    if (ensure_emission(op) != DZ_OK) return DZ_ERR;
    size_t stride = (size_t)op->kcap * 5;
    HostTimer ht(op, "h_emit_enqueue");
    /* emission kernels run on the copy stream AFTER the compute stream's
     * folds for these slots */
    hipEvent_t evA;
    {
        std::lock_guard<std::mutex> lk(op->e_mtx);
        if (!op->e_ev_pool.empty()) {
            evA = op->e_ev_pool.back();
            op->e_ev_pool.pop_back();
        } else {
            hipEventCreate(&evA);
        }
    }
    {
        HostTimer htf(op, "h_trig_fence");
        CHK(op, hipEventRecord(evA, op->stream));
        for (int s = 0; s < dz_window_op::E_CSTREAMS; s++)
            CHK(op, hipStreamWaitEvent(op->c_streams[s], evA, 0));
    }
    {
        std::lock_guard<std::mutex> lk(op->e_mtx);
        op->e_ev_pool.push_back(evA);
    }
    dz::EmitFilter ef;
    ef.on = op->has_filter ? 1 : 0;
    ef.cmp = op->f_cmp;
    ef.lit = op->f_lit;
    ef.field = 4;
    if (op->has_filter) {
        switch (op->aggs[op->f_idx].op) {
            case DZ_AGG_COUNT: ef.field = 0; break;
            case DZ_AGG_MIN: ef.field = 1; break;
            case DZ_AGG_MAX: ef.field = 2; break;
            case DZ_AGG_SUM: ef.field = 3; break;
            default: ef.field = 4; break;
        }
    }
    /* Two phases per GROUP of closes: slab readers (compact+gather) for the
     * whole group first, so the slot-release events precede the group's
     * sorts/copies on the copy stream — window slots come back to the
     * compute stream right after its folds, not behind emission work.
     * Groups are bounded by half the slab pool: phase 1 must never wait for
     * slabs whose release depends on this group's phase 2 (deadlock). */
    struct Pending { int slab; hipEvent_t ev; };
    constexpr size_t EGROUP = dz_window_op::E_POOL / 2;
    const bool dev_path_g = op->n_keys > 65536;
    /* group-batched device path: a whole trigger group of SMALL closes in
     * ONE chain (~25 host enqueues total instead of per close — the
     * push-thread enqueue serialization was the cfg3 sliding wall). Large
     * closes (hint above the threshold) keep the per-close chain with its
     * zero-copy large-output path. Worst-case safe: gstep is capped so
     * gcount * kcap <= E_GELEMS — the fused gather cannot overflow. */
    static const uint32_t group_max = [] {
        const char* v = getenv("DZ_EMIT_GROUP_MAX");
        return v ? (uint32_t)atoi(v) : 65536u;
    }();
    const bool grouped_ok = dev_path_g && op->kcap > 0 &&
        op->kcap <= dz_window_op::E_GELEMS &&
        op->e_nt_hint.load(std::memory_order_relaxed) <= group_max &&
        op->batch_seq < (1u << 19);
    size_t g0 = 0;
    while (g0 < closed.size()) {
    size_t gstep = EGROUP;
    if (grouped_ok)
        gstep = std::min<size_t>(gstep, std::max<size_t>(1,
            (size_t)(dz_window_op::E_GELEMS / op->kcap)));
    const size_t g1 = std::min(closed.size(), g0 + gstep);
    if (grouped_ok) {
        if (ensure_gd(op) != DZ_OK) return DZ_ERR;
        int gdidx;
        hipEvent_t gev;
        {
            HostTimer htw(op, "h_emit_slabwait");
            std::unique_lock<std::mutex> lk(op->e_mtx);
            op->e_cv.wait(lk, [&] { return !op->e_gdfree.empty(); });
            gdidx = op->e_gdfree.back();
            op->e_gdfree.pop_back();
            if (!op->e_ev_pool.empty()) {
                gev = op->e_ev_pool.back();
                op->e_ev_pool.pop_back();
            } else {
                hipEventCreate(&gev);
            }
        }
        auto& gd = op->e_gd[gdidx];
        const int gcount = (int)(g1 - g0);
        hipStream_t cs = op->c_streams[gdidx % dz_window_op::E_CSTREAMS];
        dz::EGatherSlots gs;
        hipEvent_t fr = op->e_frontier[op->e_frontier_idx];
        if (!fr) {
            hipEventCreateWithFlags(&fr, hipEventDisableTiming);
            op->e_frontier[op->e_frontier_idx] = fr;
        }
        op->e_frontier_idx =
            (op->e_frontier_idx + 1) % dz_window_op::E_FRONTIERS;
        uint64_t max_first = 1;
        for (size_t ci = g0; ci < g1; ci++) {
            gs.s[ci - g0] = closed[ci].slot;
            gs.base[ci - g0] = closed[ci].base;
            max_first = std::max(max_first,
                                 op->rows_seen - closed[ci].base);
            op->free_slots.push_back({closed[ci].slot, fr, false});
        }
        /* composite sort key = close_idx << cshift | (first - base); the
         * rebased first is bounded by the window's row span, so cshift
         * stays ~24-34 bits at any stream age */
        const int cshift = 64 - __builtin_clzll(max_first);
        {
            HostTimer htg(op, "h_trig_gather");
            CHK(op, hipMemsetAsync(gd.ctr, 0, 128, cs));
            dz::launch_emission_group_read(cs, op->s_base, (int64_t)stride, gs,
                                           gcount, op->n_keys, op->kcap, ef,
                                           cshift, gd.gkeys, gd.gkid,
                                           gd.giota, gd.gcnt_col, gd.gmin,
                                           gd.gmax, gd.gsum, gd.gavg,
                                           gd.gflags, gd.ctr, gd.ctr + 1);
            CHK(op, hipEventRecord(fr, cs)); /* slabs fully read */
            dz::launch_emission_group_sort(cs, dz_window_op::E_GELEMS, gcount,
                                           cshift, max_first, gd.gkeys,
                                           gd.gskeys, gd.gkid, gd.gokid,
                                           gd.giota, gd.gcnt_col, gd.gmin,
                                           gd.gmax, gd.gsum, gd.gavg,
                                           gd.gflags, gd.ctr, gd.rhist,
                                           gd.roffs, gd.pout);
            CHK(op, hipMemcpyAsync(op->e_gpcnt + 32 * gdidx, gd.ctr, 128,
                                   hipMemcpyDeviceToHost, cs));
            CHK(op, hipEventRecord(gev, cs));
        }
        auto left = std::make_shared<std::atomic<int>>(gcount);
        auto ctl = std::make_shared<GroupCtl>();
        {
            HostTimer htj(op, "h_trig_jobs");
            std::lock_guard<std::mutex> lk(op->e_mtx);
            for (size_t ci = g0; ci < g1; ci++) {
                dz_window_op::EmitJob j;
                j.ev = gev;
                j.slab = -1;
                j.wstart = closed[ci].start;
                j.wend = closed[ci].end;
                j.n_keys = op->n_keys;
                j.kcap = op->kcap;
                j.device = true;
                j.grouped = true;
                j.gbuf = gdidx;
                j.goff = (int)(ci - g0);
                j.grp_left = left;
                j.ctl = ctl;
                j.ticket = op->e_ticket_next++;
                op->e_jobs.push_back(std::move(j));
                op->e_inflight++;
            }
        }
        op->e_cv.notify_all();
        g0 = g1;
        continue;
    }
    if (!dev_path_g && op->e_gbufs[0]) {
        /* host-built closes (small keyspaces): pack the whole group's slabs
         * with one gather launch + ONE D2H into a pinned group buffer behind
         * ONE shared event — the push thread pays a fixed ~4 hip calls per
         * group instead of ~5 per close (measured ~55 µs x ~8 closes/step
         * at cfg2) */
        hipEvent_t gev;
        int gbuf;
        {
            HostTimer htw(op, "h_emit_slabwait");
            std::unique_lock<std::mutex> lk(op->e_mtx);
            op->e_cv.wait(lk, [&] { return !op->e_gfree.empty(); });
            gbuf = op->e_gfree.back();
            op->e_gfree.pop_back();
            if (!op->e_ev_pool.empty()) {
                gev = op->e_ev_pool.back();
                op->e_ev_pool.pop_back();
            } else {
                hipEventCreate(&gev);
            }
        }
        dz::EGatherSlots gs;
        const int gcount = (int)(g1 - g0);
        /* slots are readable again once the gather (not the D2H) is done:
         * gate their reuse on the next rotating frontier event */
        hipEvent_t fr = op->e_frontier[op->e_frontier_idx];
        if (!fr) {
            hipEventCreateWithFlags(&fr, hipEventDisableTiming);
            op->e_frontier[op->e_frontier_idx] = fr;
        }
        op->e_frontier_idx = (op->e_frontier_idx + 1) % dz_window_op::E_FRONTIERS;
        for (size_t ci = g0; ci < g1; ci++) {
            gs.s[ci - g0] = closed[ci].slot;
            op->free_slots.push_back({closed[ci].slot, fr, false});
        }
        {
            /* ONE launch reads the closing slots and writes their slabs
             * straight into the pinned group buffer over the host link —
             * fully asynchronous for the push thread (hipMemcpyAsync D2H
             * was measured performing the transfer at ENQUEUE time) */
            HostTimer htg(op, "h_trig_gather");
            dz::launch_egather_slabs(op->copy_stream, op->s_base, stride, gs,
                                     gcount, op->e_gbufs_dev[gbuf]);
            CHK(op, hipEventRecord(fr, op->copy_stream));
            CHK(op, hipEventRecord(gev, op->copy_stream));
        }
        auto left = std::make_shared<std::atomic<int>>(gcount);
        {
            HostTimer htj(op, "h_trig_jobs");
            std::lock_guard<std::mutex> lk(op->e_mtx);
            for (size_t ci = g0; ci < g1; ci++) {
                dz_window_op::EmitJob j;
                j.ev = gev;
                j.slab = -1;
                j.wstart = closed[ci].start;
                j.wend = closed[ci].end;
                j.n_keys = op->n_keys;
                j.kcap = op->kcap;
                j.device = false;
                j.grp_left = left;
                j.gbuf = gbuf;
                j.goff = (int)(ci - g0);
                j.ticket = op->e_ticket_next++;
                op->e_jobs.push_back(std::move(j));
                op->e_inflight++;
            }
        }
        op->e_cv.notify_all();
        g0 = g1;
        continue;
    }
    std::vector<Pending> pend;
    pend.reserve(g1 - g0);
    /* LIVENESS: a slab taken in phase 1 is dead weight until phase 2
     * pushes its job — workers can only release slabs for jobs that
     * exist. Block for the group's FIRST slab only; if the pool dries
     * mid-group, truncate the group here, run phase 2 for the slabs
     * already taken, and continue (the outer loop re-forms the rest).
     * Without this, >=17 pending closes plus a zero-copy hold backlog
     * (slabs pinned until the consumer polls) deadlock the pipeline —
     * reproduced under rocprofv3's enqueue slowdown. */
    size_t g1c = g1;
    for (size_t ci = g0; ci < g1c; ci++) {
        auto& c = closed[ci];
        int slab;
        hipEvent_t ev, slot_ev;
        {
            HostTimer htw(op, "h_emit_slabwait");
            std::unique_lock<std::mutex> lk(op->e_mtx);
            if (ci == g0) {
                op->e_cv.wait(lk, [&] { return !op->e_free.empty(); });
            } else if (op->e_free.empty()) {
                g1c = ci;
                break;
            }
            slab = op->e_free.back();
            op->e_free.pop_back();
            auto take = [&]() -> hipEvent_t {
                if (!op->e_ev_pool.empty()) {
                    hipEvent_t e = op->e_ev_pool.back();
                    op->e_ev_pool.pop_back();
                    return e;
                }
                hipEvent_t e;
                hipEventCreate(&e);
                return e;
            };
            ev = take();
            slot_ev = take();
            op->e_inflight++;
        }
        dz_window_op::DevEmit& d = op->e_dev[slab];
        const uint64_t* sl = op->s_base + (size_t)c.slot * stride;
        const bool dev_path = op->n_keys > 65536;
        hipStream_t cs = op->c_streams[(ci - g0) % dz_window_op::E_CSTREAMS];
        if (!dev_path) {
            /* small keyspaces: ONE pinned copy of the raw slab; the worker
             * sorts/filters on host (15 stream ops per close would cost more
             * than the whole host build at this scale) */
            CHK(op, hipMemcpyAsync(op->e_slabs[slab], sl, stride * 8,
                                   hipMemcpyDeviceToHost, cs));
        } else {
        dz::launch_zero_counters(cs, d.counter);
        if (op->n_keys > 0) {
            dz::launch_emission_slabread(cs,
                                         /*first*/ sl + op->kcap,
                                         /*cnt*/ sl,
                                         /*min*/ (const double*)(sl + 2 * op->kcap),
                                         /*max*/ (const double*)(sl + 3 * op->kcap),
                                         /*sum*/ (const double*)(sl + 4 * op->kcap),
                                         c.base, op->n_keys, d.ekeys, d.ekid, d.fkeys,
                                         d.fkid, d.fiota, d.counter,
                                         d.counter + 1, ef, d.ocnt, d.omin,
                                         d.omax, d.osum, d.oavg, d.oflags);
        }
        }
        CHK(op, hipEventRecord(slot_ev, cs));
        op->free_slots.push_back({c.slot, slot_ev});
        pend.push_back({slab, ev});
    }
    /* phase 2: sorts + column D2H + job hand-off */
    for (size_t ci = g0; ci < g1c; ci++) {
        auto& c = closed[ci];
        int slab = pend[ci - g0].slab;
        hipEvent_t ev = pend[ci - g0].ev;
        dz_window_op::DevEmit& d = op->e_dev[slab];
        const bool dev_path = op->n_keys > 65536;
        int csi = (int)((ci - g0) % dz_window_op::E_CSTREAMS);
        hipStream_t cs = op->c_streams[csi];
        if (dev_path && op->n_keys > 0) {
            /* adaptive: when the last close passed few groups (filtered
             * sliding closes pass ~1-2%), ONE single-block launch replaces
             * the ~15-launch multi-block radix chain — the launch enqueue
             * itself was the cfg3 push-thread wall. A misprediction is
             * slow, never wrong (the small kernel handles any nt). */
            /* sort keys are window-rebased (first - base): the bound is
             * the window's row span, so the radix pass count stays at
             * 3-4 regardless of stream age */
            const uint64_t maxk = op->rows_seen - c.base;
            static const uint32_t small_max = [] {
                const char* v = getenv("DZ_EMIT_SMALL_MAX");
                /* default OFF: A/B on one box measured the single-block
                 * form strictly slower (it hogs its emission stream); the
                 * live-count clamps in the multi-block chain won instead */
                return v ? (uint32_t)atoi(v) : 0u;
            }();
            if (small_max &&
                op->e_nt_hint.load(std::memory_order_relaxed) <= small_max)
                dz::launch_esort_small(cs, d.fkeys, d.skeys, d.fiota, d.okid,
                                       d.counter + 1, maxk);
            else
                dz::launch_emission_sort(cs, op->n_keys, d.fkeys,
                                         d.skeys, d.fiota, d.okid,
                                         d.counter + 1, op->d_rhist[csi],
                                         op->d_roffs[csi], maxk);
            /* pack the final-order columns on device: the worker pulls one
             * contiguous span and builds with sequential copies */
            dz::launch_emission_permute(cs, op->n_keys, d.counter + 1,
                                        d.fiota, d.fkid, d.ocnt, d.omin,
                                        d.omax, d.osum, d.oavg, d.oflags,
                                        d.pout);
        }
        if (dev_path)
            CHK(op, hipMemcpyAsync(&op->e_pcnt[slab], d.counter + 1, 4,
                                   hipMemcpyDeviceToHost, cs));
        CHK(op, hipEventRecord(ev, cs));
        {
            std::lock_guard<std::mutex> lk(op->e_mtx);
            dz_window_op::EmitJob j;
            j.ev = ev;
            j.slab = slab;
            j.wstart = c.start;
            j.wend = c.end;
            j.n_keys = op->n_keys;
            j.kcap = op->kcap;
            j.ticket = op->e_ticket_next++;
            j.device = dev_path;
            j.cs = csi;
            op->e_jobs.push_back(std::move(j));
        }
    }
    op->e_cv.notify_all(); /* one wakeup per group, not per close */
    g0 = g1c;
    } /* group loop */
    return DZ_OK;
}

/* ------------------------------------------------------------------ */
/* the push core (device-resident inputs)                              */
/* ------------------------------------------------------------------ */

/* Growth paths free device buffers that in-flight kernels of the previous
 * (pipelined) step may still read: quiesce both streams first. hipFree's
 * implicit device sync would cover it, but growth is rare — be explicit. */
static void quiesce(dz_window_op* op) {
    hipStreamSynchronize(op->stream);
    if (op->i_stream) hipStreamSynchronize(op->i_stream);
}

static dz_status ensure_scratch(dz_window_op* op, int C, int64_t nrec) {
    if (C > op->C_cap) {
        quiesce(op);
        for (int i = 0; i < 2; i++) {
            hipFree(op->d_ghist[i]); hipFree(op->d_gofs[i]);
            CHK(op, hipMalloc(&op->d_ghist[i], (size_t)C * dz::NB * 4));
            CHK(op, hipMalloc(&op->d_gofs[i], (size_t)C * dz::NB * 4));
            if (!op->d_total[i]) {
                CHK(op, hipMalloc(&op->d_total[i],
                                  (size_t)dz::SCAN_SSPLIT * dz::NB * 4));
                CHK(op, hipMalloc(&op->d_base[i], (dz::NB + 1) * 4));
            }
        }
        op->C_cap = C;
    }
    if (nrec > op->rec_cap) {
        quiesce(op);
        hipFree(op->d_grec);
        CHK(op, hipMalloc(&op->d_grec, (size_t)nrec * 16));
        op->rec_cap = nrec;
    }
    return DZ_OK;
}

/* Section 1 of a push: stage inputs and run the reduction whose scalars
 * (batch min/max ts, max key id) the host must read before it can lay out
 * window frames. deferred=true (device-push pipeline) runs it on i_stream
 * against an op-owned staging copy of the inputs, records ev_ready, and
 * leaves the rest for process_pending() at the next call into the op;
 * deferred=false (host-batch path) runs in-stream on the compute stream.
 * Tumbling fast path: one fused 12 B/row pass computes the histogram AND
 * min/max ts + max key id (the histogram needs no window params when every
 * row has multiplicity 1). Sliding needs min/max first (the window grid
 * anchors multiplicity), so it keeps a separate reduction pass. */
static dz_status stage_core(dz_window_op* op, int64_t n, const int64_t* d_ts,
                            const int32_t* d_kid, const double* d_vals,
                            const uint8_t* d_valbm, bool keys_are_dense,
                            bool deferred, bool borrow = false) {
    if (n >= (1LL << 31)) {
        op->err = "batch exceeds 2^31 rows; push smaller batches";
        return DZ_ERR;
    }
    const bool sliding = op->slide_ms > 0;
    int C = (int)std::min<int64_t>(512, std::max<int64_t>(1, (n + 8191) / 8192));
    int64_t chunk = (n + C - 1) / C;
    int b = op->next_buf;
    op->next_buf ^= 1;
    hipStream_t s = deferred ? op->i_stream : op->stream;
    if (deferred && !borrow) {
        HostTimer ht(op, "h_stage");
        if (n > op->s_in_cap[b]) {
            quiesce(op);
            hipFree(op->d_sts[b]); hipFree(op->d_skid[b]); hipFree(op->d_svals[b]);
            CHK(op, hipMalloc(&op->d_sts[b], (size_t)n * 8));
            CHK(op, hipMalloc(&op->d_skid[b], (size_t)n * 4));
            CHK(op, hipMalloc(&op->d_svals[b], (size_t)n * 8));
            op->s_in_cap[b] = n;
        }
        /* buffer b was last read by the scatter two pushes ago */
        if (op->consumed_valid[b])
            CHK(op, hipStreamWaitEvent(s, op->ev_consumed[b], 0));
        CHK(op, hipMemcpyAsync(op->d_sts[b], d_ts, (size_t)n * 8,
                               hipMemcpyDeviceToDevice, s));
        CHK(op, hipMemcpyAsync(op->d_skid[b], d_kid, (size_t)n * 4,
                               hipMemcpyDeviceToDevice, s));
        CHK(op, hipMemcpyAsync(op->d_svals[b], d_vals, (size_t)n * 8,
                               hipMemcpyDeviceToDevice, s));
        CHK(op, hipEventRecord(op->ev_staged[b], s));
        d_ts = op->d_sts[b];
        d_kid = op->d_skid[b];
        d_vals = op->d_svals[b];
    } else if (deferred && op->consumed_valid[b]) {
        /* borrowed push: caller keeps the inputs valid; the i_stream
         * reduction still writes d_ghist[b]/d_scalars[b], whose previous
         * contents the compute stream reads until that batch's scatter */
        CHK(op, hipStreamWaitEvent(s, op->ev_consumed[b], 0));
    } else if (op->consumed_valid[b]) {
        /* host-batch path interleaved with device pushes: the compute stream
         * must not overwrite d_ghist[b]/d_scalars[b] before the i_stream
         * reduction that last wrote them was consumed — in-stream ordering
         * covers compute-stream reuse, this wait covers the cross-stream case */
        CHK(op, hipStreamWaitEvent(s, op->ev_consumed[b], 0));
    }
    dz::launch_arm_scalars(s, op->d_scalars[b]);
    dz::WinParams wp;
    memset(&wp, 0, sizeof(wp));
    wp.len_ms = op->len_ms;
    wp.slide_ms = op->slide_ms;
    wp.is_sliding = sliding;
    if (!sliding) {
        if (ensure_scratch(op, C, n) != DZ_OK) return DZ_ERR;
        timed_on(op, s, "hist", (double)n * 12, [&] {
            dz::launch_hist(s, d_kid, d_ts, n, chunk, C, wp,
                            op->d_ghist[b], op->d_scalars[b]);
        });
        /* the scan depends only on the histogram (not on the host's window
         * decisions): run it here so process_pending's critical path starts
         * at the scatter */
        timed_on(op, s, "scan", (double)C * dz::NB * 12, [&] {
            dz::launch_scan(s, op->d_ghist[b], C, op->d_total[b],
                            op->d_base[b], op->d_gofs[b]);
        });
    } else {
        timed_on(op, s, "minmax", (double)n * 12, [&] {
            dz::launch_minmax(s, d_ts, keys_are_dense ? d_kid : nullptr,
                              n, op->d_scalars[b]);
        });
    }
    CHK(op, hipMemcpyAsync(op->h_scalars + 3 * b, op->d_scalars[b], 24,
                           hipMemcpyDeviceToHost, s));
    CHK(op, hipEventRecord(op->ev_ready[b], s));
    op->pend.active = true;
    op->pend.sliding = sliding;
    op->pend.dense = keys_are_dense;
    op->pend.deferred = deferred;
    op->pend.buf = b;
    op->pend.C = C;
    op->pend.n = n;
    op->pend.chunk = chunk;
    op->pend.ts = d_ts;
    op->pend.kid = d_kid;
    op->pend.vals = d_vals;
    op->pend.valbm = d_valbm;
    if (deferred && !borrow) {
        /* the caller may release its input buffers once we return: wait for
         * the ring copies (typically tens of µs — the ring slot is idle) */
        HostTimer ht(op, "h_stage");
        event_spin(op->ev_staged[b]);
    }
    return DZ_OK;
}

/* Sections 2-4 of a push: read the staged scalars, lay out window frames,
 * partition + fold, advance the watermark and trigger closes. */
static dz_status process_batch(dz_window_op* op, const dz_window_op::Pend& P);

static dz_status process_pending(dz_window_op* op) {
    if (!op->pend.active) return DZ_OK;
    CHK(op, hipSetDevice(op->device));
    dz_window_op::Pend P = op->pend;
    op->pend.active = false;
    return process_batch(op, P);
}

static dz_status process_batch(dz_window_op* op, const dz_window_op::Pend& P) {
    const bool sliding = P.sliding;
    const bool keys_are_dense = P.dense;
    const int64_t n = P.n, chunk = P.chunk;
    const int C = P.C;
    const int b = P.buf;
    const int64_t* d_ts = P.ts;
    const int32_t* d_kid = P.kid;
    const double* d_vals = P.vals;
    const uint8_t* d_valbm = P.valbm;
    {
        HostTimer ht(op, "h_minmax_sync");
        static const bool dbg = getenv("DZ_SPIN_DEBUG") != nullptr;
        if (dbg) {
            auto t0 = std::chrono::steady_clock::now();
            long it = 0;
            while (hipEventQuery(op->ev_ready[b]) != hipSuccess) {
                std::this_thread::yield();
                it++;
            }
            auto t1 = std::chrono::steady_clock::now();
            fprintf(stderr, "[spin] iters=%ld us=%.1f\n", it,
                    std::chrono::duration<double, std::micro>(t1 - t0).count());
        } else {
            event_spin(op->ev_ready[b]);
        }
    }
    /* compute-stream work below consumes i_stream-written buffers */
    CHK(op, hipStreamWaitEvent(op->stream, op->ev_ready[b], 0));
    drain_events(op, false);
    dz::WinParams wp;
    memset(&wp, 0, sizeof(wp));
    wp.len_ms = op->len_ms;
    wp.slide_ms = op->slide_ms;
    wp.is_sliding = sliding;
    const uint64_t* hs = op->h_scalars + 3 * b;
    int64_t mn = (int64_t)(hs[0] ^ 0x8000000000000000ULL);
    int64_t mx = (int64_t)(hs[1] ^ 0x8000000000000000ULL);
    if (mn < 0) {
        op->err = "negative timestamps not supported (reference SystemTime panics too)";
        return DZ_ERR;
    }
    if (keys_are_dense) {
        int64_t kmax = (int64_t)hs[2];
        op->n_keys = std::max(op->n_keys, kmax + 1);
    }

    if (op->n_keys > op->kcap) {
        if (state_alloc(op, std::max(op->n_keys, op->kcap * 2), op->nslots) != DZ_OK)
            return DZ_ERR;
    }

    /* 2. window ranges for this batch + frame slots */
    std::vector<int64_t> ws(dz::MAX_RANGES), we(dz::MAX_RANGES);
    int64_t nw = dz_debug_windows_for_range(mn, mx, op->len_ms, op->slide_ms,
                                            ws.data(), we.data(), dz::MAX_RANGES);
    if (nw > dz::MAX_RANGES) {
        op->err = "batch spans more than " + std::to_string(dz::MAX_RANGES) +
                  " windows";
        return DZ_ERR;
    }
    std::vector<int32_t> slotmap(nw);
    std::vector<int32_t> reset_list;
    for (int64_t r = 0; r < nw; r++) {
        auto it = op->open.find(ws[r]);
        if (it == op->open.end()) {
            if (op->free_slots.empty()) {
                int32_t cap_slots = std::max(op->max_open, 64);
                if (op->nslots >= cap_slots) {
                    op->err = "too many open windows (cap " +
                              std::to_string(cap_slots) + ")";
                    return DZ_ERR;
                }
                int32_t grow = std::min(std::max(4, op->nslots),
                                        cap_slots - op->nslots);
                if (state_alloc(op, op->kcap, op->nslots + grow) != DZ_OK)
                    return DZ_ERR;
            }
            dz_window_op::FreeSlot fs = op->free_slots.back();
            op->free_slots.pop_back();
            if (fs.ev) { /* emission read of this slot may still be in flight */
                CHK(op, hipStreamWaitEvent(op->stream, fs.ev, 0));
                if (fs.ev_owned) {
                    std::lock_guard<std::mutex> lk(op->e_mtx);
                    op->e_ev_pool.push_back(fs.ev);
                }
            }
            reset_list.push_back(fs.slot);
            op->open[ws[r]] = {we[r], fs.slot, op->rows_seen};
            slotmap[r] = fs.slot;
        } else {
            slotmap[r] = it->second.slot;
        }
    }
    if (!reset_list.empty()) { /* one batched reset launch for all new slots
                                 * (pinned staging: a pageable-source async
                                 * copy stages synchronously at enqueue) */
        int ns = (int)reset_list.size();
        memcpy(op->h_resetlist[b], reset_list.data(), (size_t)ns * 4);
        CHK(op, hipMemcpyAsync(op->d_resetlist, op->h_resetlist[b],
                               (size_t)ns * 4, hipMemcpyHostToDevice,
                               op->stream));
        dz::launch_reset_slots(op->stream, op->d_resetlist, ns, op->kcap,
                               op->s_cnt, op->s_first);
    }
    if (nw > 0) {
        memcpy(op->h_slotmap[b], slotmap.data(), (size_t)nw * 4);
        CHK(op, hipMemcpyAsync(op->d_slotmap, op->h_slotmap[b], nw * 4,
                               hipMemcpyHostToDevice, op->stream));
    }

    wp.s0 = nw > 0 ? ws[0] : 0;
    wp.nw = (int32_t)nw;

    /* 3. partition + fold */
    int64_t expand = sliding
        ? (op->len_ms + op->slide_ms - 1) / op->slide_ms + 1 : 1;
    if (expand > dz::ST_RECORDS) {
        /* one row would overflow the scatter's LDS staging on its own */
        op->err = "window length / hop ratio exceeds " +
                  std::to_string(dz::ST_RECORDS) + " (one row would land in >" +
                  std::to_string(dz::ST_RECORDS) + " windows)";
        return DZ_ERR;
    }
    int64_t nrec_max = n * expand;
    if (nrec_max >= (int64_t)UINT32_MAX) {
        op->err = "batch expands past 2^32 records; push smaller batches";
        return DZ_ERR;
    }
    if (ensure_scratch(op, C, nrec_max) != DZ_OK) return DZ_ERR;
    /* supertile row budget: st_rows * expand staged records must fit the
     * ST_RECORDS LDS staging (a 64-row floor overflowed it for window/hop
     * ratios above 32 — found by the randomized deep matrix) */
    int32_t st_rows = (int32_t)std::max<int64_t>(1, dz::ST_RECORDS / expand);
    const int64_t klocs = op->kcap >> dz::LOG_NB;
    const int64_t khigh = (klocs + 255) >> 8;
    /* Window-subrange self-split: when the batch spans so many windows that
     * neither the direct fold (klocs*nw > GCAP) nor the two-level split
     * (khigh*nw > 256) can cover them in one pass, process the batch in
     * subranges of <= 256/khigh windows each — the partition re-reads the
     * INPUT per subrange (36 B/row) instead of re-reading the expanded
     * record set per (k,w) chunk, which measured 92% of cfg3's GPU time at
     * 8-second batches (85 windows/batch). A tumbling subrange runs as
     * sliding with slide == length: identical window membership PLUS the
     * sliding path's range clamping (rows outside the subrange get
     * multiplicity 0). */
    int64_t wsub = nw;
    if (klocs * nw > dz::FOLD_GCAP && khigh * nw > 256)
        wsub = std::max<int64_t>(1, 256 / khigh);
    const bool split = wsub < nw;
    for (int64_t W0 = 0; W0 < nw; W0 += wsub) {
        const int64_t nws = std::min<int64_t>(wsub, nw - W0);
        dz::WinParams wps = wp;
        wps.s0 = ws[W0];
        wps.nw = (int32_t)nws;
        if (split && !sliding) {
            wps.is_sliding = 1;
            wps.slide_ms = wp.len_ms;
        }
        if (sliding || split) {
            /* subrange (or sliding) reduction on the compute stream; the
             * full-range tumbling histogram was fused into the ingest
             * reduction and is already in d_gofs[b] */
            timed(op, "hist", (double)n * 12, [&] {
                dz::launch_hist(op->stream, d_kid, d_ts, n, chunk, C, wps,
                                op->d_ghist[b], nullptr);
            });
            timed(op, "scan", (double)C * dz::NB * 12, [&] {
                dz::launch_scan(op->stream, op->d_ghist[b], C, op->d_total[b],
                                op->d_base[b], op->d_gofs[b]);
            });
        }
        timed(op, "scatter", (double)n * 24 + (double)nrec_max * 20, [&] {
            dz::launch_scatter(op->stream, d_kid, d_ts, d_vals, d_valbm, n,
                               chunk, C, st_rows, wps, op->d_gofs[b],
                               op->d_grec, (uint32_t)op->rec_cap, op->d_dbg);
        });
        const int64_t gtot = klocs * nws;
        const int64_t nb1 = khigh * nws;
        if (gtot > dz::FOLD_GCAP) {
            /* two-level regime (large keyspaces, e.g. cfg3's 1M keys):
             * split by (kloc>>8, widx), then the second pass splits each
             * bin1 by kloc&255 AND folds the staged bins in place (fused —
             * the split-out records and the fold's re-read of them never
             * touch HBM). nb1 <= 256 by the subrange construction. */
            if (!op->d_b1offs) {
                CHK(op, hipMalloc(&op->d_b1offs, (size_t)dz::NB * 256 * 4));
                CHK(op, hipMalloc(&op->d_b1lens, (size_t)dz::NB * 256 * 4));
            }
            if (nrec_max > op->l2_cap) {
                hipFree(op->d_grec2);
                CHK(op, hipMalloc(&op->d_grec2, (size_t)nrec_max * 16));
                op->l2_cap = nrec_max;
            }
            dz::FoldChunk fc;
            fc.w_lo = 0;
            fc.w_hi = (int32_t)nws;
            fc.k_lo = 0;
            fc.k_hi = (int32_t)klocs;
            fc.kcap = op->kcap;
            fc.row_base = op->rows_seen;
            fc.bin_stride = (int32_t)(nb1 * 256);
            fc.tl_nw = (int32_t)nws;
            timed(op, "regroup", (double)nrec_max * 40, [&] {
                dz::launch_regroup_l1(op->stream, op->d_grec, op->d_base[b],
                                      fc, op->d_b1offs, op->d_b1lens,
                                      op->d_grec2);
            });
            timed(op, "regfold", (double)nrec_max * 24, [&] {
                dz::launch_regroup_l2_fold(op->stream, op->d_grec2,
                                           op->d_base[b], fc, (int)nb1,
                                           op->d_b1offs, op->d_b1lens,
                                           op->d_slotmap + W0, op->s_cnt,
                                           op->s_min, op->s_max, op->s_sum,
                                           op->s_first,
                                           (int64_t)op->nslots * 5 * op->kcap,
                                           op->d_dbg);
            });
        } else {
            for (int64_t k_lo = 0; k_lo < klocs; k_lo += dz::FOLD_GCAP) {
                int32_t nk = (int32_t)std::min<int64_t>(dz::FOLD_GCAP,
                                                        klocs - k_lo);
                int32_t wstep = dz::FOLD_GCAP / nk;
                for (int64_t w_lo = 0; w_lo < nws; w_lo += wstep) {
                    dz::FoldChunk fc;
                    fc.w_lo = (int32_t)w_lo;
                    fc.w_hi = (int32_t)std::min<int64_t>(nws, w_lo + wstep);
                    fc.k_lo = (int32_t)k_lo;
                    fc.k_hi = (int32_t)(k_lo + nk);
                    fc.kcap = op->kcap;
                    fc.row_base = op->rows_seen;
                    fc.bin_stride = dz::FOLD_GCAP;
                    fc.tl_nw = 0;
                    timed(op, "regfold", (double)nrec_max * 24, [&] {
                        dz::launch_regroup_fold(
                            op->stream, op->d_grec, op->d_base[b], fc,
                            op->d_slotmap + W0, op->s_cnt, op->s_min,
                            op->s_max, op->s_sum, op->s_first,
                            (int64_t)op->nslots * 5 * op->kcap, op->d_dbg);
                    });
                }
            }
        }
    }
    /* the fold is the last reader of the per-buffer pipeline state (the
     * scatter reads the staged inputs + d_gofs[b]; regroup/fold read
     * d_base[b]): once it completes, buffer b may be restaged */
    CHK(op, hipEventRecord(op->ev_consumed[b], op->stream));
    op->consumed_valid[b] = true;
    op->batch_seq++;
    op->rows_seen += (uint64_t)n;

    /* 4. watermark (running max of batch minimums, :255-266) + trigger */
    if (!op->has_wm || op->watermark <= mn) {
        op->watermark = mn;
        op->has_wm = true;
    }
    if (op->d_itab && intern_sync_mirror(op) != DZ_OK)
        return DZ_ERR; /* newly interned key bytes -> emission dictionary */
    return trigger_windows(op);
}

/* Synchronous push: flush any pending device-push, stage this batch on the
 * compute stream, process it before returning (host-batch semantics). */
static dz_status push_core(dz_window_op* op, int64_t n, const int64_t* d_ts,
                           const int32_t* d_kid, const double* d_vals,
                           const uint8_t* d_valbm, bool keys_are_dense) {
    if (process_pending(op) != DZ_OK) return DZ_ERR;
    if (n <= 0) return DZ_OK; /* empty batch: reference emits empty (no-op) */
    if (stage_core(op, n, d_ts, d_kid, d_vals, d_valbm, keys_are_dense,
                   /*deferred=*/false) != DZ_OK)
        return DZ_ERR;
    return process_pending(op);
}

static dz_status ensure_zero_kid(dz_window_op* op, int64_t n) {
    if (n > op->zero_cap) {
        hipFree(op->d_zero_kid);
        CHK(op, hipMalloc(&op->d_zero_kid, (size_t)n * 4));
        CHK(op, hipMemsetAsync(op->d_zero_kid, 0, (size_t)n * 4, op->stream));
        /* i_stream reductions read d_zero_kid: make the fill visible there */
        CHK(op, hipStreamSynchronize(op->stream));
        op->zero_cap = n;
    }
    return DZ_OK;
}

extern "C" dz_status dz_window_op_push_device(dz_window_op* op, int64_t n_rows,
                                              const int64_t* d_ts_ms,
                                              const int32_t* d_key_ids,
                                              const double* d_vals) {
    if (!op) return DZ_ERR;
    CHK(op, hipSetDevice(op->device));
    if (op->no_group || !d_key_ids) {
        if (ensure_zero_kid(op, n_rows) != DZ_OK) return DZ_ERR;
        d_key_ids = op->d_zero_kid;
    }
    /* deferred pipeline: STAGE this batch first — its reduction starts on a
     * near-idle device (the previous step's kernels have drained during the
     * caller's gap) and runs ahead of the heavy kernels launched just below
     * — then process the previous push, whose reduction is long visible. */
    if (n_rows <= 0) return process_pending(op);
    int C = (int)std::min<int64_t>(512,
                                   std::max<int64_t>(1, (n_rows + 8191) / 8192));
    if (op->pend.active &&
        (C > op->C_cap || (op->slide_ms == 0 && n_rows > op->rec_cap))) {
        /* scratch growth quiesces the device: drain the pipeline first */
        if (process_pending(op) != DZ_OK) return DZ_ERR;
    }
    dz_window_op::Pend prev = op->pend;
    op->pend.active = false;
    if (stage_core(op, n_rows, d_ts_ms, d_key_ids, d_vals, nullptr,
                   /*keys_are_dense=*/true, /*deferred=*/true) != DZ_OK)
        return DZ_ERR;
    if (prev.active) return process_batch(op, prev);
    return DZ_OK;
}

/* Zero-copy variant: the op reads the caller's buffers directly, so they
 * must stay valid AND unmodified until the NEXT call into the op (the
 * deferred phase reads them then). For callers that own a resident stream
 * buffer — the bench, or an FFI caller holding the batch across its poll
 * loop iteration — this skips the staging copy (the D2D blit competes with
 * the compute kernels for CUs). */
extern "C" dz_status dz_window_op_push_device_borrowed(
        dz_window_op* op, int64_t n_rows, const int64_t* d_ts_ms,
        const int32_t* d_key_ids, const double* d_vals) {
    if (!op) return DZ_ERR;
    CHK(op, hipSetDevice(op->device));
    if (op->no_group || !d_key_ids) {
        if (ensure_zero_kid(op, n_rows) != DZ_OK) return DZ_ERR;
        d_key_ids = op->d_zero_kid;
    }
    /* stage first, then process the previous push — see push_device */
    if (n_rows <= 0) return process_pending(op);
    int C = (int)std::min<int64_t>(512,
                                   std::max<int64_t>(1, (n_rows + 8191) / 8192));
    if (op->pend.active &&
        (C > op->C_cap || (op->slide_ms == 0 && n_rows > op->rec_cap))) {
        if (process_pending(op) != DZ_OK) return DZ_ERR;
    }
    dz_window_op::Pend prev = op->pend;
    op->pend.active = false;
    if (stage_core(op, n_rows, d_ts_ms, d_key_ids, d_vals, nullptr,
                   /*keys_are_dense=*/true, /*deferred=*/true,
                   /*borrow=*/true) != DZ_OK)
        return DZ_ERR;
    if (prev.active) return process_batch(op, prev);
    return DZ_OK;
}

/* ------------------------------------------------------------------ */
/* device utf8 push: GroupValues::intern (grouped_window_agg_stream.rs:512)
 * at device rate — open-address fingerprint table + device string pool;
 * the rest of the pipeline sees dense int32 ids like every other path.  */
/* ------------------------------------------------------------------ */

static dz_status ensure_intern(dz_window_op* op, int64_t n) {
    if (!op->d_itab) {
        uint64_t want = 4 * (uint64_t)std::max<int64_t>(op->kcap, 1);
        uint32_t P = 1u << 16;
        while (P < want && P < (1u << 26)) P <<= 1;
        op->i_pmask = P - 1;
        op->i_idcap = P / 2;
        op->i_poolcap = (uint32_t)std::min<uint64_t>((uint64_t)op->i_idcap * 64,
                                                     1u << 31);
        CHK(op, hipMalloc(&op->d_itab, (size_t)P * 16));
        CHK(op, hipMalloc(&op->d_itab_row, (size_t)P * 4));
        CHK(op, hipMalloc(&op->d_ioff, (size_t)op->i_idcap * 4));
        CHK(op, hipMalloc(&op->d_ilen, (size_t)op->i_idcap * 4));
        CHK(op, hipMalloc(&op->d_ipool, op->i_poolcap));
        CHK(op, hipMalloc(&op->d_ictrs, 8));
        /* fp halves zero (empty), id halves ~0 (unassigned): zero the whole
         * table then set .z/.w lanes via a strided 0xFF fill */
        CHK(op, hipMemsetAsync(op->d_itab, 0, (size_t)P * 16, op->stream));
        CHK(op, hipMemset2DAsync((char*)op->d_itab + 8, 16, 0xFF, 8, P,
                                 op->stream));
        CHK(op, hipMemsetAsync(op->d_ictrs, 0, 8, op->stream));
        /* the intern runs on i_stream: make the init visible there */
        CHK(op, hipStreamSynchronize(op->stream));
    }
    int b = op->next_buf;
    if (n > op->i_kid_cap[b]) {
        quiesce(op);
        hipFree(op->d_ikid[b]);
        CHK(op, hipMalloc(&op->d_ikid[b], (size_t)n * 4));
        op->i_kid_cap[b] = n;
    }
    return DZ_OK;
}

/* pull newly interned key bytes into the host dictionary mirror so the
 * emission workers can format them (append-only ChunkedDict: safe alongside
 * in-flight reads below their snapshots). Runs on the push thread AFTER the
 * batch's scalars are host-visible, i.e. the intern that allocated these
 * ids has completed. */
static dz_status intern_sync_mirror(dz_window_op* op) {
    if (op->mirror_keys >= op->n_keys) return DZ_OK;
    int64_t lo = op->mirror_keys, hi = op->n_keys;
    std::vector<uint32_t> offs(hi - lo), lens(hi - lo);
    CHK(op, hipMemcpy(offs.data(), op->d_ioff + lo, (size_t)(hi - lo) * 4,
                      hipMemcpyDeviceToHost));
    CHK(op, hipMemcpy(lens.data(), op->d_ilen + lo, (size_t)(hi - lo) * 4,
                      hipMemcpyDeviceToHost));
    uint32_t pmax = 0;
    for (size_t i = 0; i < offs.size(); i++)
        pmax = std::max(pmax, offs[i] + lens[i]);
    std::vector<char> pool(pmax);
    if (pmax)
        CHK(op, hipMemcpy(pool.data(), op->d_ipool, pmax,
                          hipMemcpyDeviceToHost));
    for (size_t i = 0; i < offs.size(); i++)
        op->dict_strs.push_back(std::string(pool.data() + offs[i], lens[i]));
    op->mirror_keys = hi;
    return DZ_OK;
}

extern "C" dz_status dz_window_op_push_device_utf8(dz_window_op* op,
        int64_t n_rows, const int64_t* d_ts_ms, const int32_t* d_key_offsets,
        const char* d_key_data, const double* d_vals) {
    if (!op) return DZ_ERR;
    if (op->key_kind != DZ_KEY_UTF8 || op->no_group) {
        op->err = "push_device_utf8 requires key_kind DZ_KEY_UTF8";
        return DZ_ERR;
    }
    if (!op->dict_utf8.empty()) {
        op->err = "cannot mix host utf8 pushes and device utf8 pushes on one "
                  "operator (two dictionaries would assign conflicting ids)";
        return DZ_ERR;
    }
    CHK(op, hipSetDevice(op->device));
    if (n_rows <= 0) return process_pending(op);
    int C = (int)std::min<int64_t>(512,
                                   std::max<int64_t>(1, (n_rows + 8191) / 8192));
    if (op->pend.active &&
        (C > op->C_cap || (op->slide_ms == 0 && n_rows > op->rec_cap))) {
        if (process_pending(op) != DZ_OK) return DZ_ERR;
    }
    if (ensure_intern(op, n_rows) != DZ_OK) return DZ_ERR;
    dz_window_op::Pend prev = op->pend;
    op->pend.active = false;
    /* intern on the ingest stream ahead of the reduction; the output ring
     * buffer is gated like the input staging (the scatter two pushes back
     * was its last reader) */
    int b = op->next_buf;
    if (op->consumed_valid[b])
        CHK(op, hipStreamWaitEvent(op->i_stream, op->ev_consumed[b], 0));
    timed_on(op, op->i_stream, "intern", (double)n_rows * 18, [&] {
        dz::launch_intern(op->i_stream, d_key_offsets, d_key_data, n_rows,
                          op->d_itab, op->d_itab_row,
                          op->i_pmask, op->d_ioff, op->d_ilen, op->d_ipool,
                          op->d_ictrs, op->i_idcap, op->i_poolcap,
                          op->d_ikid[b], op->d_dbg);
    });
    if (stage_core(op, n_rows, d_ts_ms, op->d_ikid[b], d_vals, nullptr,
                   /*keys_are_dense=*/true, /*deferred=*/true,
                   /*borrow=*/true) != DZ_OK)
        return DZ_ERR;
    if (prev.active) return process_batch(op, prev);
    return DZ_OK;
}

extern "C" dz_status dz_generate_utf8(int32_t device, uint64_t seed,
                                      int64_t start_row, int64_t n_rows,
                                      int64_t n_keys, int32_t* d_lens,
                                      const int32_t* d_offsets,
                                      char* d_key_data) {
    if (hipSetDevice(device) != hipSuccess) {
        g_err = "hipSetDevice failed";
        return DZ_ERR;
    }
    dz::launch_gen_utf8(nullptr, seed, start_row, n_rows, n_keys, d_lens,
                        d_offsets, d_key_data);
    if (hipGetLastError() != hipSuccess) {
        g_err = "dz_generate_utf8 launch failed";
        return DZ_ERR;
    }
    return DZ_OK;
}

/* ------------------------------------------------------------------ */
/* host-batch push: dictionary-encode + stage + H2D                    */
/* ------------------------------------------------------------------ */

extern "C" dz_status dz_window_op_push(dz_window_op* op, const dz_batch* batch) {
    if (!op) return DZ_ERR;
    if (!batch) { op->err = "null batch"; return DZ_ERR; }
    int64_t n = batch->n_rows;
    if (n == 0) return DZ_OK;
    int32_t need = std::max(op->ts_col, op->group_col);
    for (auto& a : op->aggs) need = std::max(need, a.input_col);
    if (batch->n_cols <= need) { op->err = "batch has too few columns"; return DZ_ERR; }
    const dz_column& tsc = batch->cols[op->ts_col];
    static const dz_column k_none = {0, nullptr, nullptr, nullptr};
    const dz_column& kc = op->no_group ? k_none : batch->cols[op->group_col];
    const dz_column& vc = batch->cols[op->aggs[0].input_col];
    for (auto& a : op->aggs)
        if (a.input_col != op->aggs[0].input_col) {
            op->err = "all aggregates must share one input column (hot-path shape)";
            return DZ_ERR;
        }

    CHK(op, hipSetDevice(op->device));
    /* staging buffers */
    size_t bm_bytes = (size_t)((n + 7) / 8);
    size_t stage = (size_t)n * (8 + 4 + 8) + bm_bytes + 64;
    if (stage > op->h_stage_cap) {
        if (op->h_stage) hipHostFree(op->h_stage);
        CHK(op, hipHostMalloc((void**)&op->h_stage, stage));
        op->h_stage_cap = stage;
    }
    if (n > op->in_cap) {
        hipFree(op->d_ts); hipFree(op->d_kid); hipFree(op->d_vals); hipFree(op->d_valbm);
        CHK(op, hipMalloc(&op->d_ts, (size_t)n * 8));
        CHK(op, hipMalloc(&op->d_kid, (size_t)n * 4));
        CHK(op, hipMalloc(&op->d_vals, (size_t)n * 8));
        CHK(op, hipMalloc(&op->d_valbm, bm_bytes ? bm_bytes : 1));
        op->in_cap = n;
    }
    int64_t* h_ts = (int64_t*)op->h_stage;
    int32_t* h_kid = (int32_t*)(h_ts + n);
    double* h_vals = (double*)(h_kid + n);
    uint8_t* h_bm = (uint8_t*)(h_vals + n);

    memcpy(h_ts, tsc.data, (size_t)n * 8);
    memcpy(h_vals, vc.data, (size_t)n * 8);
    bool have_bm = vc.validity != nullptr;
    if (have_bm) memcpy(h_bm, vc.validity, bm_bytes);

    /* dictionary encode keys (first-seen dense ids; the per-frame insertion
     * order the reference emits is recovered at emission via first-row sort) */
    if (op->no_group) {
        memset(h_kid, 0, (size_t)n * 4);
        op->n_keys = 1;
    } else if (op->key_kind == DZ_KEY_UTF8) {
        const int32_t* offs = kc.offsets;
        const char* data = (const char*)kc.data;
        if (!offs || !data) { op->err = "utf8 key column needs offsets+data"; return DZ_ERR; }
        if (op->d_itab) {
            op->err = "cannot mix host utf8 pushes and device utf8 pushes on "
                      "one operator (two dictionaries would assign "
                      "conflicting ids)";
            return DZ_ERR;
        }
        for (int64_t i = 0; i < n; i++) {
            std::string s(data + offs[i], data + offs[i + 1]);
            auto it = op->dict_utf8.find(s);
            int32_t id;
            if (it == op->dict_utf8.end()) {
                id = (int32_t)op->dict_strs.size();
                op->dict_utf8.emplace(s, id);
                op->dict_strs.push_back(std::move(s));
            } else {
                id = it->second;
            }
            h_kid[i] = id;
        }
        op->n_keys = (int64_t)op->dict_strs.size();
    } else if (op->key_kind == DZ_KEY_INT64) {
        const int64_t* kv = (const int64_t*)kc.data;
        for (int64_t i = 0; i < n; i++) {
            auto it = op->dict_i64.find(kv[i]);
            int32_t id;
            if (it == op->dict_i64.end()) {
                id = (int32_t)op->dict_vals.size();
                op->dict_i64.emplace(kv[i], id);
                op->dict_vals.push_back(kv[i]);
            } else {
                id = it->second;
            }
            h_kid[i] = id;
        }
        op->n_keys = (int64_t)op->dict_vals.size();
    } else { /* DENSE: caller promises ids in [0, n_keys) */
        const int64_t* kv = (const int64_t*)kc.data;
        for (int64_t i = 0; i < n; i++) h_kid[i] = (int32_t)kv[i];
    }
    if (op->n_keys > op->kcap) {
        if (state_alloc(op, std::max(op->n_keys, op->kcap * 2), op->nslots) != DZ_OK)
            return DZ_ERR;
    }

    CHK(op, hipMemcpyAsync(op->d_ts, h_ts, (size_t)n * 8, hipMemcpyHostToDevice, op->stream));
    CHK(op, hipMemcpyAsync(op->d_kid, h_kid, (size_t)n * 4, hipMemcpyHostToDevice, op->stream));
    CHK(op, hipMemcpyAsync(op->d_vals, h_vals, (size_t)n * 8, hipMemcpyHostToDevice, op->stream));
    if (have_bm)
        CHK(op, hipMemcpyAsync(op->d_valbm, h_bm, bm_bytes, hipMemcpyHostToDevice, op->stream));

    return push_core(op, n, op->d_ts, op->d_kid, op->d_vals,
                     have_bm ? op->d_valbm : nullptr,
                     op->key_kind == DZ_KEY_DENSE_INT64);
}

/* ------------------------------------------------------------------ */
/* poll / finish / watermark / filter / stats                          */
/* ------------------------------------------------------------------ */

extern "C" dz_status dz_window_op_poll(dz_window_op* op, const dz_out_batch** out) {
    if (!op || !out) return DZ_ERR;
    *out = nullptr;
    if (op->pend.active) {
        /* opportunistic, non-blocking: finish a deferred device push whose
         * reduction is already host-visible — but only when the caller would
         * otherwise see nothing at all (keeps a poll-only loop live without
         * perturbing the push/process phase of a pipelined caller, whose
         * next push will process it anyway) */
        bool starved;
        {
            std::lock_guard<std::mutex> lk(op->out_mtx);
            starved = op->outq.empty();
        }
        if (starved) {
            std::lock_guard<std::mutex> lk(op->e_mtx);
            starved = op->e_inflight == 0 && op->e_jobs.empty();
        }
        if (starved) {
            hipSetDevice(op->device);
            if (hipEventQuery(op->ev_ready[op->pend.buf]) == hipSuccess &&
                process_pending(op) != DZ_OK)
                return DZ_ERR;
        }
    }
    int release = -1;
    uint32_t release_gen = 0;
    std::shared_ptr<GroupCtl> release_group;
    {
        std::lock_guard<std::mutex> lk(op->out_mtx);
        if (op->outq.empty()) return DZ_OK;
        if (op->has_current) {
            /* the previous batch's validity ends here: release its slab /
             * group span (zero-copy batches) and recycle the buffer husk */
            release = op->current.hold_slab;
            release_gen = op->current.slab_gen;
            release_group = std::move(op->current.hold_group);
            op->current.hold_slab = -1;
            op->current.packed = nullptr;
            op->current.hold_group.reset();
            if (op->ob_pool.size() < 16)
                op->ob_pool.push_back(std::move(op->current));
        }
        op->current = std::move(op->outq.front());
        op->outq.pop_front();
    }
    if (release >= 0 || release_group) {
        {
            std::lock_guard<std::mutex> lk(op->e_mtx);
            /* a hold that survived a pool regrow references graveyarded
             * memory: its slot number must not re-enter the new pool */
            if (release >= 0 && release_gen == op->e_slab_gen)
                op->e_free.push_back(release);
            if (release_group &&
                release_group->holders.fetch_sub(1) == 1 &&
                release_group->all_built.load()) {
                if (release_group->span >= 0) {
                    op->e_gspan_free.push_back(release_group->span);
                    release_group->span = -1;
                }
                free(release_group->heap);
                release_group->heap = nullptr;
            }
        }
        op->e_cv.notify_all();
    }
    op->has_current = true;
    OutBuf& ob = op->current;
    ob.agg_ptrs.clear();
    if (ob.packed) {
        /* zero-copy batch: aggregate columns (+ dense keys + validity) are
         * views into the pinned packed span the OutBuf holds; group-span
         * slices carry their offset in packed_off (whole-slab batches have
         * packed_off 0) */
        const char* p = ob.packed;
        const size_t ntp = ob.packed_nt;
        const size_t po = ob.packed_off;
        for (size_t a = 0; a < op->aggs.size(); a++) {
            const void* ptr = nullptr;
            switch (op->aggs[a].op) {
                case DZ_AGG_COUNT: ptr = p + ntp * 8 + po * 8; break;
                case DZ_AGG_MIN: ptr = p + ntp * 16 + po * 8; break;
                case DZ_AGG_MAX: ptr = p + ntp * 24 + po * 8; break;
                case DZ_AGG_SUM: ptr = p + ntp * 32 + po * 8; break;
                case DZ_AGG_AVG: ptr = p + ntp * 40 + po * 8; break;
            }
            ob.agg_ptrs.push_back(ptr);
        }
        bool dense_key = !op->no_group && op->key_kind == DZ_KEY_DENSE_INT64;
        ob.view.key_i64 = dense_key ? (const int64_t*)p + po
                                    : (ob.key_i64.empty() ? nullptr
                                                          : ob.key_i64.data());
        ob.view.key_offsets =
            ob.key_offsets.empty() ? nullptr : ob.key_offsets.data();
        ob.view.key_data = ob.key_data.empty() ? nullptr : ob.key_data.data();
        ob.view.agg_valid = (const uint8_t*)(p + ntp * 52 + po);
    } else {
        for (size_t a = 0; a < op->aggs.size(); a++) {
            if (op->aggs[a].op == DZ_AGG_COUNT)
                ob.agg_ptrs.push_back((const void*)ob.agg_i64[a].data());
            else
                ob.agg_ptrs.push_back((const void*)ob.agg_f64[a].data());
        }
        ob.view.key_i64 = ob.key_i64.empty() ? nullptr : ob.key_i64.data();
        ob.view.key_offsets =
            ob.key_offsets.empty() ? nullptr : ob.key_offsets.data();
        ob.view.key_data = ob.key_data.empty() ? nullptr : ob.key_data.data();
        ob.view.agg_valid = ob.agg_valid.data();
    }
    ob.view.agg_cols = ob.agg_ptrs.data();
    ob.view.window_start_ms = ob.wstart.data();
    ob.view.window_end_ms = ob.wend.data();
    *out = &ob.view;
    return DZ_OK;
}

/* read the kernels' bounds-guard cells; a nonzero cell means a kernel
 * detected an out-of-range write it refused to perform — surface it as a
 * hard error (results may be incomplete, never corrupted) */
static dz_status check_dbg(dz_window_op* op) {
    uint32_t cells[4] = {0, 0, 0, 0};
    if (op->d_dbg &&
        hipMemcpy(cells, op->d_dbg, 16, hipMemcpyDeviceToHost) == hipSuccess) {
        for (int i = 0; i < 4; i++)
            if (cells[i]) {
                op->err = "internal: kernel bounds guard tripped (cell " +
                          std::to_string(i) + " = " + std::to_string(cells[i]) +
                          ") — a kernel skipped an out-of-range write";
                return DZ_ERR;
            }
    }
    return DZ_OK;
}

extern "C" dz_status dz_window_op_finish(dz_window_op* op) {
    if (!op) return DZ_ERR;
    CHK(op, hipSetDevice(op->device));
    if (process_pending(op) != DZ_OK) return DZ_ERR;
    int64_t mx = op->has_wm ? op->watermark : INT64_MIN;
    for (auto& kv : op->open) mx = std::max(mx, kv.second.end);
    if (mx != INT64_MIN) {
        op->watermark = mx;
        op->has_wm = true;
    }
    if (trigger_windows(op) != DZ_OK) return DZ_ERR;
    emit_drain(op);
    CHK(op, hipStreamSynchronize(op->stream));
    drain_events(op, true);
    return check_dbg(op);
}

extern "C" dz_status dz_window_op_drain(dz_window_op* op) {
    if (!op) return DZ_ERR;
    if (op->pend.active) {
        CHK(op, hipSetDevice(op->device));
        if (process_pending(op) != DZ_OK) return DZ_ERR;
    }
    emit_drain(op);
    return DZ_OK;
}

extern "C" dz_status dz_window_op_advance_watermark(dz_window_op* op, int64_t wm) {
    if (!op) return DZ_ERR;
    CHK(op, hipSetDevice(op->device));
    /* the external watermark may close windows the deferred batch still has
     * rows for: fold it in first */
    if (process_pending(op) != DZ_OK) return DZ_ERR;
    if (!op->has_wm || op->watermark <= wm) {
        op->watermark = wm;
        op->has_wm = true;
    }
    return trigger_windows(op);
}

extern "C" int64_t dz_window_op_watermark(dz_window_op* op) {
    return (op && op->has_wm) ? op->watermark : INT64_MIN;
}

extern "C" int64_t dz_window_op_open_windows(dz_window_op* op) {
    if (!op) return 0;
    if (op->pend.active) {
        hipSetDevice(op->device);
        process_pending(op);
    }
    return (int64_t)op->open.size();
}

extern "C" dz_status dz_window_op_set_filter(dz_window_op* op, int32_t agg_idx,
                                             int32_t cmp, double literal) {
    if (!op) return DZ_ERR;
    if (op->pend.active) { /* keep filter-applies-at-trigger-time semantics */
        CHK(op, hipSetDevice(op->device));
        if (process_pending(op) != DZ_OK) return DZ_ERR;
    }
    if (agg_idx < 0 || agg_idx >= (int32_t)op->aggs.size()) {
        op->err = "filter agg index out of range";
        return DZ_ERR;
    }
    op->has_filter = true;
    op->f_idx = agg_idx;
    op->f_cmp = cmp;
    op->f_lit = literal;
    return DZ_OK;
}

extern "C" dz_status dz_window_op_kernel_stats(dz_window_op* op,
                                               dz_kernel_stat* out, int32_t cap,
                                               int32_t* n_out) {
    if (!op) return DZ_ERR;
    CHK(op, hipSetDevice(op->device));
    if (process_pending(op) != DZ_OK) return DZ_ERR;
    emit_drain(op);
    CHK(op, hipStreamSynchronize(op->stream));
    drain_events(op, true);
    if (check_dbg(op) != DZ_OK) return DZ_ERR;
    {
        KStatAcc& s = op->stats["h_emit_build"];
        s.launches = op->e_builds.load();
        s.ms = op->e_build_ns.load() / 1e6;
        op->stats["h_emit_zc"].launches = op->e_zc_builds.load();
        op->stats["h_emit_copybuild"].launches = op->e_copy_builds.load();
    }
    int32_t n = 0;
    for (auto& kv : op->stats) {
        if (n >= cap) break;
        snprintf(out[n].name, sizeof(out[n].name), "%s", kv.first.c_str());
        out[n].launches = kv.second.launches;
        /* device-kernel timing is sampled: scale back to an estimated total
         * (host timers have timed == 0 and report exactly) */
        out[n].total_ms = kv.second.timed
            ? kv.second.ms * (double)kv.second.launches / (double)kv.second.timed
            : kv.second.ms;
        out[n].bytes_per_launch_alg = kv.second.last_bytes;
        n++;
    }
    if (n_out) *n_out = n;
    return DZ_OK;
}

/* ------------------------------------------------------------------ */
/* generator + device helpers                                          */
/* ------------------------------------------------------------------ */

static std::string g_util_err;

extern "C" dz_status dz_generate(int32_t device, uint64_t seed, int64_t t0_ms,
                                 int64_t start_row, int64_t n_rows, int64_t n_keys,
                                 int64_t rows_per_ms, int64_t* d_ts, int64_t* d_keys,
                                 int32_t* d_key_ids, double* d_vals) {
    if (hipSetDevice(device) != hipSuccess) { g_err = "hipSetDevice failed"; return DZ_ERR; }
    dz::launch_gen(nullptr, seed, t0_ms, start_row, n_rows, n_keys, rows_per_ms,
                   d_ts, d_keys, d_key_ids, d_vals);
    if (hipGetLastError() != hipSuccess) { g_err = "dz_generate launch failed"; return DZ_ERR; }
    return DZ_OK;
}

extern "C" dz_status dz_device_malloc(int32_t device, size_t bytes, void** out) {
    if (hipSetDevice(device) != hipSuccess || hipMalloc(out, bytes) != hipSuccess) {
        g_err = "hipMalloc failed";
        return DZ_ERR;
    }
    return DZ_OK;
}

extern "C" dz_status dz_device_free(void* p) {
    return hipFree(p) == hipSuccess ? DZ_OK : DZ_ERR;
}

extern "C" dz_status dz_device_synchronize(int32_t device) {
    if (hipSetDevice(device) != hipSuccess || hipDeviceSynchronize() != hipSuccess) {
        g_err = "hipDeviceSynchronize failed";
        return DZ_ERR;
    }
    return DZ_OK;
}

extern "C" dz_status dz_memcpy_d2h(void* dst, const void* src, size_t bytes) {
    return hipMemcpy(dst, src, bytes, hipMemcpyDeviceToHost) == hipSuccess ? DZ_OK : DZ_ERR;
}

extern "C" dz_status dz_memcpy_h2d(void* dst, const void* src, size_t bytes) {
    return hipMemcpy(dst, src, bytes, hipMemcpyHostToDevice) == hipSuccess ? DZ_OK : DZ_ERR;
}

/* ------------------------------------------------------------------ */
/* stream join operator (BASELINE cfg5): inner equi-join on trip_id    */
/* feeding the windowed group-by. The reference lowers .join() to      */
/* DataFusion's inner hash join (datastream.rs:126-175); the streaming */
/* emission discipline here matches oracle.c::orc_join_* exactly:      */
/* matched probe rows emit IN ROW ORDER, unmatched rows buffer in row  */
/* order and re-emit (original order) when their build row arrives.    */
/* Outputs are DEVICE-RESIDENT (ts, dense driver kid, value) columns   */
/* sized for a zero-copy borrowed push into a dz_window_op — the       */
/* join+window pipeline stays on HBM end to end.                       */
/* ------------------------------------------------------------------ */

struct dz_join_op {
    int device = 0;
    std::string err;
    hipStream_t stream = nullptr;
    int64_t* tab_trip = nullptr; /* open-address (trip -> driver), JEMPTY=empty */
    int64_t* tab_drv = nullptr;
    uint64_t p_mask = 0;
    /* unmatched ping-pong buffers (order-preserving) */
    int64_t* u_ts[2] = {};
    int64_t* u_trip[2] = {};
    double* u_val[2] = {};
    int64_t u_cap = 0, u_n = 0;
    int u_cur = 0;
    /* matched output, double-buffered: buffer b stays valid until the
     * SECOND-next push on the join op (a window op's borrowed push reads
     * it until the window's next call) */
    int64_t* o_ts[2] = {};
    int32_t* o_kid[2] = {};
    double* o_val[2] = {};
    int64_t o_cap = 0;
    int o_cur = 0;
    int64_t o_n = 0;
    /* probe scratch */
    int32_t* d_drvtmp = nullptr;
    int64_t drv_cap = 0;
    uint32_t* d_jcnt = nullptr;  /* [512] counts + mbase + ubase + tot */
    uint32_t* h_tot = nullptr;   /* pinned {matched, unmatched} */
    uint32_t* d_dbg = nullptr;
};

extern "C" void dz_join_op_destroy(dz_join_op* op);

#define JCHK(op, call)                                                     \
    do {                                                                   \
        hipError_t e_ = (call);                                            \
        if (e_ != hipSuccess) {                                            \
            (op)->err = std::string(#call) + ": " + hipGetErrorString(e_); \
            return DZ_ERR;                                                 \
        }                                                                  \
    } while (0)

extern "C" dz_join_op* dz_join_op_create(int32_t device, int64_t n_trips_hint) {
    g_err.clear();
    int ndev = 0;
    if (hipGetDeviceCount(&ndev) != hipSuccess || ndev <= device) {
        g_err = "no HIP device available (this operator has no CPU fallback)";
        return nullptr;
    }
    auto* op = new dz_join_op();
    op->device = device;
    uint64_t want = 4 * (uint64_t)std::max<int64_t>(n_trips_hint, 1);
    uint64_t P = 1u << 16;
    while (P < want && P < (1ull << 28)) P <<= 1;
    op->p_mask = P - 1;
    if (hipSetDevice(device) != hipSuccess ||
        hipStreamCreate(&op->stream) != hipSuccess ||
        hipMalloc(&op->tab_trip, P * 8) != hipSuccess ||
        hipMalloc(&op->tab_drv, P * 8) != hipSuccess ||
        hipMalloc(&op->d_jcnt, (3 * 512 + 2) * 4) != hipSuccess ||
        hipMalloc(&op->d_dbg, 16) != hipSuccess ||
        hipHostMalloc((void**)&op->h_tot, 8) != hipSuccess) {
        g_err = "join op allocation failed";
        dz_join_op_destroy(op); /* frees whatever was allocated */
        return nullptr;
    }
    hipMemset(op->d_dbg, 0, 16);
    dz::launch_fill_i64(op->stream, op->tab_trip, (int64_t)P, INT64_MIN);
    if (hipStreamSynchronize(op->stream) != hipSuccess) {
        g_err = "join table init failed";
        dz_join_op_destroy(op);
        return nullptr;
    }
    return op;
}

extern "C" void dz_join_op_destroy(dz_join_op* op) {
    if (!op) return;
    hipSetDevice(op->device);
    if (op->stream) hipStreamSynchronize(op->stream);
    hipFree(op->tab_trip); hipFree(op->tab_drv);
    for (int i = 0; i < 2; i++) {
        hipFree(op->u_ts[i]); hipFree(op->u_trip[i]); hipFree(op->u_val[i]);
        hipFree(op->o_ts[i]); hipFree(op->o_kid[i]); hipFree(op->o_val[i]);
    }
    hipFree(op->d_drvtmp); hipFree(op->d_jcnt); hipFree(op->d_dbg);
    if (op->h_tot) hipHostFree(op->h_tot);
    if (op->stream) hipStreamDestroy(op->stream);
    delete op;
}

extern "C" const char* dz_join_last_error(dz_join_op* op) {
    if (!op) return g_err.empty() ? nullptr : g_err.c_str();
    return op->err.empty() ? nullptr : op->err.c_str();
}

static dz_status join_ensure_probe(dz_join_op* op, int64_t n) {
    /* the buffer re-probe on a build push marks u_n rows, not n */
    int64_t need_drv = std::max(n, op->u_n);
    if (need_drv > op->drv_cap) {
        JCHK(op, hipStreamSynchronize(op->stream));
        hipFree(op->d_drvtmp);
        JCHK(op, hipMalloc(&op->d_drvtmp, (size_t)need_drv * 4));
        op->drv_cap = need_drv;
    }
    /* output must hold this probe's matches OR a full buffer re-probe.
     * GROWTH frees buffers a downstream window op may still be reading
     * under the borrowed-push contract (its streams, not ours): take a
     * full-device barrier before freeing. */
    int64_t need_o = std::max(n, op->u_n);
    if (need_o > op->o_cap) {
        JCHK(op, hipDeviceSynchronize());
        for (int i = 0; i < 2; i++) {
            hipFree(op->o_ts[i]); hipFree(op->o_kid[i]); hipFree(op->o_val[i]);
            JCHK(op, hipMalloc(&op->o_ts[i], (size_t)need_o * 8));
            JCHK(op, hipMalloc(&op->o_kid[i], (size_t)need_o * 4));
            JCHK(op, hipMalloc(&op->o_val[i], (size_t)need_o * 8));
        }
        op->o_cap = need_o;
    }
    int64_t need_u = op->u_n + n;
    if (need_u > op->u_cap) {
        JCHK(op, hipStreamSynchronize(op->stream));
        int64_t cap = std::max<int64_t>(need_u, op->u_cap * 2);
        for (int i = 0; i < 2; i++) {
            int64_t* nts; int64_t* ntr; double* nv;
            JCHK(op, hipMalloc(&nts, (size_t)cap * 8));
            JCHK(op, hipMalloc(&ntr, (size_t)cap * 8));
            JCHK(op, hipMalloc(&nv, (size_t)cap * 8));
            if (op->u_n > 0 && i == op->u_cur) {
                JCHK(op, hipMemcpy(nts, op->u_ts[i], (size_t)op->u_n * 8,
                                   hipMemcpyDeviceToDevice));
                JCHK(op, hipMemcpy(ntr, op->u_trip[i], (size_t)op->u_n * 8,
                                   hipMemcpyDeviceToDevice));
                JCHK(op, hipMemcpy(nv, op->u_val[i], (size_t)op->u_n * 8,
                                   hipMemcpyDeviceToDevice));
            }
            hipFree(op->u_ts[i]); hipFree(op->u_trip[i]); hipFree(op->u_val[i]);
            op->u_ts[i] = nts; op->u_trip[i] = ntr; op->u_val[i] = nv;
        }
        op->u_cap = cap;
    }
    return DZ_OK;
}

static dz_status join_check_dbg(dz_join_op* op) {
    uint32_t cells[4] = {0, 0, 0, 0};
    if (hipMemcpy(cells, op->d_dbg, 16, hipMemcpyDeviceToHost) == hipSuccess &&
        cells[3] == 8) {
        op->err = "join build table full — raise n_trips_hint";
        return DZ_ERR;
    }
    return DZ_OK;
}

extern "C" dz_status dz_join_op_push_build(dz_join_op* op, int64_t n,
                                           const int64_t* d_trip_ids,
                                           const int64_t* d_driver_ids) {
    if (!op) return DZ_ERR;
    JCHK(op, hipSetDevice(op->device));
    op->o_n = 0;
    if (n > 0)
        dz::launch_join_build(op->stream, d_trip_ids, d_driver_ids, n,
                              op->tab_trip, op->tab_drv, op->p_mask, op->d_dbg);
    if (op->u_n > 0) {
        /* re-probe the unmatched buffer: newly matched rows emit in their
         * original buffered order; the rest compact into the other buffer */
        if (join_ensure_probe(op, 0) != DZ_OK) return DZ_ERR;
        const int src = op->u_cur, dst = src ^ 1;
        int C = (int)std::min<int64_t>(512,
                                       std::max<int64_t>(1, (op->u_n + 8191) / 8192));
        int64_t chunk = (op->u_n + C - 1) / C;
        const int nxt = op->o_cur ^ 1;
        dz::launch_join_probe(op->stream, op->u_ts[src], op->u_trip[src],
                              op->u_val[src], op->u_n, chunk, C, op->tab_trip,
                              op->tab_drv, op->p_mask, op->d_drvtmp,
                              op->d_jcnt, op->d_jcnt + 512, op->d_jcnt + 1024,
                              op->d_jcnt + 1536, op->o_ts[nxt], op->o_kid[nxt],
                              op->o_val[nxt], 0, op->u_ts[dst],
                              op->u_trip[dst], op->u_val[dst]);
        JCHK(op, hipMemcpyAsync(op->h_tot, op->d_jcnt + 1536, 8,
                                hipMemcpyDeviceToHost, op->stream));
        JCHK(op, hipStreamSynchronize(op->stream));
        op->o_cur = nxt;
        op->o_n = op->h_tot[0];
        op->u_n = op->h_tot[1];
        op->u_cur = dst;
    } else {
        JCHK(op, hipStreamSynchronize(op->stream));
    }
    return join_check_dbg(op);
}

extern "C" dz_status dz_join_op_push_probe(dz_join_op* op, int64_t n,
                                           const int64_t* d_ts_ms,
                                           const int64_t* d_trip_ids,
                                           const double* d_vals) {
    if (!op) return DZ_ERR;
    JCHK(op, hipSetDevice(op->device));
    op->o_n = 0;
    if (n <= 0) return DZ_OK;
    if (join_ensure_probe(op, n) != DZ_OK) return DZ_ERR;
    int C = (int)std::min<int64_t>(512, std::max<int64_t>(1, (n + 8191) / 8192));
    int64_t chunk = (n + C - 1) / C;
    const int nxt = op->o_cur ^ 1;
    const int ub = op->u_cur;
    dz::launch_join_probe(op->stream, d_ts_ms, d_trip_ids, d_vals, n, chunk, C,
                          op->tab_trip, op->tab_drv, op->p_mask, op->d_drvtmp,
                          op->d_jcnt, op->d_jcnt + 512, op->d_jcnt + 1024,
                          op->d_jcnt + 1536, op->o_ts[nxt], op->o_kid[nxt],
                          op->o_val[nxt], op->u_n, op->u_ts[ub],
                          op->u_trip[ub], op->u_val[ub]);
    JCHK(op, hipMemcpyAsync(op->h_tot, op->d_jcnt + 1536, 8,
                            hipMemcpyDeviceToHost, op->stream));
    JCHK(op, hipStreamSynchronize(op->stream));
    op->o_cur = nxt;
    op->o_n = op->h_tot[0];
    op->u_n += op->h_tot[1];
    return join_check_dbg(op);
}

/* matched output of the LAST push (device-resident). Valid until the
 * SECOND-next push on this join op (double-buffered), which covers a
 * dz_window_op borrowed push consuming it across one pipeline step. */
extern "C" dz_status dz_join_op_matches(dz_join_op* op, int64_t* n_out,
                                        const int64_t** d_ts,
                                        const int32_t** d_kid,
                                        const double** d_vals) {
    if (!op || !n_out) return DZ_ERR;
    *n_out = op->o_n;
    if (d_ts) *d_ts = op->o_ts[op->o_cur];
    if (d_kid) *d_kid = op->o_kid[op->o_cur];
    if (d_vals) *d_vals = op->o_val[op->o_cur];
    return DZ_OK;
}

extern "C" int64_t dz_join_op_unmatched(dz_join_op* op) {
    return op ? op->u_n : 0;
}

/* ------------------------------------------------------------------ */
/* JSON ingest decoder: device-side replacement for the  */
/* reference's serde_json decode of Kafka payload bytes                */
/* (formats/decoders/json.rs:23-46 via kafka_stream_read.rs:165-296).  */
/* Newline-delimited records -> (ts int64, utf8 key column, f64 value) */
/* device columns shaped for dz_window_op_push_device_utf8. Documented */
/* subset (flagged loudly otherwise): no escape sequences inside the   */
/* schema fields' strings/names, numeric literals within the exact     */
/* Clinger fast path (<=15 significant digits, |decimal exp| <= 22).   */
/* ------------------------------------------------------------------ */

struct dz_json_decoder {
    int device = 0;
    std::string err;
    hipStream_t stream = nullptr;
    dz::JsonFields jf;
    uint32_t* d_cnt = nullptr;   /* [512] + base [512] + tot [1] + ktot [1] */
    uint32_t* h_tot = nullptr;   /* pinned */
    uint32_t* d_dbg = nullptr;
    uint32_t* d_blocksum = nullptr;
    int64_t bs_cap = 0;
    /* per-decode outputs, double-buffered (valid until the second-next
     * decode: a utf8 borrowed push consumes them across one step) */
    int64_t* d_recoff[2] = {};
    int64_t* o_ts[2] = {};
    int64_t* o_kbeg[2] = {};
    int32_t* o_klen[2] = {};
    double* o_val[2] = {};
    int32_t* o_koff[2] = {};
    char* o_kdata[2] = {};
    int64_t rec_cap = 0, kdata_cap = 0;
    int cur = 0;
    int64_t n_rec = 0;
};

extern "C" void dz_json_decoder_destroy(dz_json_decoder* d);

extern "C" dz_json_decoder* dz_json_decoder_create(int32_t device,
                                                   const char* ts_field,
                                                   const char* key_field,
                                                   const char* val_field) {
    g_err.clear();
    int ndev = 0;
    if (hipGetDeviceCount(&ndev) != hipSuccess || ndev <= device) {
        g_err = "no HIP device available (this operator has no CPU fallback)";
        return nullptr;
    }
    if (!ts_field || !key_field || !val_field ||
        strlen(ts_field) > 31 || strlen(key_field) > 31 ||
        strlen(val_field) > 31) {
        g_err = "field names must be 1..31 bytes";
        return nullptr;
    }
    auto* d = new dz_json_decoder();
    d->device = device;
    snprintf(d->jf.ts_name, 32, "%s", ts_field);
    snprintf(d->jf.key_name, 32, "%s", key_field);
    snprintf(d->jf.val_name, 32, "%s", val_field);
    d->jf.ts_len = (int32_t)strlen(ts_field);
    d->jf.key_len = (int32_t)strlen(key_field);
    d->jf.val_len = (int32_t)strlen(val_field);
    if (hipSetDevice(device) != hipSuccess ||
        hipStreamCreate(&d->stream) != hipSuccess ||
        hipMalloc(&d->d_cnt, (512 + 512 + 2) * 4) != hipSuccess ||
        hipMalloc(&d->d_dbg, 16) != hipSuccess ||
        hipHostMalloc((void**)&d->h_tot, 8) != hipSuccess) {
        g_err = "json decoder allocation failed";
        dz_json_decoder_destroy(d); /* frees whatever was allocated */
        return nullptr;
    }
    hipMemset(d->d_dbg, 0, 16);
    return d;
}

extern "C" void dz_json_decoder_destroy(dz_json_decoder* d) {
    if (!d) return;
    hipSetDevice(d->device);
    if (d->stream) hipStreamSynchronize(d->stream);
    hipFree(d->d_cnt); hipFree(d->d_dbg); hipFree(d->d_blocksum);
    for (int i = 0; i < 2; i++) {
        hipFree(d->d_recoff[i]); hipFree(d->o_ts[i]); hipFree(d->o_kbeg[i]);
        hipFree(d->o_klen[i]); hipFree(d->o_val[i]); hipFree(d->o_koff[i]);
        hipFree(d->o_kdata[i]);
    }
    if (d->h_tot) hipHostFree(d->h_tot);
    if (d->stream) hipStreamDestroy(d->stream);
    delete d;
}

extern "C" const char* dz_json_decoder_last_error(dz_json_decoder* d) {
    if (!d) return g_err.empty() ? nullptr : g_err.c_str();
    return d->err.empty() ? nullptr : d->err.c_str();
}

#define DCHK(op, call)                                                     \
    do {                                                                   \
        hipError_t e_ = (call);                                            \
        if (e_ != hipSuccess) {                                            \
            (op)->err = std::string(#call) + ": " + hipGetErrorString(e_); \
            return DZ_ERR;                                                 \
        }                                                                  \
    } while (0)

extern "C" dz_status dz_json_decode(dz_json_decoder* d, const char* d_bytes,
                                    int64_t n_bytes) {
    if (!d) return DZ_ERR;
    DCHK(d, hipSetDevice(d->device));
    d->n_rec = 0;
    if (n_bytes <= 0) return DZ_OK;
    int C = (int)std::min<int64_t>(512,
                                   std::max<int64_t>(1, (n_bytes + 8191) / 8192));
    int64_t chunk = (n_bytes + C - 1) / C;
    /* tail byte decides whether bytes after the last newline are a record */
    char tail = 0;
    DCHK(d, hipMemcpy(&tail, d_bytes + n_bytes - 1, 1, hipMemcpyDeviceToHost));
    dz::launch_json_count(d->stream, d_bytes, n_bytes, C, chunk,
                          tail != '\n' ? 1 : 0, d->d_cnt, d->d_cnt + 512,
                          d->d_cnt + 1024);
    DCHK(d, hipMemcpyAsync(d->h_tot, d->d_cnt + 1024, 4,
                           hipMemcpyDeviceToHost, d->stream));
    DCHK(d, hipStreamSynchronize(d->stream));
    const int64_t nrec = d->h_tot[0];
    if (nrec == 0) return DZ_OK;
    if (nrec > d->rec_cap || n_bytes > d->kdata_cap) {
        /* growth frees columns a downstream window op may still read
         * under the borrowed-push contract: full-device barrier first */
        DCHK(d, hipDeviceSynchronize());
        int64_t rc = std::max(nrec, d->rec_cap);
        int64_t kc = std::max(n_bytes, d->kdata_cap);
        for (int i = 0; i < 2; i++) {
            hipFree(d->d_recoff[i]); hipFree(d->o_ts[i]); hipFree(d->o_kbeg[i]);
            hipFree(d->o_klen[i]); hipFree(d->o_val[i]); hipFree(d->o_koff[i]);
            hipFree(d->o_kdata[i]);
            DCHK(d, hipMalloc(&d->d_recoff[i], (size_t)rc * 8));
            DCHK(d, hipMalloc(&d->o_ts[i], (size_t)rc * 8));
            DCHK(d, hipMalloc(&d->o_kbeg[i], (size_t)rc * 8));
            DCHK(d, hipMalloc(&d->o_klen[i], (size_t)rc * 4));
            DCHK(d, hipMalloc(&d->o_val[i], (size_t)rc * 8));
            DCHK(d, hipMalloc(&d->o_koff[i], ((size_t)rc + 1) * 4));
            DCHK(d, hipMalloc(&d->o_kdata[i], (size_t)kc));
        }
        d->rec_cap = rc;
        d->kdata_cap = kc;
    }
    const int64_t nb = (nrec + 4095) / 4096;
    if (nb > d->bs_cap) {
        hipFree(d->d_blocksum);
        DCHK(d, hipMalloc(&d->d_blocksum, (size_t)nb * 4));
        d->bs_cap = nb;
    }
    const int b = d->cur ^ 1;
    dz::launch_json_parse(d->stream, d_bytes, n_bytes, C, chunk, d->jf,
                          d->d_cnt + 512, d->d_cnt + 1024, d->d_recoff[b],
                          nrec, d->o_ts[b], d->o_kbeg[b], d->o_klen[b],
                          d->o_val[b], d->d_blocksum, d->d_cnt + 1025,
                          d->o_koff[b], d->o_kdata[b], d->d_dbg);
    DCHK(d, hipStreamSynchronize(d->stream));
    uint32_t cells[4] = {0, 0, 0, 0};
    DCHK(d, hipMemcpy(cells, d->d_dbg, 16, hipMemcpyDeviceToHost));
    if (cells[2]) {
        d->err = cells[2] == 2
                     ? "numeric literal outside the exact parse subset "
                       "(<=15 significant digits, |decimal exponent| <= 22)"
                     : (cells[2] == 1 ? "malformed JSON record"
                                      : "record missing a schema field");
        return DZ_ERR;
    }
    d->cur = b;
    d->n_rec = nrec;
    return DZ_OK;
}

/* the LAST decode's columns (device pointers, valid until the second-next
 * decode) — exactly the shape dz_window_op_push_device_utf8 consumes */
extern "C" dz_status dz_json_decoder_batch(dz_json_decoder* d, int64_t* n_out,
                                           const int64_t** d_ts,
                                           const int32_t** d_key_offsets,
                                           const char** d_key_data,
                                           const double** d_vals) {
    if (!d || !n_out) return DZ_ERR;
    *n_out = d->n_rec;
    if (d_ts) *d_ts = d->o_ts[d->cur];
    if (d_key_offsets) *d_key_offsets = d->o_koff[d->cur];
    if (d_key_data) *d_key_data = d->o_kdata[d->cur];
    if (d_vals) *d_vals = d->o_val[d->cur];
    return DZ_OK;
}

extern "C" dz_status dz_generate_json(int32_t device, uint64_t seed,
                                      int64_t t0_ms, int64_t start_row,
                                      int64_t n_rows, int64_t n_keys,
                                      int64_t rows_per_ms, int32_t* d_lens,
                                      const int64_t* d_offsets, char* d_data) {
    if (hipSetDevice(device) != hipSuccess) {
        g_err = "hipSetDevice failed";
        return DZ_ERR;
    }
    dz::launch_gen_json(nullptr, seed, t0_ms, start_row, n_rows, n_keys,
                        rows_per_ms, d_lens, d_offsets, d_data);
    if (hipGetLastError() != hipSuccess) {
        g_err = "dz_generate_json launch failed";
        return DZ_ERR;
    }
    return DZ_OK;
}
