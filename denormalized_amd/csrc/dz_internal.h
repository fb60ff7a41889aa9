/* dz_internal.h — shared internal declarations between the engine host code
 * (window_op.cpp) and the CDNA4 kernels (kernels.hip). Not part of the C ABI.
 */
#pragma once
#include <cstdint>
#include <hip/hip_runtime.h>

namespace dz {

/* Partition geometry. Rows are bucketed by (key_id & (NB-1)): with dense
 * dictionary ids this balances buckets to ±1 key. One wave folds one bucket,
 * so NB also sets fold parallelism (4096 waves = 16 waves/CU on 256 CUs). */
constexpr int NB = 512;
constexpr int LOG_NB = 9;
constexpr int BLOCK = 256;          /* 4 waves */
constexpr int WAVES_PER_BLOCK = BLOCK / 64;
constexpr int MAX_RANGES = 4096;    /* window frames touched by one batch */
constexpr int ST_RECORDS = 2048;    /* scatter staging records per supertile */

/* meta word packed per record: kloc (16b) | widx (12b) | valid (1b) */
constexpr uint32_t META_KLOC_MASK = 0xFFFFu;
constexpr int META_WIDX_SHIFT = 16;
constexpr uint32_t META_WIDX_MASK = 0xFFFu;
constexpr int META_VALID_SHIFT = 28;

struct WinParams {
    int64_t s0;          /* first window start of this batch's range list */
    int64_t len_ms;
    int64_t slide_ms;    /* 0 => tumbling */
    int32_t nw;          /* ranges in this batch */
    int32_t is_sliding;
};

struct FoldChunk {
    int32_t w_lo, w_hi;  /* widx range [w_lo, w_hi) this launch folds */
    int32_t k_lo, k_hi;  /* kloc range [k_lo, k_hi) */
    int64_t kcap;        /* state key capacity (multiple of NB) */
    uint64_t row_base;   /* global row id of this batch's row 0: first-seen
                          * = row_base + rowidx, a monotone stream-lifetime
                          * counter (same order as the old batch<<32|row
                          * composite, but window-rebased sort keys stay
                          * small forever — fixed radix pass count) */
    int32_t bin_stride;  /* binoffs/binlens row stride per bucket */
    int32_t tl_nw;       /* two-level mode: windows per batch (0 = off) */
};

/* Launch wrappers implemented in kernels.hip. All run on `stream`;
 * n = rows in batch, C = partition chunks (hist/scatter grid). */
void launch_gen(hipStream_t stream, uint64_t seed, int64_t t0, int64_t start_row,
                int64_t n, int64_t nkeys, int64_t rows_per_ms,
                int64_t* d_ts, int64_t* d_keys, int32_t* d_kid, double* d_vals);

/* device utf8 intern (GroupValues::intern analog at device rate) */
void launch_intern(hipStream_t stream, const int32_t* d_offs, const char* d_data,
                   int64_t n, uint4* tab /* {fp, id, len|off<<6} slots */,
                   uint32_t* tab_row, uint32_t p_mask, uint32_t* id_off,
                   uint32_t* id_len, char* pool, uint32_t* ctrs,
                   uint32_t id_cap, uint32_t pool_cap, int32_t* out_kid,
                   uint32_t* d_dbg);

/* synthetic utf8 key generator ("sensor_{k}"): lens pass and/or fill pass */
void launch_gen_utf8(hipStream_t stream, uint64_t seed, int64_t start_row,
                     int64_t n, int64_t nkeys, int32_t* d_lens,
                     const int32_t* d_offs, char* d_data);

/* d_scalars: [0]=min ts (order-mapped u64), [1]=max ts, [2]=max kid (u64).
 * Host must pre-set [0]=UINT64_MAX, [1]=0, [2]=0. */
void launch_minmax(hipStream_t stream, const int64_t* d_ts, const int32_t* d_kid,
                   int64_t n, uint64_t* d_scalars);

void launch_hist(hipStream_t stream, const int32_t* d_kid, const int64_t* d_ts,
                 int64_t n, int64_t chunk, int C, const WinParams& wp,
                 uint32_t* d_ghist, uint64_t* d_scalars /*fused minmax or null*/);

constexpr int SCAN_SSPLIT = 32; /* d_psum is SCAN_SSPLIT * NB u32 */
void launch_scan(hipStream_t stream, const uint32_t* d_ghist, int C,
                 uint32_t* d_psum, uint32_t* d_bucket_base /*NB+1*/,
                 uint32_t* d_gofs);

/* record payloads are 16 B uint4s: {val lo, val hi, rowidx, meta}; the
 * rowidx word carries the validity bit (bit 31) in FOLD-stage records and
 * .w carries the meta word (kloc|widx|valid) — there is no separate meta
 * array */
void launch_scatter(hipStream_t stream, const int32_t* d_kid, const int64_t* d_ts,
                    const double* d_vals, const uint8_t* d_validity /*bitmap|null*/,
                    int64_t n, int64_t chunk, int C, int32_t st_rows,
                    const WinParams& wp, const uint32_t* d_gofs, uint4* d_grec,
                    uint32_t rec_limit, uint32_t* d_dbg /*bounds-guard cells*/);

constexpr int FOLD_GCAP = 256; /* groups per bucket per fold chunk */

/* fused stable-split + fold: per-(window,key) bins staged in LDS fold
 * straight into register accumulators seeded from the window-slot slabs —
 * no reordered records in HBM */
void launch_regroup_fold(hipStream_t stream, const uint4* d_grec,
                         const uint32_t* d_bucket_base,
                         const FoldChunk& fc, const int32_t* d_slot_of_widx,
                         uint64_t* s_cnt, double* s_min, double* s_max,
                         double* s_sum, uint64_t* s_first, int64_t slab_cells,
                         uint32_t* d_dbg);

void launch_regroup_l1(hipStream_t stream, const uint4* d_grec,
                       const uint32_t* d_bucket_base,
                       const FoldChunk& fc, uint32_t* d_b1offs,
                       uint32_t* d_b1lens, uint4* d_grec2);

void launch_regroup_l2_fold(hipStream_t stream, const uint4* d_grec2,
                            const uint32_t* d_bucket_base,
                            const FoldChunk& fc, int nb1,
                            const uint32_t* d_b1offs, const uint32_t* d_b1lens,
                            const int32_t* d_slot_of_widx, uint64_t* s_cnt,
                            double* s_min, double* s_max, double* s_sum,
                            uint64_t* s_first, int64_t slab_cells,
                            uint32_t* d_dbg);

struct EGatherSlots {
    int32_t s[16];
    uint64_t base[16]; /* per-close window-open row base: the sort key is
                        * first - base, bounded by the window's row span */
};

struct EmitFilter {
    int32_t on;      /* 0 = no filter */
    int32_t field;   /* 0 cnt, 1 min, 2 max, 3 sum, 4 avg */
    int32_t cmp;     /* 0 < 1 <= 2 > 3 >= 4 == 5 != */
    double lit;
};

/* device-side emission at window close, two stages on the copy stream:
 * slabread = compact touched groups + gather aggregate columns + filter
 * flags (after which the window slot is reusable); sort = stable radix by
 * first-seen row, leaving the sorted compact-index permutation in `skid`. */
void launch_emission_slabread(hipStream_t stream, const uint64_t* slab_first,
                              const uint64_t* slab_cnt, const double* slab_min,
                              const double* slab_max, const double* slab_sum,
                              uint64_t base, int64_t K, uint64_t* ekeys, uint32_t* ekid,
                              uint64_t* fkeys, uint32_t* fkid, uint32_t* fiota,
                              uint32_t* counter, uint32_t* counter2,
                              const EmitFilter& ef, uint64_t* ocnt, double* omin,
                              double* omax, double* osum, double* oavg,
                              uint8_t* oflags);
void launch_emission_sort(hipStream_t stream, int64_t K, uint64_t* fkeys,
                          uint64_t* skeys, uint32_t* fiota, uint32_t* okid,
                          uint32_t* counter2, uint32_t* rhist, uint32_t* roffs,
                          uint64_t max_key /* host-known bound on `first` */);
/* group-batched device emission (<=16 closes in one chain; see
 * kernels.hip §GROUP-BATCHED): read phase (slot slabs release after it),
 * then sort+pack */
void launch_emission_group_read(hipStream_t stream, const uint64_t* s_base,
                                int64_t stride_u64, const EGatherSlots& slots,
                                int gcount, int64_t K, int64_t kcap,
                                const EmitFilter& ef, int cshift,
                                uint64_t* gkeys, uint32_t* gkid,
                                uint32_t* giota, uint64_t* gcnt_col,
                                double* gmin, double* gmax, double* gsum,
                                double* gavg, uint8_t* gflags, uint32_t* gtot,
                                uint32_t* gcnt);
void launch_emission_group_sort(hipStream_t stream, int64_t elem_cap,
                                int gcount, int cshift, uint64_t max_first,
                                uint64_t* gkeys, uint64_t* gskeys,
                                uint32_t* gkid, uint32_t* gokid,
                                uint32_t* giota, uint64_t* gcnt_col,
                                double* gmin, double* gmax, double* gsum,
                                double* gavg, uint8_t* gflags, uint32_t* gtot,
                                uint32_t* rhist, uint32_t* roffs, char* pout);

void launch_esort_small(hipStream_t stream, uint64_t* fkeys, uint64_t* skeys,
                        uint32_t* fiota, uint32_t* okid, uint32_t* counter2,
                        uint64_t max_key);
constexpr int EMIT_RCHUNK = 4096;
constexpr int EMIT_RBINS = 2048;

void launch_reset_slots(hipStream_t stream, const int32_t* d_slots, int ns,
                        int64_t kcap, uint64_t* s_cnt, uint64_t* s_first);

/* host-path emission: pack up to 16 closing window slots' state slabs into
 * one contiguous staging area (one launch + ONE big D2H replaces a per-close
 * copy on the push thread; slot ids travel by value in the kernel args) */
void launch_egather_slabs(hipStream_t stream, const uint64_t* s_base,
                          int64_t stride_u64, EGatherSlots slots, int gcount,
                          uint64_t* out);
void launch_arm_scalars(hipStream_t stream, uint64_t* scalars);

void launch_fill_i64(hipStream_t stream, int64_t* d_p, int64_t n, int64_t v);

/* JSON ingest (kernels.hip §JSON ingest): newline count/scan, then
 * emit+parse+key compaction with the record count known host-side */
struct JsonFields {
    char ts_name[32];
    char key_name[32];
    char val_name[32];
    int32_t ts_len, key_len, val_len;
};
void launch_json_count(hipStream_t stream, const char* d_buf, int64_t nbytes,
                       int C, int64_t chunk, int32_t tail_record,
                       uint32_t* d_cnt, uint32_t* d_base, uint32_t* d_tot);
void launch_json_parse(hipStream_t stream, const char* d_buf, int64_t nbytes,
                       int C, int64_t chunk, const JsonFields& jf,
                       const uint32_t* d_base, const uint32_t* d_tot,
                       int64_t* d_recoff, int64_t nrec, int64_t* o_ts,
                       int64_t* o_kbeg, int32_t* o_klen, double* o_val,
                       uint32_t* d_blocksum, uint32_t* d_ktot, int32_t* o_koff,
                       char* o_kdata, uint32_t* d_dbg);
void launch_gen_json(hipStream_t stream, uint64_t seed, int64_t t0,
                     int64_t start_row, int64_t n, int64_t nkeys,
                     int64_t rows_per_ms, int32_t* d_lens,
                     const int64_t* d_offs, char* d_data);

/* stream join (cfg5): open-address trip->driver build table + stable
 * order-preserving probe/emit (kernels.hip §stream join) */
void launch_join_build(hipStream_t stream, const int64_t* d_trips,
                       const int64_t* d_drivers, int64_t n, int64_t* tab_trip,
                       int64_t* tab_drv, uint64_t p_mask, uint32_t* d_dbg);
void launch_join_probe(hipStream_t stream, const int64_t* d_ts,
                       const int64_t* d_trips, const double* d_vals, int64_t n,
                       int64_t chunk, int C, const int64_t* tab_trip,
                       const int64_t* tab_drv, uint64_t p_mask,
                       int32_t* d_drvtmp, uint32_t* d_jcnt, uint32_t* d_mbase,
                       uint32_t* d_ubase, uint32_t* d_tot, int64_t* o_ts,
                       int32_t* o_kid, double* o_val, int64_t ubuf_base,
                       int64_t* u_ts, int64_t* u_trip, double* u_val);
void launch_zero_counters(hipStream_t stream, uint32_t* two_u32);
void launch_emission_permute(hipStream_t stream, int64_t K,
                             const uint32_t* counter2, const uint32_t* sidx,
                             const uint32_t* fkid, const uint64_t* ocnt,
                             const double* omin, const double* omax,
                             const double* osum, const double* oavg,
                             const uint8_t* oflags, char* out);

} // namespace dz
