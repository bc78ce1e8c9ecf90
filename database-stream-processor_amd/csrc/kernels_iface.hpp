// kernels_iface.hpp — internal interface between kernels.hip (device code)
// and engine.cpp (host-side Circuit/Stream mirror).  Not part of the C ABI.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>
#include "../../include/dbsp_hip.h"

// spine-as-kernel-argument descriptor (multi-batch join/probe kernels)
#define MAX_TRACE_BATCHES 24
struct TraceArgs {
    int nb;
    const uint64_t *k[MAX_TRACE_BATCHES];
    const uint64_t *v[MAX_TRACE_BATCHES];
    const int64_t *w[MAX_TRACE_BATCHES];
    int64_t n[MAX_TRACE_BATCHES];
};

#define SORT_BATCH_MAX 8
struct SortArgs {
    int nb;
    const uint64_t *kin[SORT_BATCH_MAX];
    const uint64_t *vin[SORT_BATCH_MAX];
    const int64_t *win[SORT_BATCH_MAX];
    int64_t n[SORT_BATCH_MAX];
    uint64_t *tk[SORT_BATCH_MAX];
    uint64_t *tv[SORT_BATCH_MAX];
    int64_t *tw[SORT_BATCH_MAX];
    uint64_t *ok[SORT_BATCH_MAX];
    uint64_t *ov[SORT_BATCH_MAX];
    int64_t *ow[SORT_BATCH_MAX];
    int64_t *d_len;  // device array, one length per batch
    // optional per-batch DEVICE length (overrides n[b]): lets the sort launch
    // chain behind the producing kernel without a host sync.  If the device
    // length exceeds the fused capacity the kernel writes out_len = -1 and
    // the host re-sorts through the sized paths after its next sync.
    const int64_t *n_dev[SORT_BATCH_MAX];
};

// batched single-workgroup merges (one pair per workgroup; each na+nb <= 32768)
#define MERGE_BATCH_MAX 4
struct MergeArgs {
    int np;
    const uint64_t *ak[MERGE_BATCH_MAX];
    const uint64_t *av[MERGE_BATCH_MAX];
    const int64_t *aw[MERGE_BATCH_MAX];
    int64_t na[MERGE_BATCH_MAX];
    const uint64_t *bk[MERGE_BATCH_MAX];
    const uint64_t *bv[MERGE_BATCH_MAX];
    const int64_t *bw[MERGE_BATCH_MAX];
    int64_t nb[MERGE_BATCH_MAX];
    // optional device-side b length (the in-train accumulator merge: b is the
    // tick's delta, whose consolidated length lives in a d_len slot); null =>
    // use nb.  A negative device length (failed speculation) => emit -1.
    const int64_t *dnb[MERGE_BATCH_MAX];
    uint64_t *ok[MERGE_BATCH_MAX];
    uint64_t *ov[MERGE_BATCH_MAX];
    int64_t *ow[MERGE_BATCH_MAX];
    int64_t *d_len;
};

// fused chained-emit descriptor (bases + all plans' emits in one launch)
struct FusedEmitArgs {
    int np;
    const uint64_t *dk[3], *dv[3];
    const int64_t *dw[3];
    const int64_t *nd_dev[3];
    TraceArgs t[3];
    const int64_t *tn_dev[3];
    const uint32_t *cnts[3];
    const uint64_t *offsets[3];
    int proj[3];
    const int64_t *totals;  // per-plan count totals (device)
    int64_t cap;
    int64_t *d_total;  // combined total out (or -1)
    int64_t *d_flag;   // overflow flag out
    uint64_t *ok, *ov;
    int64_t *ow;
};
struct JoinCountArgs {
    int np;
    const uint64_t *dk[3];
    int64_t nd[3];
    TraceArgs t[3];
    uint32_t *cnts[3];
    uint64_t *offsets[3];
    int64_t *d_total;
    // optional device lengths: nd_dev overrides nd (the delta length), and
    // tn_dev overrides t.n[0] for single-batch traces that are themselves
    // tick-fresh deltas.  Negative device lengths (speculative-sort overflow)
    // make the plan write d_total = -1 for the host to detect at its sync.
    const int64_t *nd_dev[3];
    const int64_t *tn_dev[3];
};


namespace dbspk {
// big-buffer free-list over hipMallocAsync (see kernels.hip): same signature
// shape as hipMallocAsync/hipFreeAsync so call sites stay HIP_CHECK-able
hipError_t cache_malloc(void **out, size_t bytes, hipStream_t s);
hipError_t cache_free(void *p, hipStream_t s);
void cache_trim(hipStream_t s);
// scope the tick-scale (64 KB..4 MB) free-list band: trace-scale engines
// turn it off at init (see the definition for the measured pool interaction)
void cache_small_set(bool on, hipStream_t s);
}  // namespace dbspk

namespace dbspk {

dbsp_status scan_excl(hipStream_t s, const uint64_t *in, uint64_t *out,
                      int64_t n, uint64_t *h_total);
void fill_u64(hipStream_t s, uint64_t *p, uint64_t v, int64_t n);

dbsp_status sort_rows(hipStream_t s, uint64_t *kk, uint64_t *vv, int64_t *ww,
                      int64_t n, uint64_t *kk2, uint64_t *vv2, int64_t *ww2,
                      bool *result_in_scratch);

dbsp_status consolidate_sorted(hipStream_t s, const uint64_t *kk,
                               const uint64_t *vv, const int64_t *ww, int64_t n,
                               uint64_t **ok, uint64_t **ov, int64_t **ow,
                               int64_t *out_n);

dbsp_status merge_rows(hipStream_t s, const uint64_t *ak, const uint64_t *av,
                       const int64_t *aw, int64_t na, const uint64_t *bk,
                       const uint64_t *bv, const int64_t *bw, int64_t nb,
                       uint64_t **ok, uint64_t **ov, int64_t **ow,
                       int64_t *out_n);

// fused single-workgroup sort+consolidate batch descriptors: up to 4
// independent small (n <= 8192) raw batches sorted+consolidated concurrently,
// one workgroup each, lengths left in d_len[i] (device)
dbsp_status wm_update(hipStream_t s, const uint64_t *ak, const int64_t *n_dev,
                      uint64_t width, uint64_t tumble, uint64_t lag,
                      unsigned long long *state, unsigned long long *bounds);
dbsp_status wm_update_n(hipStream_t s, const uint64_t *ak, int64_t n,
                        uint64_t width, uint64_t tumble, uint64_t lag,
                        unsigned long long *state, unsigned long long *bounds);
dbsp_status wm_update_g(hipStream_t s, const unsigned long long *gmax_dev,
                        uint64_t width, uint64_t tumble, uint64_t lag,
                        unsigned long long *state, unsigned long long *bounds);
dbsp_status window_ranges_chain(hipStream_t s, const TraceArgs &t,
                                const uint64_t *bk, int64_t bn,
                                const int64_t *bn_dev,
                                const unsigned long long *bounds,
                                int64_t *table, int64_t *d_total);
dbsp_status minmax_rows_chain(hipStream_t s, const uint64_t *k,
                              const uint64_t *v, int64_t cap,
                              const int64_t *n_dev,
                              unsigned long long *mm_dev);
dbsp_status shard_rows_pair(hipStream_t s, const uint64_t *k0,
                            const uint64_t *v0, const int64_t *w0, int64_t n0,
                            const uint64_t *k1, const uint64_t *v1,
                            const int64_t *w1, int64_t n1, int nshards,
                            uint64_t *ok0, uint64_t *ov0, int64_t *ow0,
                            uint64_t *ok1, uint64_t *ov1, int64_t *ow1,
                            int64_t *h_off0, int64_t *h_off1);
// device event columns (the §8f3 columnizer, input.rs:591-721): staged
// events split once into SoA columns so the per-tick flatmaps read
// coalesced 8 B streams instead of 56 B strided AoS structs
struct EventCols {
    const uint64_t *kind, *f0, *f1, *f2, *f3, *f4;
    const int64_t *w;
};
dbsp_status columnize_events(hipStream_t s, const dbsp_event *ev, int64_t n,
                             uint64_t *kind, uint64_t *f0, uint64_t *f1,
                             uint64_t *f2, uint64_t *f3, uint64_t *f4,
                             int64_t *w);
dbsp_status flatmap_events_chain(hipStream_t s, const dbsp_event *ev,
                                 const EventCols *cols, int64_t n, int query,
                                 uint64_t *k0, uint64_t *v0, int64_t *w0,
                                 uint64_t *k1, uint64_t *v1, int64_t *w1,
                                 uint64_t *ctr);

// fixed-frame pair exchange (engine exchange fast path): both partitioned
// streams packed into per-peer segments of fixed capacity P0/P1 rows with a
// 2-word count header, so ONE equal-count ncclAllToAll replaces the
// counts-allgather + host sync + grouped send/recv mesh.  Segment layout
// (u64 words): [cnt0, cnt1, k0(P0), v0(P0), w0(P0), k1(P1), v1(P1), w1(P1)].
// A count above its capacity travels in the header as-is; the unpack kernel
// then writes -1 totals (sentinel) and the engine replays the exchange
// through the dynamic alltoallv path on every rank.
struct FramePairArgs {
    const uint64_t *p0k, *p0v;
    const int64_t *p0w;
    const uint64_t *p1k, *p1v;
    const int64_t *p1w;
    int64_t off0[9], off1[9];  // host partition offsets (nshards <= 8)
    int world;
    int64_t P0, P1;
    uint64_t *frame;
};
dbsp_status frames_pack_pair(hipStream_t s, const FramePairArgs &a);
// chained variant: hash-partition both RAW streams straight into the frames
// (lengths read from device counters; n_cap bounds the launch grid)
dbsp_status shard_frames_chain(hipStream_t s, const uint64_t *k0,
                               const uint64_t *v0, const int64_t *w0,
                               const int64_t *n0_dev, const uint64_t *k1,
                               const uint64_t *v1, const int64_t *w1,
                               const int64_t *n1_dev, int world, int64_t P0,
                               int64_t P1, int64_t n_cap, uint64_t *frame);
dbsp_status frames_unpack_pair(hipStream_t s, const uint64_t *frame,
                               int world, int64_t P0, int64_t P1,
                               uint64_t *r0k, uint64_t *r0v, int64_t *r0w,
                               uint64_t *r1k, uint64_t *r1v, int64_t *r1w,
                               int64_t *d_tot0, int64_t *d_tot1);
dbsp_status sort_cons_small_batch(hipStream_t s, const SortArgs &args);
// min/max of k and v (one sync): mm = {kmin, vmin, kmax, vmax}
dbsp_status minmax_rows(hipStream_t s, const uint64_t *k, const uint64_t *v,
                        int64_t n, uint64_t mm[4]);
// dense-range consolidate: weights scattered into a kspan*vspan histogram,
// nonzero cells emitted in index order (sorted); i64 only
dbsp_status sort_cons_dense(hipStream_t s, const uint64_t *k, const uint64_t *v,
                            const int64_t *w, int64_t n, uint64_t kbase,
                            uint64_t vbase, int64_t kspan, int64_t vspan,
                            uint64_t *ok, uint64_t *ov, int64_t *ow,
                            int64_t *out_n);
// chained dense consolidate: length to *out_n_dev, no sync
dbsp_status sort_cons_dense_chain(hipStream_t s, const uint64_t *k,
                                  const uint64_t *v, const int64_t *w,
                                  int64_t n, uint64_t kbase, uint64_t vbase,
                                  int64_t kspan, int64_t vspan, uint64_t *ok,
                                  uint64_t *ov, int64_t *ow,
                                  int64_t *out_n_dev);

// single-workgroup merge of two consolidated batches (na+nb <= 8192):
// one launch, no host sync; length left in *d_len (device)
dbsp_status merge_small_batch(hipStream_t s, const MergeArgs &args);
// fixed-grid multi-WG merge with optional device-side b lengths (in-train
// accumulator fold); scratch holds np * (MERGE_MID_SCRATCH) int64 slots
#define MERGE_MID_SCRATCH 33
dbsp_status merge_mid_batch(hipStream_t s, const MergeArgs &args,
                            int64_t *scratch);
dbsp_status merge_mid_batch_f64(hipStream_t s, const MergeArgs &args,
                                int64_t *scratch);

// up to 3 single-workgroup join count+scan plans in ONE launch (nd <= 8192
// each): per-plan per-row/per-batch cnts, exclusive offsets, totals to
// d_total[i] (device)
dbsp_status join_count_scan_batch(hipStream_t s, const JoinCountArgs &args);
// chained emit machinery: per-plan bases from device totals, emit with every
// size read device-side (flag short-circuits on lost speculation/overflow)
dbsp_status emit_bases(hipStream_t s, const int64_t *totals, int np,
                       int64_t cap, int64_t *bases, int64_t *out_total,
                       int64_t *out_flag);
dbsp_status join_emit_chain(hipStream_t s, const uint64_t *dk,
                            const uint64_t *dv, const int64_t *dw,
                            const int64_t *nd_dev, const TraceArgs &t,
                            const int64_t *tn_dev, const uint32_t *cnts,
                            const uint64_t *offsets, const int64_t *total_dev,
                            const int64_t *base_dev, const int64_t *flag_dev,
                            int64_t grid_cap, int proj, uint64_t param,
                            uint64_t *ok, uint64_t *ov, int64_t *ow);
// emit phase over precomputed cnts/offsets
dbsp_status join_emit_prepared(hipStream_t s, const uint64_t *dk,
                               const uint64_t *dv, const int64_t *dw,
                               int64_t nd, const TraceArgs &t,
                               const uint32_t *cnts, const uint64_t *offsets,
                               int64_t n_out, int proj, uint64_t param,
                               uint64_t *ok, uint64_t *ov, int64_t *ow);

// join delta against a whole spine (TraceArgs) in one count/emit pair
dbsp_status join_spine_rows(hipStream_t s, const uint64_t *dk,
                            const uint64_t *dv, const int64_t *dw, int64_t nd,
                            const TraceArgs &t, int proj, uint64_t param,
                            uint64_t **ok, uint64_t **ov, int64_t **ow,
                            int64_t *out_n);

dbsp_status join_rows(hipStream_t s, const uint64_t *dk, const uint64_t *dv,
                      const int64_t *dw, int64_t nd, const uint64_t *tk,
                      const uint64_t *tv, const int64_t *tw, int64_t nt,
                      int proj, uint64_t param, uint64_t **ok, uint64_t **ov,
                      int64_t **ow, int64_t *out_n);

// incremental distinct over a whole spine (non-linear: the per-pair total
// must be summed across batches before the indicator)
dbsp_status distinct_inc_rows(hipStream_t s, const uint64_t *dk,
                              const uint64_t *dv, const int64_t *dw,
                              int64_t nd, const TraceArgs &t, uint64_t **ok,
                              uint64_t **ov, int64_t **ow, int64_t *out_n);

dbsp_status join_emit_fused(hipStream_t s, const FusedEmitArgs &a);

dbsp_status agg_linear_upsert_rows(hipStream_t s, const uint64_t *keys,
                                   int64_t nd, const uint64_t *ik,
                                   const uint64_t *iv, const int64_t *iw,
                                   int64_t ni, const uint64_t *tok,
                                   const uint64_t *tov, const int64_t *tow,
                                   int64_t no, uint64_t **ok, uint64_t **ov,
                                   int64_t **ow, int64_t *out_n);

// q6's fold: avg of the last <= 10 vals in the key's run (cursor order)
dbsp_status agg_last10_upsert_rows(hipStream_t s, const uint64_t *keys,
                                   int64_t nd, const uint64_t *ik,
                                   const uint64_t *iv, const int64_t *iw,
                                   int64_t ni, const uint64_t *tok,
                                   const uint64_t *tov, const int64_t *tow,
                                   int64_t no, uint64_t **ok, uint64_t **ov,
                                   int64_t **ow, int64_t *out_n);
dbsp_status agg_max_upsert_rows(hipStream_t s, const uint64_t *keys, int64_t nd,
                                const uint64_t *ik, const uint64_t *iv,
                                const int64_t *iw, int64_t ni,
                                const uint64_t *tok, const uint64_t *tov,
                                const int64_t *tow, int64_t no, uint64_t **ok,
                                uint64_t **ov, int64_t **ow, int64_t *out_n);

// multi-batch linear aggregate: accumulate per-delta-key weight sums of one
// trace batch into acc[nd] (aggregation is linear in the trace, so spines are
// summed batch by batch)
dbsp_status agg_sum_batch(hipStream_t s, const uint64_t *keys, int64_t nd,
                          const uint64_t *ik, const int64_t *iw, int64_t ni,
                          int64_t *acc);
// emit (key, acc, +1) for acc != 0 (ticket append; caller consolidates)
dbsp_status emit_nonzero(hipStream_t s, const uint64_t *keys,
                         const int64_t *acc, int64_t nd, uint64_t *ok,
                         uint64_t *ov, int64_t *ow, int64_t *h_count);

dbsp_status window_rows(hipStream_t s, const uint64_t *tk, const uint64_t *tv,
                        const int64_t *tw, int64_t nt, const uint64_t *bk,
                        const uint64_t *bv, const int64_t *bw, int64_t nb,
                        int have_prev, uint64_t s0, uint64_t e0, uint64_t s1,
                        uint64_t e1, uint64_t **ok, uint64_t **ov, int64_t **ow,
                        int64_t *out_n);

// multi-batch window: ranges for all spine batches + tick batch in one launch
// (region table rows [src,lo,len,sign,goff]; total left in *d_total), then one
// grid-stride emit
dbsp_status window_ranges_multi(hipStream_t s, const TraceArgs &t,
                                const uint64_t *bk, int64_t bn, int have_prev,
                                uint64_t s0, uint64_t e0, uint64_t s1,
                                uint64_t e1, int64_t *table, int64_t *d_total);
dbsp_status window_emit_multi(hipStream_t s, const TraceArgs &t,
                              const uint64_t *bk, const uint64_t *bv,
                              const int64_t *bw, const int64_t *table,
                              int nreg, int64_t total, uint64_t *ok,
                              uint64_t *ov, int64_t *ow);

dbsp_status shard_rows(hipStream_t s, const uint64_t *k, const uint64_t *v,
                       const int64_t *w, int64_t n, int nshards, uint64_t *ok,
                       uint64_t *ov, int64_t *ow, int64_t *h_offsets);

dbsp_status flatmap_events(hipStream_t s, const dbsp_event *ev,
                           const EventCols *cols, int64_t n, int query,
                           uint64_t *k0, uint64_t *v0, int64_t *w0,
                           int64_t *n0, uint64_t *k1, uint64_t *v1,
                           int64_t *w1, int64_t *n1);

dbsp_status map_rows(hipStream_t s, const uint64_t *k, const uint64_t *v,
                     const int64_t *w, int64_t n, int mode, uint64_t *ok,
                     uint64_t *ov, int64_t *ow);

// distinct keys of a consolidated (sorted) batch
dbsp_status unique_keys(hipStream_t s, const uint64_t *kk, int64_t n,
                        uint64_t **okeys, int64_t *out_n);

// ---- f64-weight variants (config C5: f64 sum aggregate; deterministic
// position-fixed reduction order, tolerance 2 ulp * reduction depth) ----
dbsp_status sort_cons_small_batch_f64(hipStream_t s, const SortArgs &args);
dbsp_status merge_small_batch_f64(hipStream_t s, const MergeArgs &args);
dbsp_status merge_rows_f64(hipStream_t s, const uint64_t *ak,
                           const uint64_t *av, const double *aw, int64_t na,
                           const uint64_t *bk, const uint64_t *bv,
                           const double *bw, int64_t nb, uint64_t **ok,
                           uint64_t **ov, double **ow, int64_t *out_n);
dbsp_status consolidate_sorted_f64(hipStream_t s, const uint64_t *kk,
                                   const uint64_t *vv, double *ww, int64_t n,
                                   uint64_t **ok, uint64_t **ov, double **ow,
                                   int64_t *out_n);
dbsp_status agg_sum_batch_f64(hipStream_t s, const uint64_t *keys, int64_t nd,
                              const uint64_t *ik, const double *iw, int64_t ni,
                              double *acc);
dbsp_status emit_nonzero_f64(hipStream_t s, const uint64_t *keys,
                             const double *acc, int64_t nd, uint64_t *ok,
                             uint64_t *ov, int64_t *ow, int64_t *h_count);

// C5 synthetic operands: sorted-unique rows by construction
// (k_i = stride*i + h(i)%jitter, jitter < stride; val_mode 0 = uniform f64
// bits, 1 = zero)
dbsp_status c5_gen_rows(hipStream_t s, int64_t n, uint64_t stride,
                        uint64_t jitter, uint64_t seed, int val_mode,
                        uint64_t *k, uint64_t *v, int64_t *w);

// radix-tree rolling aggregate (§8f4): per input row (partition, ts, w),
// the weight sum over the partition's rows with time in [ts-width, ts]
dbsp_status rolling_agg_rows(hipStream_t s, const uint64_t *k,
                             const uint64_t *v, const int64_t *w, int64_t n,
                             uint64_t width, uint64_t *ok, uint64_t *ov,
                             int64_t *ow);

uint64_t host_xxh3_u64(uint64_t key, uint64_t seed);

}  // namespace dbspk
