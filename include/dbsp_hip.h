/* dbsp_hip.h — C-ABI boundary of the MI355X-native DBSP hot path.
 *
 * This is the FFI line that sits beneath operator `eval` in the reference's
 * Circuit/Stream API (reference: crates/dbsp/src/circuit/operator_traits.rs:18-430,
 * registered via circuit_builder.rs:626-900).  The reference is pure Rust with no
 * FFI; its hot-path subsystems (trace merge, consolidation, join, aggregate,
 * sharding) are re-implemented here as CDNA4 HIP kernels behind these extern "C"
 * entry points.  A Rust host (the north-star configuration) would bind exactly
 * these symbols; see INTEGRATION.md for the bindgen/extern-"C" stub a
 * crates/dbsp maintainer would add.  In this environment (no Rust toolchain)
 * the host-side mirror of the Stream/Circuit API is C++
 * (database-stream-processor_amd/csrc/engine.cpp) and calls the same symbols.
 *
 * Data model: a Z-set / indexed-Z-set batch is a struct-of-arrays of rows
 *   (key: u64, val: u64, weight: i64), sorted lexicographically by (key, val),
 * resident in device HBM.  This is the MI355X-native restatement of
 *   - ColumnLayer{keys,diffs}             (trace/layers/column_layer/mod.rs:31-36)
 *   - OrderedLayer{keys,offs,vals}        (trace/layers/ordered/mod.rs:32-44)
 *   - OrdZSet / OrdIndexedZSet            (trace/ord/zset_batch.rs:28,
 *                                          trace/ord/indexed_zset_batch.rs:27-41)
 * The CSR `offs` index of the reference is not materialised: kernels locate a
 * key's value range by binary search over the sorted (key,val) rows; this trades
 * the offs array for two log2(n) probes and keeps batches a single layout.
 *
 * All calls are asynchronous on the caller's hipStream_t and ordered by it
 * (the reference's per-worker scheduler thread maps to one HIP stream).
 * Output sizes are data-dependent (zero-weight elimination); calls take a
 * caller-owned device scratch arena and return lengths via device or host
 * pointers as documented per entry point.
 */
#ifndef DBSP_HIP_H
#define DBSP_HIP_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- status codes ---- */
typedef int32_t dbsp_status;
#define DBSP_OK                0
#define DBSP_ERR_NOGPU        -1  /* no HIP device: the product path fails loudly, never falls back */
#define DBSP_ERR_OOM          -2
#define DBSP_ERR_INVALID      -3
#define DBSP_ERR_OVERFLOW     -4  /* output exceeded provided capacity */
#define DBSP_ERR_INTERNAL     -5

/* ---- event row (Nexmark ingress; model: crates/nexmark/src/model.rs:14-69) ----
 * kind: 0 = Person, 1 = Auction, 2 = Bid.
 * Person : f0=id, f1=name_id, f2=city_id, f3=state_id, f4=date_time
 * Auction: f0=id, f1=seller,  f2=category, f3=date_time, f4=expires
 * Bid    : f0=auction, f1=bidder, f2=price, f3=date_time, f4=0
 * Strings are carried as dictionary ids end-to-end (SURVEY.md §7 hard part (b));
 * egress would map ids back through the dictionary. */
typedef struct {
    uint64_t kind;
    uint64_t f0, f1, f2, f3, f4;
    int64_t  w;
} dbsp_event;

/* ---- Z-set row ---- */
typedef struct {
    uint64_t k;
    uint64_t v;
    int64_t  w;
} dbsp_row;

/* ---- batch descriptor: SoA device pointers, rows sorted lex by (k,v) ---- */
typedef struct {
    uint64_t *k;     /* device */
    uint64_t *v;     /* device */
    int64_t  *w;     /* device */
    int64_t   len;
} dbsp_batch;

/* ---- join output projections ----
 * The reference's join takes a per-query `join_func` closure
 * (operator/join.rs:180,217: Fn(&K,&V1,&V2)->Z::Key).  Kernels monomorphise the
 * projections the Nexmark hot path needs (the same move the reference makes for
 * its typed, non-JIT path).  Each mode maps (k, v1, v2) -> output (hi, lo). */
typedef enum {
    DBSP_PROJ_HI_V2_LO_V1    = 0, /* q3: delta=auction_by_seller, trace=person_by_id */
    DBSP_PROJ_HI_V1_LO_V2    = 1, /* q3 swapped side */
    DBSP_PROJ_HI_K_LO_V1V2   = 2, /* generic pair join (reference join.rs:886 tests): lo=v1<<32|v2 */
    DBSP_PROJ_HI_K_LO_V1RND  = 3, /* q8: lo = (v1 & hi32) | round_down(v1 & lo32, param) */
    DBSP_PROJ_HI_K_LO_V2RND  = 4, /* q8 swapped side */
    DBSP_PROJ_HI_V2_LO_K     = 5, /* q5 final join */
    DBSP_PROJ_HI_V1_LO_K     = 6, /* q5 final join swapped side */
    DBSP_PROJ_HI_K_LO_V2V1   = 7, /* generic pair join, swapped side: lo=v2<<32|v1 */
    DBSP_PROJ_HI_K_LO_V2     = 8, /* (k, v2): keeps the join key, takes the trace val
                                     (full 64-bit; the C5 join -> f64-sum pipeline) */
    DBSP_PROJ_Q4_BID_X_AUC   = 9, /* q4.rs:58-68 join_func: delta=bid
                                     (v=bid_dt<<20|price), trace=auction
                                     (v=a_dt<<28|(expires-a_dt)<<4|(cat&0xF)):
                                     emit ((auction<<4)|cat, price) iff
                                     a_dt <= bid_dt <= expires, else weight 0
                                     (dropped by the consolidate) */
    DBSP_PROJ_Q4_AUC_X_BID   = 10, /* q4 swapped side */
    DBSP_PROJ_Q6_BID_X_AUC   = 11, /* q6.rs:60-80: delta=bid, trace=auction
                                      (v=a_dt<<36|(expires-a_dt)<<20|seller):
                                      emit ((auction<<20)|seller, price) in
                                      the validity window, else weight 0 */
    DBSP_PROJ_Q6_AUC_X_BID   = 12, /* q6 swapped side */
} dbsp_proj;

/* ======================================================================
 * Kernel-level entry points (each replaces a reference hot-path routine)
 * ====================================================================== */

/* Opaque per-device context: owns the HIP stream, scratch arena, RCCL comm. */
typedef struct dbsp_ctx dbsp_ctx;

dbsp_status dbsp_ctx_create(dbsp_ctx **out, int device);
dbsp_status dbsp_ctx_destroy(dbsp_ctx *ctx);
/* Blocks until all queued work on the context stream is complete. */
dbsp_status dbsp_ctx_sync(dbsp_ctx *ctx);

/* Device memory management (HBM arena; 288 GB per GPU). */
dbsp_status dbsp_dev_alloc(dbsp_ctx *ctx, size_t bytes, void **out);
dbsp_status dbsp_dev_free(dbsp_ctx *ctx, void *p);
dbsp_status dbsp_h2d(dbsp_ctx *ctx, void *dst_dev, const void *src_host, size_t bytes);
dbsp_status dbsp_d2h(dbsp_ctx *ctx, void *dst_host, const void *src_dev, size_t bytes);

/* Sort + consolidate unsorted rows into a batch.
 * Replaces consolidate_slice / consolidate_paired_slices
 * (trace/consolidation/mod.rs:91-110,212-231 — "the hottest code within the
 * entirety of the program") and the MergeBatcher seal path
 * (trace/ord/merge_batcher/mod.rs:22-197).
 * In : rows_in (device, len n, unsorted, arbitrary weights)
 * Out: batch `out` (device arrays allocated by the context arena), out->len set
 *      (host-visible after dbsp_ctx_sync). */
dbsp_status dbsp_sort_consolidate(dbsp_ctx *ctx,
                                  const uint64_t *k_in, const uint64_t *v_in,
                                  const int64_t *w_in, int64_t n,
                                  dbsp_batch *out);

/* Merge two sorted batches, summing weights of equal (k,v), dropping zeros.
 * Replaces ColumnLayerBuilder::push_merge (trace/layers/column_layer/builders.rs:98-169),
 * OrderedBuilder::merge_step (trace/layers/ordered/mod.rs:344-396) and the
 * fueled variants (ordered/mod.rs:493-575): a 2-level CSR merge over
 * OrdIndexedZSet is exactly a 1-level merge over (key,val) composite rows.
 * Merge-path (diagonal) partitioned; two-phase count+emit. */
dbsp_status dbsp_merge(dbsp_ctx *ctx, const dbsp_batch *a, const dbsp_batch *b,
                       dbsp_batch *out);

/* Delta x trace join with projection.
 * Replaces Join::eval (operator/join.rs:436-473) and the JoinTrace::eval inner
 * loop (operator/join.rs:732-787) at root scope (Time = (), time/mod.rs:223-235).
 * For every delta row (k,v1,w1) and every trace row (k,v2,w2) with equal k,
 * emits proj(k,v1,v2) with weight w1*w2.  Output is RAW (unconsolidated);
 * callers consolidate once per tick as the reference's output batcher does
 * (join.rs:792-850 degenerates to a single consolidate at Time=()).
 * The trace side may be a spine of several batches: join is linear in the
 * trace, so callers loop batches and concatenate. */
dbsp_status dbsp_join(dbsp_ctx *ctx, const dbsp_batch *delta,
                      const dbsp_batch *trace, dbsp_proj proj, uint64_t param,
                      dbsp_batch *out_raw);

/* Linear aggregate + upsert for the aggregate_linear path.
 * Replaces AggregateIncremental::eval_key with the WeightedCount aggregator
 * (operator/aggregate/mod.rs:129-156,479-547) followed by Upsert::eval
 * (operator/upsert.rs:161-208).
 * In : delta_keys (device, nd distinct sorted keys = keys of this tick's delta),
 *      input trace (integral of the weighed stream, incl. this tick),
 *      output trace (integral of this operator's own output, excl. this tick).
 * Out: raw update rows: for each key with new aggregate s != 0 emit (key, s, +1);
 *      for each existing (key, v, w_sum != 0) in the output trace emit (key, v, -w_sum).
 *      Caller consolidates. */
/* Radix-tree rolling aggregate (operator/time_series/radix_tree/mod.rs:1-75
 * + rolling_aggregate.rs:235-280): per row (partition, ts, w)
 * of a consolidated batch, the weight sum over the partition rows with time
 * in [ts-width, ts] (RelRange::range_of, range.rs:93-110).  Implemented as a
 * flat radix-16 prefix-aggregate tree over the sorted rows (O(log) query). */
dbsp_status dbsp_rolling_agg(dbsp_ctx *ctx, const dbsp_batch *in,
                             uint64_t width, dbsp_batch *out);

dbsp_status dbsp_agg_linear_upsert(dbsp_ctx *ctx,
                                   const uint64_t *delta_keys, int64_t nd,
                                   const dbsp_batch *in_trace,
                                   const dbsp_batch *out_trace,
                                   dbsp_batch *out_raw);

/* Max aggregate + upsert (reference operator/aggregate/max.rs:26-60). Same
 * contract as above but the new per-key value is the largest val whose total
 * weight in in_trace is non-zero (None if no such val). */
dbsp_status dbsp_agg_max_upsert(dbsp_ctx *ctx,
                                const uint64_t *delta_keys, int64_t nd,
                                const dbsp_batch *in_trace,
                                const dbsp_batch *out_trace,
                                dbsp_batch *out_raw);

/* Window operator: 3-region retract/insert range scan over a time-keyed trace.
 * Replaces Window::eval (operator/time_series/window.rs:144-220).
 * trace = integral of the stream up to but NOT including this tick;
 * batch = this tick's delta;  (s0,e0) = previous window (s0=e0=0, have_prev=0 on
 * the first tick);  (s1,e1) = new bounds.  Output raw rows. */
dbsp_status dbsp_window(dbsp_ctx *ctx, const dbsp_batch *trace,
                        const dbsp_batch *batch,
                        int have_prev, uint64_t s0, uint64_t e0,
                        uint64_t s1, uint64_t e1,
                        dbsp_batch *out_raw);

/* Key partitioning for the worker exchange.
 * Replaces shard_batch (operator/communication/shard.rs:165-199) with the hash
 * of hash.rs:9-13 (xxh3_64 with seed 0x7f95_ef85_be33_c337 over the 8 key
 * bytes).  Rows of `in` are stably scattered into nshards contiguous runs of
 * `out` by xxh3(k) % nshards; host-visible shard offsets written to
 * offsets_host[nshards+1] after sync.  The exchange itself (reference
 * exchange.rs:45-251, an N^2 mailbox) is an RCCL grouped send/recv over xGMI —
 * see dbsp_comm_* below. */
dbsp_status dbsp_shard_partition(dbsp_ctx *ctx, const dbsp_batch *in,
                                 int nshards, dbsp_batch *out,
                                 int64_t *offsets_host);

/* ---- f64-weight variants (config C5: the f64 sum aggregate path).
 * Weights travel in the same 8-byte column as f64 bit patterns.  All f64
 * reductions use a position-fixed order (segmented tree / batch order);
 * documented tolerance vs a sequential sum: |err| <= 2 ulp * reduction depth
 * (SURVEY.md §8d).  Zero-weight elimination follows the reference's F64
 * is_zero (== 0.0, so -0.0 is eliminated; algebra/floats.rs:24). */
dbsp_status dbsp_sort_consolidate_f64(dbsp_ctx *ctx, const uint64_t *k_in,
                                      const uint64_t *v_in, const double *w_in,
                                      int64_t n, dbsp_batch *out);
dbsp_status dbsp_merge_f64(dbsp_ctx *ctx, const dbsp_batch *a,
                           const dbsp_batch *b, dbsp_batch *out);
/* weigh (aggregate/mod.rs:297-323) with f(k, v) = f64_from_bits(v):
 * (k, v, w) -> raw rows (k, (), f64(v) * w); caller consolidates. */
dbsp_status dbsp_weigh_f64(dbsp_ctx *ctx, const dbsp_batch *in,
                           dbsp_batch *out_raw);
/* linear aggregate over an f64-weighted input trace + upsert against an
 * i64-weighted output trace whose values are f64 bit patterns. */
dbsp_status dbsp_agg_linear_upsert_f64(dbsp_ctx *ctx,
                                       const uint64_t *delta_keys, int64_t nd,
                                       const dbsp_batch *in_trace,
                                       const dbsp_batch *out_trace,
                                       dbsp_batch *out_raw);

/* Incremental distinct (operator/distinct.rs:273: DistinctIncremental at
 * root scope): for each delta pair, out = [w_before + dw > 0] - [w_before > 0]
 * where w_before is the pair's total weight in the delayed integral (passed
 * as its spine batches: distinct is NOT linear in the trace).  The output is
 * consolidated by construction.  Unlocks antijoin/outer-join and Nexmark
 * q4/q6/q9. */
dbsp_status dbsp_distinct_inc(dbsp_ctx *ctx, const dbsp_batch *delta,
                              const dbsp_batch *trace_batches, int n_batches,
                              dbsp_batch *out);

/* Distinct keys of a consolidated batch (device out, caller frees). */
dbsp_status dbsp_unique_keys(dbsp_ctx *ctx, const dbsp_batch *in,
                             uint64_t **out_keys, int64_t *n_out);

/* Reference hash (host-callable too, for tests): xxh3_64(le_bytes(key), seed). */
uint64_t dbsp_xxh3_u64(uint64_t key, uint64_t seed);

/* ---- RCCL exchange over xGMI (replaces exchange.rs:45-251) ---- */
/* nccl_id is the 128-byte ncclUniqueId obtained by rank 0 and distributed
 * out-of-band (e.g. torch.distributed broadcast). */
dbsp_status dbsp_comm_init(dbsp_ctx *ctx, int rank, int world,
                           const void *nccl_id /* 128 bytes */);
/* All-to-all-v of row columns: send_counts[world] rows from `send` (contiguous
 * runs in shard order, as produced by dbsp_shard_partition); receives into
 * `recv` (capacity recv_cap rows); recv_counts_host[world] written after sync. */
dbsp_status dbsp_comm_alltoallv(dbsp_ctx *ctx, const dbsp_batch *send,
                                const int64_t *send_counts,
                                dbsp_batch *recv, int64_t recv_cap,
                                int64_t *recv_counts_host);

/* ======================================================================
 * Engine-level entry points (host-side mirror of the Circuit/Stream API;
 * see engine.cpp for the operator-by-operator mapping)
 * ====================================================================== */

typedef struct dbsp_engine dbsp_engine;

/* query: 0, 3, 5, 8 (Nexmark — the same operator DAG as
 * crates/nexmark/src/queries/q{0,3,5,8}.rs over the mirrored API), or 100
 * (config C5: the synthetic 1B-row OrdIndexedZSet x 10M-row delta
 * incremental join + f64 sum aggregate of BASELINE configs[4]). */
dbsp_status dbsp_engine_create(dbsp_engine **out, dbsp_ctx *ctx, int query,
                               int rank, int world);
dbsp_status dbsp_engine_destroy(dbsp_engine *e);

/* C5 (query 100) setup: device-generates the n_trace-row (k, f64-bits, +1)
 * indexed trace (counter-based splitmix64; sorted-unique by construction)
 * and fixes the per-tick delta size.  After this, step ranges index delta
 * ROWS: dbsp_engine_run_staged(e, 0, steps*n_delta, n_delta) runs `steps`
 * ticks, each one a fresh device-generated n_delta-row delta joined and
 * aggregated incrementally (operator mapping in the engine source). */
dbsp_status dbsp_engine_c5_init(dbsp_engine *e, int64_t n_trace,
                                int64_t n_delta, uint64_t seed);

/* One clock tick: ingest `events` (host array, this tick's input delta) and run
 * every operator once in dependency order — the mirror of DBSPHandle::step
 * (circuit/dbsp_handle.rs:246) + the static scheduler (circuit/schedule/).
 * The tick's output delta is retained on device; fetch with
 * dbsp_engine_output. */
dbsp_status dbsp_engine_step(dbsp_engine *e, const dbsp_event *events, int64_t n);

/* Events may instead be pre-staged to HBM once (bench: inputs resident before
 * the timed region) and stepped by index range. */
dbsp_status dbsp_engine_stage_events(dbsp_engine *e, const dbsp_event *events,
                                     int64_t n);
dbsp_status dbsp_engine_step_staged(dbsp_engine *e, int64_t lo, int64_t hi);
/* Run staged events [lo,hi) in tick-sized steps inside one call (the
 * benchmark loop; per-tick outputs are produced and replaced in turn —
 * read dbsp_engine_output afterwards for the LAST tick only). */
dbsp_status dbsp_engine_run_staged(dbsp_engine *e, int64_t lo, int64_t hi,
                                   int64_t tick);

/* Copy last tick's output delta to host rows; returns count via *n_out
 * (capacity cap rows; DBSP_ERR_OVERFLOW if larger). For q0 the output rows are
 * the consolidated event zset re-packed as (hi,lo,w) row pairs per event
 * (see engine.cpp q0 notes). */
dbsp_status dbsp_engine_output(dbsp_engine *e, dbsp_row *out, int64_t cap,
                               int64_t *n_out);

/* Perf counters for the roofline leg: total device ns and bytes of the
 * dominant kernel class since engine creation (measured with hipEvents on the
 * engine's own stream). kind: 0=merge/consolidate sort passes, 1=merge-path,
 * 2=join, 3=aggregate, 4=window, 5=everything-else. */
dbsp_status dbsp_engine_kernel_stats(dbsp_engine *e, int kind,
                                     double *total_ms, double *algo_bytes,
                                     int64_t *launches);

#ifdef __cplusplus
}
#endif
#endif /* DBSP_HIP_H */
