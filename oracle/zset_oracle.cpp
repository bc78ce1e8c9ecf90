// oracle/zset_oracle.cpp — CPU oracle for the MI355X DBSP hot path.
//
// *** TEST INFRASTRUCTURE ONLY ***
// This library is the CPU restatement of the reference's hot-path algorithms
// (vmware/database-stream-processor, mounted read-only at /root/reference).
// It exists to (a) pin parity of the HIP kernels and the GPU engine, and
// (b) serve as the measured CPU baseline (`cpu_baseline` leg of bench.py,
// kind "port").  Only tests/, __graft_entry__.smoke() and bench.py's
// cpu_baseline leg may link or call it.  The product path
// (database-stream-processor_amd/) never routes through this code and fails
// loudly when its HIP extension is missing.
//
// Pinning: the reference is Rust and cannot be compiled in this environment
// (no rustc/cargo, no network — SURVEY.md §0/§8c), so there is no oracle/_ref
// binary.  Instead this oracle is pinned against golden vectors transcribed
// from the reference's own tests into tests/golden/ :
//   - operator/join.rs:886-1017      (stream_join + incremental join outputs)
//   - nexmark/src/queries/q3.rs:74-220, q5.rs:130-212, q8.rs:104-216
//   - trace/consolidation semantics  (consolidation/mod.rs:32-110)
// plus randomized model tests in tests/test_oracle.py following the
// reference's TestBatch pattern (trace/test_batch.rs:1-60).
//
// Every function cites the reference file:line it restates.
//
// Build: g++ -O2 -std=c++17 -shared -fPIC zset_oracle.cpp -o liboracle_dbsp.so

#include <algorithm>
#include <cstdint>
#include <cstring>
#include <vector>

#include "../include/dbsp_hip.h"  // ABI structs/enums only (dbsp_event, dbsp_row, dbsp_proj)

namespace {

struct Row {
    uint64_t k, v;
    int64_t w;
};

static bool row_lt(const Row &a, const Row &b) {
    if (a.k != b.k) return a.k < b.k;
    return a.v < b.v;
}

// consolidate: sort by (k,v), accumulate weights of equal rows, drop zeros.
// Restates consolidate_slice / consolidate_paired_slices
// (trace/consolidation/mod.rs:91-110,212-231: sort_unstable by key, then
// dedup-accumulate, then retain non-zero weights).  The "key" of an indexed
// Z-set row is the (key,val) pair (trace/ord/indexed_zset_batch.rs:27-41).
void consolidate(std::vector<Row> &rows) {
    if (rows.empty()) return;
    std::sort(rows.begin(), rows.end(), row_lt);
    size_t off = 0;  // index of current accumulation (consolidation/mod.rs:117-174)
    for (size_t i = 1; i < rows.size(); i++) {
        if (rows[off].k == rows[i].k && rows[off].v == rows[i].v) {
            rows[off].w += rows[i].w;
        } else {
            if (rows[off].w != 0) off++;
            rows[off] = rows[i];
        }
    }
    if (rows[off].w != 0) off++;
    rows.resize(off);
}

// Two-way sorted merge with weight accumulation and zero elimination.
// Restates ColumnLayerBuilder::push_merge (trace/layers/column_layer/builders.rs:98-169)
// lifted to (key,val) composite rows, which also covers the two-level
// OrderedBuilder::merge_step (trace/layers/ordered/mod.rs:344-396): merging a
// CSR level and recursing into equal keys' value layers is exactly the
// composite-row merge, with keys whose merged value layer is empty dropped.
std::vector<Row> merge(const std::vector<Row> &a, const std::vector<Row> &b) {
    std::vector<Row> out;
    out.reserve(a.size() + b.size());
    size_t i = 0, j = 0;
    while (i < a.size() && j < b.size()) {
        if (row_lt(a[i], b[j])) {
            out.push_back(a[i++]);
        } else if (row_lt(b[j], a[i])) {
            out.push_back(b[j++]);
        } else {
            int64_t s = a[i].w + b[j].w;  // builders.rs:134-141
            if (s != 0) out.push_back({a[i].k, a[i].v, s});
            i++; j++;
        }
    }
    for (; i < a.size(); i++) out.push_back(a[i]);
    for (; j < b.size(); j++) out.push_back(b[j]);
    return out;
}

// Join projection: the monomorphised join_func set (see include/dbsp_hip.h).
inline void proj_emit(std::vector<Row> &out, dbsp_proj proj, uint64_t param,
                      uint64_t k, uint64_t v1, uint64_t v2, int64_t w) {
    uint64_t hi = 0, lo = 0;
    switch (proj) {
        case DBSP_PROJ_HI_V2_LO_V1: hi = v2; lo = v1; break;
        case DBSP_PROJ_HI_V1_LO_V2: hi = v1; lo = v2; break;
        case DBSP_PROJ_HI_K_LO_V1V2: hi = k; lo = (v1 << 32) | (v2 & 0xFFFFFFFFull); break;
        case DBSP_PROJ_HI_K_LO_V1RND: {
            uint64_t dt = v1 & 0xFFFFFFFFull;
            hi = k; lo = (v1 & 0xFFFFFFFF00000000ull) | (dt - dt % param);
            break;
        }
        case DBSP_PROJ_HI_K_LO_V2RND: {
            uint64_t dt = v2 & 0xFFFFFFFFull;
            hi = k; lo = (v2 & 0xFFFFFFFF00000000ull) | (dt - dt % param);
            break;
        }
        case DBSP_PROJ_HI_V2_LO_K: hi = v2; lo = k; break;
        case DBSP_PROJ_HI_V1_LO_K: hi = v1; lo = k; break;
        case DBSP_PROJ_HI_K_LO_V2V1: hi = k; lo = (v2 << 32) | (v1 & 0xFFFFFFFFull); break;
        case DBSP_PROJ_HI_K_LO_V2: hi = k; lo = v2; break;
        case DBSP_PROJ_Q4_BID_X_AUC: {
            // q4.rs:58-68 join_func: delta = bid (v1 = bid_dt<<27|price),
            // trace = auction (v2 = a_dt<<28|(expires-a_dt)<<4|cat&0xF);
            // invalid pairs keep their slot with weight 0 (dropped by the
            // consolidate), mirroring the GPU emit contract
            const uint64_t bid_dt = v1 >> 27, price = v1 & 0x7FFFFFFull;
            const uint64_t a_dt = v2 >> 28, dur = (v2 >> 4) & 0xFFFFFFull;
            hi = (k << 4) | (v2 & 0xFull);
            lo = price;
            if (!(bid_dt >= a_dt && bid_dt <= a_dt + dur)) w = 0;
            break;
        }
        case DBSP_PROJ_Q6_BID_X_AUC: {
            // q6.rs:60-80: delta = bid (v1 = bid_dt<<27|price), trace =
            // auction (v2 = a_dt<<34 | (expires-a_dt)<<20 | seller)
            const uint64_t bid_dt = v1 >> 27, price = v1 & 0x7FFFFFFull;
            const uint64_t a_dt = v2 >> 36, dur = (v2 >> 20) & 0xFFFFull;
            hi = (k << 20) | (v2 & 0xFFFFFull);
            lo = price;
            if (!(bid_dt >= a_dt && bid_dt <= a_dt + dur)) w = 0;
            break;
        }
        case DBSP_PROJ_Q6_AUC_X_BID: {
            const uint64_t bid_dt = v2 >> 27, price = v2 & 0x7FFFFFFull;
            const uint64_t a_dt = v1 >> 36, dur = (v1 >> 20) & 0xFFFFull;
            hi = (k << 20) | (v1 & 0xFFFFFull);
            lo = price;
            if (!(bid_dt >= a_dt && bid_dt <= a_dt + dur)) w = 0;
            break;
        }
        case DBSP_PROJ_Q4_AUC_X_BID: {
            const uint64_t bid_dt = v2 >> 27, price = v2 & 0x7FFFFFFull;
            const uint64_t a_dt = v1 >> 28, dur = (v1 >> 4) & 0xFFFFFFull;
            hi = (k << 4) | (v1 & 0xFull);
            lo = price;
            if (!(bid_dt >= a_dt && bid_dt <= a_dt + dur)) w = 0;
            break;
        }
    }
    out.push_back({hi, lo, w});
}

// delta x trace join: two-cursor sorted intersection with per-key val x val
// cross product, weights w1*w2.  Restates Join::eval (operator/join.rs:436-473)
// and the JoinTrace::eval inner loop (operator/join.rs:751-787) at Time=()
// (root scope; time/mod.rs:223-235 — per-time batching degenerates to a single
// consolidate, which the caller performs).
void join_raw(const std::vector<Row> &delta, const std::vector<Row> &trace,
              dbsp_proj proj, uint64_t param, std::vector<Row> &out) {
    size_t i = 0, j = 0;
    while (i < delta.size() && j < trace.size()) {
        if (delta[i].k < trace[j].k) {
            i++;
        } else if (trace[j].k < delta[i].k) {
            j++;
        } else {
            uint64_t key = delta[i].k;
            size_t i_end = i, j_end = j;
            while (i_end < delta.size() && delta[i_end].k == key) i_end++;
            while (j_end < trace.size() && trace[j_end].k == key) j_end++;
            for (size_t a = i; a < i_end; a++)
                for (size_t b = j; b < j_end; b++)
                    proj_emit(out, proj, param, key, delta[a].v, trace[b].v,
                              delta[a].w * trace[b].w);
            i = i_end; j = j_end;
        }
    }
}

// Linear aggregate (WeightedCount) + upsert.
// Restates AggregateIncremental::eval_key with WeightedCount
// (operator/aggregate/mod.rs:129-156: Some(sum of weights) or None if zero;
// mod.rs:479-547: seek key in input trace) and Upsert::eval
// (operator/upsert.rs:161-208: push (key,new,+1), retract every existing
// (key,val) of the output trace with its accumulated weight negated,
// consolidate per key).  delta_keys must be sorted+distinct (upsert.rs:162-166
// debug asserts).  in_trace includes the current tick (trace.rs TraceAppend
// semantics, see operator/trace.rs:324-460); out_trace excludes it.
void agg_linear_upsert(const std::vector<uint64_t> &delta_keys,
                       const std::vector<Row> &in_trace,
                       const std::vector<Row> &out_trace,
                       std::vector<Row> &out) {
    for (uint64_t key : delta_keys) {
        // aggregate: sum of weights of `key` in in_trace (vals are ())
        auto lo = std::lower_bound(in_trace.begin(), in_trace.end(), key,
                                   [](const Row &r, uint64_t k) { return r.k < k; });
        int64_t s = 0;
        for (auto it = lo; it != in_trace.end() && it->k == key; ++it) s += it->w;
        std::vector<Row> key_updates;
        if (s != 0) key_updates.push_back({key, (uint64_t)s, 1});
        // upsert retraction against the output trace
        auto lo2 = std::lower_bound(out_trace.begin(), out_trace.end(), key,
                                    [](const Row &r, uint64_t k) { return r.k < k; });
        for (auto it = lo2; it != out_trace.end() && it->k == key; ++it)
            if (it->w != 0) key_updates.push_back({key, it->v, -it->w});
        consolidate(key_updates);  // upsert.rs:198
        for (auto &r : key_updates) out.push_back(r);
    }
}

// Max aggregate + upsert.  Restates Max::aggregate
// (operator/aggregate/max.rs:36-55: fast-forward to the largest val, step in
// reverse to the first val with non-zero total weight) + the same upsert.
void agg_max_upsert(const std::vector<uint64_t> &delta_keys,
                    const std::vector<Row> &in_trace,
                    const std::vector<Row> &out_trace,
                    std::vector<Row> &out) {
    for (uint64_t key : delta_keys) {
        auto lo = std::lower_bound(in_trace.begin(), in_trace.end(), key,
                                   [](const Row &r, uint64_t k) { return r.k < k; });
        auto hi = lo;
        while (hi != in_trace.end() && hi->k == key) ++hi;
        bool found = false;
        uint64_t maxv = 0;
        for (auto it = hi; it != lo;) {
            --it;
            if (it->w != 0) { found = true; maxv = it->v; break; }
        }
        std::vector<Row> key_updates;
        if (found) key_updates.push_back({key, maxv, 1});
        auto lo2 = std::lower_bound(out_trace.begin(), out_trace.end(), key,
                                    [](const Row &r, uint64_t k) { return r.k < k; });
        for (auto it = lo2; it != out_trace.end() && it->k == key; ++it)
            if (it->w != 0) key_updates.push_back({key, it->v, -it->w});
        consolidate(key_updates);
        for (auto &r : key_updates) out.push_back(r);
    }
}

// Window operator: 3-region retract/insert scan.
// Restates Window::eval (operator/time_series/window.rs:144-220):
//   region 1  [s0 .. min(s1,e0)): retract from trace
//   shrink    [e1 .. e0): retract from trace (when e1 < e0)
//   region 3  [max(e0,s1) .. e1): insert from trace
//   batch     [s1 .. e1): insert from this tick's batch
// trace excludes the current tick's batch.
void window_eval(const std::vector<Row> &trace, const std::vector<Row> &batch,
                 bool have_prev, uint64_t s0, uint64_t e0, uint64_t s1,
                 uint64_t e1, std::vector<Row> &out) {
    auto seek = [](const std::vector<Row> &v, uint64_t key) {
        return std::lower_bound(v.begin(), v.end(), key,
                                [](const Row &r, uint64_t k) { return r.k < k; });
    };
    if (have_prev) {
        // region 1
        for (auto it = seek(trace, s0); it != trace.end() && it->k < s1 && it->k < e0; ++it)
            out.push_back({it->k, it->v, -it->w});
        // window shrunk on the right (window.rs:186-196)
        if (e1 < e0)
            for (auto it = seek(trace, e1); it != trace.end() && it->k < e0; ++it)
                out.push_back({it->k, it->v, -it->w});
        // region 3
        for (auto it = seek(trace, std::max(e0, s1)); it != trace.end() && it->k < e1; ++it)
            out.push_back({it->k, it->v, it->w});
    }
    // batch region (window.rs:209-216)
    for (auto it = seek(batch, s1); it != batch.end() && it->k < e1; ++it)
        out.push_back({it->k, it->v, it->w});
}

// xxh3_64 of the 8 little-endian key bytes with the reference's seed
// (hash.rs:6: 0x7f95_ef85_be33_c337; xxhash-rust's Xxh3::with_seed over a u64
// write).  This is the 4-to-8-byte one-shot path of the public XXH3
// specification (XXH3_len_4to8_64b).
uint64_t xxh3_u64(uint64_t key, uint64_t seed) {
    // XXH3 constants / secret prefix (public spec, xxhash v0.8)
    static const uint64_t PRIME64_1 = 0x9E3779B185EBCA87ull;
    static const uint64_t PRIME64_2 = 0xC2B2AE3D27D4EB4Full;
    static const uint64_t PRIME64_3 = 0x165667B19E3779F9ull;
    static const uint32_t PRIME32_2 = 0x85EBCA77u;
    static const uint8_t kSecret[24 + 8] = {
        // bytes 0..32 of XXH3_kSecret (spec)
        0xb8, 0xfe, 0x6c, 0x39, 0x23, 0xa4, 0x4b, 0xbe, 0x7c, 0x01, 0x81, 0x2c,
        0xf7, 0x21, 0xad, 0x1c, 0xde, 0xd4, 0x6d, 0xe9, 0x83, 0x90, 0x97, 0xdb,
        0x72, 0x40, 0xa4, 0xa4, 0xb7, 0xb3, 0x67, 0x1f,
    };
    auto read64 = [](const uint8_t *p) {
        uint64_t x; std::memcpy(&x, p, 8); return x;
    };
    // XXH3_len_4to8_64b(input, 8, kSecret, seed)
    uint64_t s = seed ^ ((uint64_t)__builtin_bswap32((uint32_t)seed) << 32);
    uint32_t in_lo = (uint32_t)(key & 0xFFFFFFFFull);
    uint32_t in_hi = (uint32_t)(key >> 32);
    uint64_t bitflip = (read64(kSecret + 8) ^ read64(kSecret + 16)) - s;
    uint64_t input64 = (uint64_t)in_hi + ((uint64_t)in_lo << 32);
    uint64_t keyed = input64 ^ bitflip;
    // XXH3_rrmxmx(keyed, len=8)
    uint64_t h = keyed;
    auto rotl = [](uint64_t x, int r) { return (x << r) | (x >> (64 - r)); };
    h ^= rotl(h, 49) ^ rotl(h, 24);
    h *= 0x9FB21C651E98DF25ull;
    h ^= (h >> 35) + 8 /* len */;
    h *= 0x9FB21C651E98DF25ull;
    h ^= (h >> 28);
    (void)PRIME64_1; (void)PRIME64_2; (void)PRIME64_3; (void)PRIME32_2;
    return h;
}

// ---------------------------------------------------------------------------
// Nexmark query drivers over the primitives above.  Each mirrors the operator
// DAG of crates/nexmark/src/queries/q{0,3,5,8}.rs; incremental joins use the
// bilinear decomposition out_t = dA ⋈ B_{t-1} + A_t ⋈ dB (operator/join.rs:217-292:
// two JoinTrace operators, one over the delayed trace).
// ---------------------------------------------------------------------------

// State id mapping used by the generator/test dictionaries (see
// database-stream-processor_amd/csrc/nexmark_gen.hpp): index into the
// reference's US_STATES table (generator/people.rs:18-25):
//   0=AZ 1=CA 2=ID 3=OR 4=WA 5=WY
// q3's states of interest {OR, ID, CA} = {3, 2, 1} (queries/q3.rs:30).
inline bool q3_state_of_interest(uint64_t state_id) {
    return state_id == 1 || state_id == 2 || state_id == 3;
}
constexpr uint64_t Q3_CATEGORY = 10;  // queries/q3.rs:31 (FIRST_CATEGORY_ID=10)

inline uint64_t pack_person(uint64_t name_id, uint64_t city_id, uint64_t state_id) {
    return (name_id << 8) | ((city_id & 0xF) << 4) | (state_id & 0xF);
}

struct Oracle {
    int query;
    // consolidated-event staging (input.rs:664 OrdZSet::from_tuples per tick)
    // q3 state
    std::vector<Row> a_int, p_int;
    // q8 state
    std::vector<Row> pt_int, at_int;       // people/auctions by time (trace for window)
    std::vector<Row> wp_int, wa_int;       // windowed people by id / auctions by seller
    bool q8_have_prev_p = false, q8_have_prev_a = false;
    uint64_t q8_s0p = 0, q8_e0p = 0, q8_s0a = 0, q8_e0a = 0;
    uint64_t q8_wm = 0;
    // q4 state
    std::vector<Row> q4_a_int, q4_b_int;   // auctions / bids by auction id
    std::vector<Row> q4_maxin, q4_maxout;  // max in/out integrals
    std::vector<Row> q4_avg_int, q4_avgout;  // packed avg integral + output
    // q6 state
    std::vector<Row> q6_a_int, q6_b_int;
    std::vector<Row> q6_maxin, q6_maxout;    // per-(auction,seller) max
    std::vector<Row> q6_fold_in, q6_foldout; // seller-keyed last-10 fold
    // q5 state
    std::vector<Row> bt_int;               // bids by time (trace for window)
    std::vector<Row> wb_int;               // windowed bids by auction (weighed integral)
    std::vector<Row> counts_int;           // aggregate output trace (auction -> count)
    std::vector<Row> maxin_int;            // (() -> count) aggregate input integral
    std::vector<Row> maxout_int;           // (() -> max) aggregate output trace
    std::vector<Row> maxz_int;             // max as zset keyed by count
    std::vector<Row> bc_int;               // (count -> auction)
    bool q5_have_prev = false;
    uint64_t q5_s0 = 0, q5_e0 = 0;
    uint64_t q5_wm = 0;
};

// Consolidate the tick's event multiset — the input path's
// OrdZSet::from_tuples / MergeBatcher seal (input.rs:664, trace/mod.rs:259-263).
void consolidate_events(std::vector<dbsp_event> &ev) {
    std::sort(ev.begin(), ev.end(), [](const dbsp_event &a, const dbsp_event &b) {
        if (a.kind != b.kind) return a.kind < b.kind;
        if (a.f0 != b.f0) return a.f0 < b.f0;
        if (a.f1 != b.f1) return a.f1 < b.f1;
        if (a.f2 != b.f2) return a.f2 < b.f2;
        if (a.f3 != b.f3) return a.f3 < b.f3;
        return a.f4 < b.f4;
    });
    size_t off = 0;
    auto eq = [](const dbsp_event &a, const dbsp_event &b) {
        return a.kind == b.kind && a.f0 == b.f0 && a.f1 == b.f1 &&
               a.f2 == b.f2 && a.f3 == b.f3 && a.f4 == b.f4;
    };
    if (ev.empty()) return;
    for (size_t i = 1; i < ev.size(); i++) {
        if (eq(ev[off], ev[i])) {
            ev[off].w += ev[i].w;
        } else {
            if (ev[off].w != 0) off++;
            ev[off] = ev[i];
        }
    }
    if (ev[off].w != 0) off++;
    ev.resize(off);
}

std::vector<Row> q3_step(Oracle &o, const std::vector<dbsp_event> &ev) {
    // flat_map_index (queries/q3.rs:37-49) + sort/consolidate builds
    std::vector<Row> dA, dP;
    for (auto &e : ev) {
        if (e.kind == 1 && e.f2 == Q3_CATEGORY) dA.push_back({e.f1, e.f0, e.w});
        if (e.kind == 0 && q3_state_of_interest(e.f3))
            dP.push_back({e.f0, pack_person(e.f1, e.f2, e.f3), e.w});
    }
    consolidate(dA);
    consolidate(dP);
    std::vector<Row> out;
    join_raw(dA, o.p_int, DBSP_PROJ_HI_V2_LO_V1, 0, out);  // dA ⋈ P_{t-1}
    o.a_int = merge(o.a_int, dA);
    join_raw(dP, o.a_int, DBSP_PROJ_HI_V1_LO_V2, 0, out);  // A_t ⋈ dP
    o.p_int = merge(o.p_int, dP);
    consolidate(out);
    return out;
}

// queries/q4.rs: join (validity-filtered) -> Max per (auction,category) ->
// Average per category (packed (sum<<20)+count linear aggregate; see the
// engine's q4_step for the operator mapping)
std::vector<Row> q4_step(Oracle &o, const std::vector<dbsp_event> &ev) {
    std::vector<Row> dA, dB;
    for (auto &e : ev) {
        if (e.kind == 1)
            dA.push_back({e.f0, (e.f3 << 28) | (((e.f4 - e.f3) & 0xFFFFFFull) << 4) |
                                    (e.f2 & 0xFull),
                          e.w});
        if (e.kind == 2)
            dB.push_back({e.f0, (e.f3 << 27) | (e.f2 & 0x7FFFFFFull), e.w});
    }
    consolidate(dA);
    consolidate(dB);
    std::vector<Row> dWinIn;
    join_raw(dB, o.q4_a_int, DBSP_PROJ_Q4_BID_X_AUC, 0, dWinIn);  // dB ⋈ A_{t-1}
    o.q4_b_int = merge(o.q4_b_int, dB);
    join_raw(dA, o.q4_b_int, DBSP_PROJ_Q4_AUC_X_BID, 0, dWinIn);  // dA ⋈ B_t
    o.q4_a_int = merge(o.q4_a_int, dA);
    consolidate(dWinIn);
    std::vector<Row> dWin;
    if (!dWinIn.empty()) {
        o.q4_maxin = merge(o.q4_maxin, dWinIn);
        std::vector<uint64_t> keys;
        for (auto &r : dWinIn)
            if (keys.empty() || keys.back() != r.k) keys.push_back(r.k);
        agg_max_upsert(keys, o.q4_maxin, o.q4_maxout, dWin);
        consolidate(dWin);
        o.q4_maxout = merge(o.q4_maxout, dWin);
    }
    // average per category: weigh to the packed (sum<<20)+count weight
    std::vector<Row> dAvgIn;
    for (auto &r : dWin)
        dAvgIn.push_back({r.k & 0xFull, 0, r.w * (int64_t)((r.v << 18) | 1ull)});
    consolidate(dAvgIn);
    std::vector<Row> out;
    if (!dAvgIn.empty()) {
        o.q4_avg_int = merge(o.q4_avg_int, dAvgIn);
        std::vector<uint64_t> keys;
        for (auto &r : dAvgIn)
            if (keys.empty() || keys.back() != r.k) keys.push_back(r.k);
        std::vector<Row> upd;
        agg_linear_upsert(keys, o.q4_avg_int, o.q4_avgout, upd);
        consolidate(upd);
        o.q4_avgout = merge(o.q4_avgout, upd);
        for (auto &r : upd) {
            const uint64_t cnt = r.v & 0x3FFFFull;
            out.push_back({r.k, cnt ? (r.v >> 18) / cnt : 0, r.w});
        }
        consolidate(out);
    }
    return out;
}

// queries/q6.rs: join -> Max per (auction<<20|seller) -> per-seller average
// of the last <= 10 winning bids (the VecDeque fold in cursor order,
// q6.rs:96-110; vals (auction<<20)|price keep that order)
std::vector<Row> q6_step(Oracle &o, const std::vector<dbsp_event> &ev) {
    std::vector<Row> dA, dB;
    for (auto &e : ev) {
        if (e.kind == 1)
            dA.push_back({e.f0, (e.f3 << 36) | (((e.f4 - e.f3) & 0xFFFFull) << 20) |
                                    (e.f1 & 0xFFFFFull),
                          e.w});
        if (e.kind == 2)
            dB.push_back({e.f0, (e.f3 << 27) | (e.f2 & 0x7FFFFFFull), e.w});
    }
    consolidate(dA);
    consolidate(dB);
    std::vector<Row> dWinIn;
    join_raw(dB, o.q6_a_int, DBSP_PROJ_Q6_BID_X_AUC, 0, dWinIn);
    o.q6_b_int = merge(o.q6_b_int, dB);
    join_raw(dA, o.q6_b_int, DBSP_PROJ_Q6_AUC_X_BID, 0, dWinIn);
    o.q6_a_int = merge(o.q6_a_int, dA);
    consolidate(dWinIn);
    std::vector<Row> dWin;
    if (!dWinIn.empty()) {
        o.q6_maxin = merge(o.q6_maxin, dWinIn);
        std::vector<uint64_t> keys;
        for (auto &r : dWinIn)
            if (keys.empty() || keys.back() != r.k) keys.push_back(r.k);
        agg_max_upsert(keys, o.q6_maxin, o.q6_maxout, dWin);
        consolidate(dWin);
        o.q6_maxout = merge(o.q6_maxout, dWin);
    }
    std::vector<Row> dFoldIn;
    for (auto &r : dWin)
        dFoldIn.push_back({r.k & 0xFFFFFull, ((r.k >> 20) << 27) | (r.v & 0x7FFFFFFull), r.w});
    consolidate(dFoldIn);
    std::vector<Row> out;
    if (!dFoldIn.empty()) {
        o.q6_fold_in = merge(o.q6_fold_in, dFoldIn);
        std::vector<uint64_t> keys;
        for (auto &r : dFoldIn)
            if (keys.empty() || keys.back() != r.k) keys.push_back(r.k);
        // per affected seller: avg of the last <= 10 vals' prices; then the
        // upsert against the output integral
        for (uint64_t key : keys) {
            // gather the seller's current run
            std::vector<uint64_t> vals;
            for (auto &r : o.q6_fold_in)
                if (r.k == key && r.w > 0) vals.push_back(r.v);
            if (!vals.empty()) {
                size_t n10 = vals.size() < 10 ? vals.size() : 10;
                uint64_t sum = 0;
                for (size_t t = vals.size() - n10; t < vals.size(); t++)
                    sum += vals[t] & 0x7FFFFFFull;
                out.push_back({key, sum / n10, 1});
            }
            for (auto &r : o.q6_foldout)
                if (r.k == key) out.push_back({key, r.v, -r.w});
        }
        consolidate(out);
        o.q6_foldout = merge(o.q6_foldout, out);
    }
    return out;
}

std::vector<Row> q8_step(Oracle &o, const std::vector<dbsp_event> &ev) {
    constexpr uint64_t TUMBLE_MS = 10'000;  // queries/q8.rs:46
    std::vector<Row> dPT, dAT;
    for (auto &e : ev) {
        if (e.kind == 0)  // (id, name) packed order-preserving as id*1024+name
            dPT.push_back({e.f4, (e.f0 << 10) | (e.f1 & 0x3FFull), e.w});
        if (e.kind == 1) dAT.push_back({e.f3, e.f1, e.w});
    }
    consolidate(dPT);
    consolidate(dAT);
    // watermark_monotonic over auctions_by_time (q8.rs:63-65): last key of the
    // tick's batch (watermark.rs:38-45 stream_fold of max over fast-forwarded key).
    if (!dAT.empty()) o.q8_wm = std::max(o.q8_wm, dAT.back().k - TUMBLE_MS);
    uint64_t rounded = o.q8_wm - o.q8_wm % TUMBLE_MS;
    uint64_t s1 = rounded >= TUMBLE_MS ? rounded - TUMBLE_MS : 0;  // saturating_sub (q8.rs:69)
    uint64_t e1 = rounded;
    // windowed people / auctions (window.rs:144-220); trace excludes this tick
    std::vector<Row> dWPraw, dWAraw;
    window_eval(o.pt_int, dPT, o.q8_have_prev_p, o.q8_s0p, o.q8_e0p, s1, e1, dWPraw);
    o.q8_have_prev_p = true; o.q8_s0p = s1; o.q8_e0p = e1;
    o.pt_int = merge(o.pt_int, dPT);
    window_eval(o.at_int, dAT, o.q8_have_prev_a, o.q8_s0a, o.q8_e0a, s1, e1, dWAraw);
    o.q8_have_prev_a = true; o.q8_s0a = s1; o.q8_e0a = e1;
    o.at_int = merge(o.at_int, dAT);
    // map_index (q8.rs:84): (dt,(id,name)) -> (id, name<<32|dt)
    std::vector<Row> dWP, dWA;
    for (auto &r : dWPraw)
        dWP.push_back({r.v >> 10, ((r.v & 0x3FFull) << 32) | (r.k & 0xFFFFFFFFull), r.w});
    // map (q8.rs:87): (dt,seller) -> (seller,())
    for (auto &r : dWAraw) dWA.push_back({r.v, 0, r.w});
    consolidate(dWP);
    consolidate(dWA);
    std::vector<Row> out;
    join_raw(dWP, o.wa_int, DBSP_PROJ_HI_K_LO_V1RND, TUMBLE_MS, out);  // dWP ⋈ WA_{t-1}
    o.wp_int = merge(o.wp_int, dWP);
    join_raw(dWA, o.wp_int, DBSP_PROJ_HI_K_LO_V2RND, TUMBLE_MS, out);  // WP_t ⋈ dWA
    o.wa_int = merge(o.wa_int, dWA);
    consolidate(out);
    return out;
}

std::vector<Row> q5_step(Oracle &o, const std::vector<dbsp_event> &ev) {
    constexpr uint64_t WIDTH_MS = 10'000, TUMBLE_MS = 2'000;  // queries/q5.rs:74-75
    constexpr uint64_t WM_LAG_MS = 4'000;  // queries/mod.rs:12
    std::vector<Row> dBT;
    for (auto &e : ev)
        if (e.kind == 2) dBT.push_back({e.f3, e.f0, e.w});
    consolidate(dBT);
    if (!dBT.empty()) o.q5_wm = std::max(o.q5_wm, dBT.back().k - WM_LAG_MS);
    uint64_t rounded = o.q5_wm - o.q5_wm % TUMBLE_MS;  // q5.rs:92-97
    uint64_t s1 = rounded >= WIDTH_MS ? rounded - WIDTH_MS : 0;
    uint64_t e1 = rounded;
    std::vector<Row> dWBraw;
    window_eval(o.bt_int, dBT, o.q5_have_prev, o.q5_s0, o.q5_e0, s1, e1, dWBraw);
    o.q5_have_prev = true; o.q5_s0 = s1; o.q5_e0 = e1;
    o.bt_int = merge(o.bt_int, dBT);
    // map (q5.rs:101): (time,auction) -> (auction,()); then weigh(|_|1)
    // (aggregate/mod.rs:297-323) leaves weights unchanged.
    std::vector<Row> dWB;
    for (auto &r : dWBraw) dWB.push_back({r.v, 0, r.w});
    consolidate(dWB);
    // aggregate_linear: integral includes this tick (trace.rs TraceAppend);
    // delta keys = distinct keys of the consolidated delta batch.
    o.wb_int = merge(o.wb_int, dWB);
    std::vector<uint64_t> keys;
    for (auto &r : dWB)
        if (keys.empty() || keys.back() != r.k) keys.push_back(r.k);
    std::vector<Row> dCounts;
    agg_linear_upsert(keys, o.wb_int, o.counts_int, dCounts);
    consolidate(dCounts);
    o.counts_int = merge(o.counts_int, dCounts);
    // max side: map_index (q5.rs:107) -> ((), count)
    std::vector<Row> dMaxIn;
    for (auto &r : dCounts) dMaxIn.push_back({0, r.v, r.w});
    consolidate(dMaxIn);
    o.maxin_int = merge(o.maxin_int, dMaxIn);
    std::vector<Row> dMaxOut;
    if (!dMaxIn.empty()) {
        std::vector<uint64_t> unit_key{0};
        agg_max_upsert(unit_key, o.maxin_int, o.maxout_int, dMaxOut);
        consolidate(dMaxOut);
        o.maxout_int = merge(o.maxout_int, dMaxOut);
    }
    // map (q5.rs:110): ((), max) -> zset keyed max
    std::vector<Row> dMaxZ;
    for (auto &r : dMaxOut) dMaxZ.push_back({r.v, 0, r.w});
    consolidate(dMaxZ);
    // by_count: map_index (q5.rs:116): (auction,count) -> (count,auction)
    std::vector<Row> dBC;
    for (auto &r : dCounts) dBC.push_back({r.v, r.k, r.w});
    consolidate(dBC);
    // final incremental join (q5.rs:118-120)
    std::vector<Row> out;
    join_raw(dMaxZ, o.bc_int, DBSP_PROJ_HI_V2_LO_K, 0, out);  // dMax ⋈ BC_{t-1}
    o.maxz_int = merge(o.maxz_int, dMaxZ);
    join_raw(dBC, o.maxz_int, DBSP_PROJ_HI_V1_LO_K, 0, out);  // Max_t ⋈ dBC
    o.bc_int = merge(o.bc_int, dBC);
    consolidate(out);
    return out;
}

}  // namespace

// ---------------------------------------------------------------------------
// extern "C" surface (ctypes-friendly; used ONLY by tests and bench cpu_baseline)
// ---------------------------------------------------------------------------
extern "C" {

int64_t oracle_consolidate(dbsp_row *rows, int64_t n) {
    std::vector<Row> v(n);
    for (int64_t i = 0; i < n; i++) v[i] = {rows[i].k, rows[i].v, rows[i].w};
    consolidate(v);
    for (size_t i = 0; i < v.size(); i++) rows[i] = {v[i].k, v[i].v, v[i].w};
    return (int64_t)v.size();
}

int64_t oracle_merge(const dbsp_row *a, int64_t na, const dbsp_row *b, int64_t nb,
                     dbsp_row *out) {
    std::vector<Row> va(na), vb(nb);
    for (int64_t i = 0; i < na; i++) va[i] = {a[i].k, a[i].v, a[i].w};
    for (int64_t i = 0; i < nb; i++) vb[i] = {b[i].k, b[i].v, b[i].w};
    auto r = merge(va, vb);
    for (size_t i = 0; i < r.size(); i++) out[i] = {r[i].k, r[i].v, r[i].w};
    return (int64_t)r.size();
}

// Raw (unconsolidated) join output, for kernel-level parity.
int64_t oracle_join(const dbsp_row *delta, int64_t nd, const dbsp_row *trace,
                    int64_t nt, int proj, uint64_t param, dbsp_row *out,
                    int64_t cap) {
    std::vector<Row> vd(nd), vt(nt), vo;
    for (int64_t i = 0; i < nd; i++) vd[i] = {delta[i].k, delta[i].v, delta[i].w};
    for (int64_t i = 0; i < nt; i++) vt[i] = {trace[i].k, trace[i].v, trace[i].w};
    join_raw(vd, vt, (dbsp_proj)proj, param, vo);
    if ((int64_t)vo.size() > cap) return -1;
    for (size_t i = 0; i < vo.size(); i++) out[i] = {vo[i].k, vo[i].v, vo[i].w};
    return (int64_t)vo.size();
}

int64_t oracle_agg_linear_upsert(const uint64_t *keys, int64_t nk,
                                 const dbsp_row *in_trace, int64_t nin,
                                 const dbsp_row *out_trace, int64_t nout,
                                 dbsp_row *out, int64_t cap) {
    std::vector<uint64_t> vk(keys, keys + nk);
    std::vector<Row> vi(nin), vo(nout), res;
    for (int64_t i = 0; i < nin; i++) vi[i] = {in_trace[i].k, in_trace[i].v, in_trace[i].w};
    for (int64_t i = 0; i < nout; i++) vo[i] = {out_trace[i].k, out_trace[i].v, out_trace[i].w};
    agg_linear_upsert(vk, vi, vo, res);
    if ((int64_t)res.size() > cap) return -1;
    for (size_t i = 0; i < res.size(); i++) out[i] = {res[i].k, res[i].v, res[i].w};
    return (int64_t)res.size();
}

int64_t oracle_agg_max_upsert(const uint64_t *keys, int64_t nk,
                              const dbsp_row *in_trace, int64_t nin,
                              const dbsp_row *out_trace, int64_t nout,
                              dbsp_row *out, int64_t cap) {
    std::vector<uint64_t> vk(keys, keys + nk);
    std::vector<Row> vi(nin), vo(nout), res;
    for (int64_t i = 0; i < nin; i++) vi[i] = {in_trace[i].k, in_trace[i].v, in_trace[i].w};
    for (int64_t i = 0; i < nout; i++) vo[i] = {out_trace[i].k, out_trace[i].v, out_trace[i].w};
    agg_max_upsert(vk, vi, vo, res);
    if ((int64_t)res.size() > cap) return -1;
    for (size_t i = 0; i < res.size(); i++) out[i] = {res[i].k, res[i].v, res[i].w};
    return (int64_t)res.size();
}

int64_t oracle_window(const dbsp_row *trace, int64_t nt, const dbsp_row *batch,
                      int64_t nb, int have_prev, uint64_t s0, uint64_t e0,
                      uint64_t s1, uint64_t e1, dbsp_row *out, int64_t cap) {
    std::vector<Row> vt(nt), vb(nb), vo;
    for (int64_t i = 0; i < nt; i++) vt[i] = {trace[i].k, trace[i].v, trace[i].w};
    for (int64_t i = 0; i < nb; i++) vb[i] = {batch[i].k, batch[i].v, batch[i].w};
    window_eval(vt, vb, have_prev != 0, s0, e0, s1, e1, vo);
    if ((int64_t)vo.size() > cap) return -1;
    for (size_t i = 0; i < vo.size(); i++) out[i] = {vo[i].k, vo[i].v, vo[i].w};
    return (int64_t)vo.size();
}

// naive rolling aggregate (the radix-tree operators' semantic oracle:
// rolling_aggregate.rs:235-280 with the linear weight-sum aggregate;
// range_of(ts) = [ts - width, ts], range.rs:93-110, saturating at 0)
int64_t oracle_rolling_agg(const dbsp_row *in, int64_t n, uint64_t width,
                           dbsp_row *out) {
    for (int64_t i = 0; i < n; i++) {
        const uint64_t p = in[i].k, ts = in[i].v;
        const uint64_t t0 = ts >= width ? ts - width : 0;
        int64_t s = 0;
        for (int64_t j = 0; j < n; j++)
            if (in[j].k == p && in[j].v >= t0 && in[j].v <= ts) s += in[j].w;
        out[i] = {p, ts, s};
    }
    return n;
}

uint64_t oracle_xxh3_u64(uint64_t key, uint64_t seed) { return xxh3_u64(key, seed); }

// Incremental distinct (operator/distinct.rs:404-462 at root scope, depth 1:
// distinct_vals = {f(t), f(t-1)}; partial derivative = dist(after)-dist(before))
int64_t oracle_distinct_inc(const dbsp_row *delta, int64_t nd,
                            const dbsp_row *trace, int64_t nt, dbsp_row *out) {
    int64_t o = 0;
    for (int64_t i = 0; i < nd; i++) {
        int64_t before = 0;
        for (int64_t j = 0; j < nt; j++)
            if (trace[j].k == delta[i].k && trace[j].v == delta[i].v)
                before += trace[j].w;
        int64_t after = before + delta[i].w;
        int64_t w = (int64_t)(after > 0) - (int64_t)(before > 0);
        if (w != 0) out[o++] = {delta[i].k, delta[i].v, w};
    }
    return o;
}

// ---- f64-weight variants (config C5).  The oracle is the SEQUENTIAL
// reference: sums accumulate left-to-right in sorted order; the GPU's
// position-fixed tree order is compared against it within the stated
// tolerance (|err| <= 2 ulp * reduction depth).  Zero
// elimination follows F64 is_zero (== 0.0; algebra/floats.rs:24). ----

static inline double bits_to_f64(int64_t w) {
    double d;
    std::memcpy(&d, &w, 8);
    return d;
}
static inline int64_t f64_to_bits(double d) {
    int64_t w;
    std::memcpy(&w, &d, 8);
    return w;
}

int64_t oracle_consolidate_f64(dbsp_row *rows, int64_t n) {
    std::vector<Row> v(n);
    for (int64_t i = 0; i < n; i++) v[i] = {rows[i].k, rows[i].v, rows[i].w};
    std::sort(v.begin(), v.end(), row_lt);
    std::vector<Row> out;
    size_t i = 0;
    while (i < v.size()) {
        size_t j = i;
        double sum = 0.0;
        while (j < v.size() && v[j].k == v[i].k && v[j].v == v[i].v) {
            sum += bits_to_f64(v[j].w);
            j++;
        }
        if (sum != 0.0) out.push_back({v[i].k, v[i].v, f64_to_bits(sum)});
        i = j;
    }
    for (size_t t = 0; t < out.size(); t++)
        rows[t] = {out[t].k, out[t].v, out[t].w};
    return (int64_t)out.size();
}

int64_t oracle_merge_f64(const dbsp_row *a, int64_t na, const dbsp_row *b,
                         int64_t nb, dbsp_row *out) {
    size_t i = 0, j = 0, o = 0;
    while (i < (size_t)na && j < (size_t)nb) {
        Row ra{a[i].k, a[i].v, a[i].w}, rb{b[j].k, b[j].v, b[j].w};
        if (row_lt(ra, rb)) {
            out[o++] = a[i++];
        } else if (row_lt(rb, ra)) {
            out[o++] = b[j++];
        } else {
            double s = bits_to_f64(a[i].w) + bits_to_f64(b[j].w);
            if (s != 0.0) out[o++] = {a[i].k, a[i].v, f64_to_bits(s)};
            i++; j++;
        }
    }
    while (i < (size_t)na) out[o++] = a[i++];
    while (j < (size_t)nb) out[o++] = b[j++];
    return (int64_t)o;
}

// weigh (aggregate/mod.rs:297-323) with f(k,v) = f64_from_bits(v)
int64_t oracle_weigh_f64(const dbsp_row *in, int64_t n, dbsp_row *out) {
    for (int64_t i = 0; i < n; i++) {
        double wp = bits_to_f64((int64_t)in[i].v) * (double)in[i].w;
        out[i] = {in[i].k, 0, f64_to_bits(wp)};
    }
    return n;
}

int64_t oracle_agg_linear_upsert_f64(const uint64_t *keys, int64_t nk,
                                     const dbsp_row *in_trace, int64_t nin,
                                     const dbsp_row *out_trace, int64_t nout_t,
                                     dbsp_row *out, int64_t cap) {
    int64_t o = 0;
    for (int64_t t = 0; t < nk; t++) {
        uint64_t key = keys[t];
        double s = 0.0;
        for (int64_t i = 0; i < nin; i++)
            if (in_trace[i].k == key) s += bits_to_f64(in_trace[i].w);
        std::vector<Row> upd;
        if (s != 0.0) upd.push_back({key, (uint64_t)f64_to_bits(s), 1});
        for (int64_t i = 0; i < nout_t; i++)
            if (out_trace[i].k == key && out_trace[i].w != 0)
                upd.push_back({key, out_trace[i].v, -out_trace[i].w});
        consolidate(upd);
        for (auto &r : upd) {
            if (o >= cap) return -1;
            out[o++] = {r.k, r.v, r.w};
        }
    }
    return o;
}

// --- query drivers ---
void *oracle_query_new(int query) {
    Oracle *o = new Oracle();
    o->query = query;
    return o;
}

void oracle_query_free(void *h) { delete (Oracle *)h; }

// Steps the query with this tick's events; writes consolidated output rows.
// Returns output count, or -1 on overflow, -2 on bad query id.
int64_t oracle_query_step(void *h, const dbsp_event *events, int64_t n,
                          dbsp_row *out, int64_t cap) {
    Oracle *o = (Oracle *)h;
    std::vector<dbsp_event> ev(events, events + n);
    consolidate_events(ev);
    std::vector<Row> res;
    switch (o->query) {
        case 3: res = q3_step(*o, ev); break;
        case 4: res = q4_step(*o, ev); break;
        case 6: res = q6_step(*o, ev); break;
        case 5: res = q5_step(*o, ev); break;
        case 8: res = q8_step(*o, ev); break;
        default: return -2;
    }
    if ((int64_t)res.size() > cap) return -1;
    for (size_t i = 0; i < res.size(); i++) out[i] = {res[i].k, res[i].v, res[i].w};
    return (int64_t)res.size();
}

// q0 passthrough (queries/q0.rs:6-8): output = the consolidated input event
// zset itself.  Returns event count after consolidation.
int64_t oracle_q0_step(const dbsp_event *events, int64_t n, dbsp_event *out,
                       int64_t cap) {
    std::vector<dbsp_event> ev(events, events + n);
    consolidate_events(ev);
    if ((int64_t)ev.size() > cap) return -1;
    std::memcpy(out, ev.data(), ev.size() * sizeof(dbsp_event));
    return (int64_t)ev.size();
}

}  // extern "C"
