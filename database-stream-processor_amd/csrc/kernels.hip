// kernels.hip — hand-written CDNA4 (gfx950) kernels for the DBSP Z-set hot path.
//
// MI355X-native design notes (see DESIGN.md for the full rationale):
//  * Every op here is HBM-bound integer/byte work (no dense contraction ⇒ no
//    MFMA).  The levers are coalescing (SoA u64 columns, 8B/lane loads),
//    LDS staging for the block-local sort/merge phases, and ≫256-workgroup
//    launches with grid-stride loops (256 CUs / 8 XCDs want >2048 blocks
//    before per-XCD L2 effects matter).
//  * Wave width is 64; block size 256 (4 waves) everywhere.
//  * Output sizes are data-dependent (zero-weight elimination,
//    trace/consolidation/mod.rs:32-51) ⇒ count→scan→emit two-phase kernels.
//
// Reference mapping:
//   radix sort + consolidate  <- consolidate_slice / quicksort
//                                (trace/consolidation/mod.rs:91-110, quicksort.rs)
//   merge-path merge          <- ColumnLayerBuilder::push_merge
//                                (trace/layers/column_layer/builders.rs:98-169)
//                                + OrderedBuilder::merge_step (ordered/mod.rs:344-396)
//   join probe/expand         <- Join::eval / JoinTrace::eval
//                                (operator/join.rs:436-473,751-787)
//   aggregate + upsert        <- aggregate/mod.rs:479-547 + upsert.rs:161-208
//   window                    <- time_series/window.rs:144-220
//   shard partition           <- communication/shard.rs:165-199 + hash.rs:9-13

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>

#include "../../include/dbsp_hip.h"
#include "kernels_iface.hpp"

// ---------------------------------------------------------------------------
// Big-buffer cache.  hipMallocAsync's pool hands a freed multi-GB block back
// to the SAME size-class only a couple of allocations later, so trace-scale
// merges (6.5 GB per output column at 1B rows) were re-carving physical
// pages on almost every call at ~9 GB/s — a 720 ms host stall per column
// (profiles/r01_merge_host_stall.txt).  All product allocations go through
// this free-list instead: blocks >= CACHE_MIN bytes are kept and reused
// best-fit (everything runs on the one ctx stream, so reuse is ordered by
// stream order alone); smaller ones pass through to the pool, which handles
// tick-scale buffers fine.  On allocation failure the cache is dropped and
// the carve retried.
// ---------------------------------------------------------------------------
#include <atomic>
#include <map>
#include <mutex>
#include <unordered_map>

namespace dbspk {
namespace {
// 64 KB: originally only trace-scale buffers (>= 32 MB) were cached, but the
// pipelined tick makes ~12 pool alloc/free calls per tick for ~0.1-3 MB
// buffers, and those hipMallocAsync round-trips (~2-4 us each) were the
// largest remaining host cost at the tick boundary — the free-list reuse is
// ~0.2 us.  Footprint cost is bounded by the pow2 rounding (<= 2x per size
// class) against 288 GB of HBM.
// Two cached bands: tick-scale blocks (64 KB .. small-ceiling) whose pool
// round-trips dominate the pipelined tick's host cost, and trace-scale
// blocks (>= 32 MB) whose pool re-carves cost ~100 ms/GB.  The band between
// stays on the stream-ordered pool: C5-scale ticks allocate many MB-range
// buffers and measured 2x slower with them free-listed.
const size_t CACHE_MIN = []() -> size_t {
    const char *v = getenv("DBSP_CACHE_MIN");  // diagnostic A/B knob
    return v ? (size_t)atoll(v) : (64u << 10);
}();
const size_t CACHE_SMALL_CEIL = []() -> size_t {
    const char *v = getenv("DBSP_CACHE_SMALL_CEIL");
    return v ? (size_t)atoll(v) : (4u << 20);
}();
constexpr size_t CACHE_BIG_MIN = 32u << 20;
// Small-band caching is SELF-DISABLING: if any carve fails (trace-scale
// workloads near the memory limit — the hoarded small blocks keep the pool
// from satisfying a multi-GB request), the failure path drops the free list
// and turns the small band off for the rest of the process, so the cost is
// one sync, not one per tick.  Tick-scale engines never fail and keep it.
std::atomic<bool> g_small_cache_on{true};
inline bool cacheable(size_t bytes) {
    if (bytes >= CACHE_BIG_MIN) return true;
    return bytes >= CACHE_MIN && bytes < CACHE_SMALL_CEIL &&
           g_small_cache_on.load(std::memory_order_relaxed);
}
std::mutex g_cache_mu;
std::unordered_map<void *, size_t> g_cache_live;
std::multimap<size_t, void *> g_cache_free;

void cache_drop_locked(hipStream_t s) {
    for (auto &e : g_cache_free) (void)hipFreeAsync(e.second, s);
    g_cache_free.clear();
}
}  // namespace

hipError_t cache_malloc(void **out, size_t bytes, hipStream_t s) {
    if (cacheable(bytes)) {
        // round carves up to the next power of two and accept cached blocks
        // up to 2x the request: a workload with growing buffers (spine
        // merges, trace cascades) then carves only log2(max/min) times
        // instead of once per size, at <= 2x footprint — cheap against
        // 288 GB of HBM, and the page-mapping stalls dominate otherwise
        size_t msb = (size_t)1 << (63 - __builtin_clzll(bytes));
        if (msb < bytes) bytes = msb << 1;
        std::lock_guard<std::mutex> g(g_cache_mu);
        auto it = g_cache_free.lower_bound(bytes);
        // small classes accept only <= 2x (a 64 MB slack there would let a
        // trace-scale block be wasted on a tick-scale request)
        const size_t slack = bytes >= (32u << 20) ? (64u << 20) : 0;
        if (it != g_cache_free.end() && it->first <= 2 * bytes + slack) {
            *out = it->second;
            g_cache_live.emplace(it->second, it->first);
            g_cache_free.erase(it);
            return hipSuccess;
        }
    }
    if (bytes >= CACHE_BIG_MIN) {
        // big-band carve: flush the small band back to the pool first —
        // stream-ordered frees are reusable by this very carve, and a
        // hoarded small band otherwise forces the pool to map fresh pages
        // per multi-GB request (~120 ms each at trace scale).  Tick-scale
        // engines do no big carves in steady state, so their band persists.
        std::lock_guard<std::mutex> g(g_cache_mu);
        for (auto it = g_cache_free.begin();
             it != g_cache_free.end() && it->first < CACHE_BIG_MIN;) {
            (void)hipFreeAsync(it->second, s);
            it = g_cache_free.erase(it);
        }
    }
    static const bool cstats = []() {
        const char *v = getenv("DBSP_CACHE_STATS");
        return v && v[0] == '1';
    }();
    double tc0 = 0;
    if (cstats && bytes >= CACHE_BIG_MIN) {
        struct timespec ts;
        clock_gettime(CLOCK_MONOTONIC, &ts);
        tc0 = ts.tv_sec * 1e3 + ts.tv_nsec * 1e-6;
    }
    hipError_t e = hipMallocAsync(out, bytes, s);
    if (cstats && bytes >= CACHE_BIG_MIN) {
        struct timespec ts;
        clock_gettime(CLOCK_MONOTONIC, &ts);
        double dt = ts.tv_sec * 1e3 + ts.tv_nsec * 1e-6 - tc0;
        size_t nsm = 0, live = 0;
        {
            std::lock_guard<std::mutex> g(g_cache_mu);
            for (auto &kv : g_cache_live)
                if (kv.second < CACHE_BIG_MIN) { nsm++; live += kv.second; }
        }
        fprintf(stderr, "[carve] %.1f MB in %.1f ms (small live: %zu blks %.1f MB)\n",
                bytes / 1048576.0, dt, nsm, live / 1048576.0);
    }
    if (e != hipSuccess) {
        g_small_cache_on.store(false, std::memory_order_relaxed);
        std::lock_guard<std::mutex> g(g_cache_mu);
        cache_drop_locked(s);
        (void)hipStreamSynchronize(s);
        e = hipMallocAsync(out, bytes, s);
    }
    if (e == hipSuccess && cacheable(bytes)) {
        std::lock_guard<std::mutex> g(g_cache_mu);
        g_cache_live.emplace(*out, bytes);
    }
    return e;
}

hipError_t cache_free(void *p, hipStream_t s) {
    if (!p) return hipSuccess;
    {
        std::lock_guard<std::mutex> g(g_cache_mu);
        auto it = g_cache_live.find(p);
        if (it != g_cache_live.end()) {
            const size_t sz = it->second;
            g_cache_live.erase(it);
            if (sz < CACHE_BIG_MIN &&
                !g_small_cache_on.load(std::memory_order_relaxed)) {
                // small band disabled: return to the pool instead of hoarding
            } else {
                g_cache_free.emplace(sz, p);
                return hipSuccess;
            }
        }
    }
    return hipFreeAsync(p, s);
}

void cache_trim(hipStream_t s) {
    std::lock_guard<std::mutex> g(g_cache_mu);
    cache_drop_locked(s);
}

// Explicit policy switch: trace-scale engines (C5's 1B-row spine) turn the
// small band OFF at init — measured on MI355X, routing their tick-scale
// churn away from the stream-ordered pool leaves the pool unable to serve
// the per-tick multi-GB spine carves from retained slabs (0.3 ms -> 240 ms
// per carve), tripling the step.  Tick-scale engines keep it on.
void cache_small_set(bool on, hipStream_t s) {
    if (getenv("DBSP_CACHE_STATS"))
        fprintf(stderr, "[cache] small band %s\n", on ? "on" : "off");
    g_small_cache_on.store(on, std::memory_order_relaxed);
    if (!on) {
        std::lock_guard<std::mutex> g(g_cache_mu);
        for (auto it = g_cache_free.begin();
             it != g_cache_free.end() && it->first < CACHE_BIG_MIN;) {
            (void)hipFreeAsync(it->second, s);
            it = g_cache_free.erase(it);
        }
    }
}
}  // namespace dbspk


#define WAVE 64
#define BLK 256
#define SORT_ITEMS 8
#define SORT_TILE (BLK * SORT_ITEMS)  // 2048 rows per block

#define HIP_CHECK(x)                                                      \
    do {                                                                  \
        hipError_t err_ = (x);                                            \
        if (err_ != hipSuccess) {                                         \
            fprintf(stderr, "HIP error %s at %s:%d\n",                    \
                    hipGetErrorString(err_), __FILE__, __LINE__);         \
            return DBSP_ERR_INTERNAL;                                     \
        }                                                                 \
    } while (0)

static inline int64_t ceil_div(int64_t a, int64_t b) { return (a + b - 1) / b; }

// ---------------------------------------------------------------------------
// generic grid-stride helpers
// ---------------------------------------------------------------------------

static inline dim3 grid_for(int64_t n, int per_block = BLK) {
    int64_t b = ceil_div(n, per_block);
    if (b < 1) b = 1;
    if (b > 16384) b = 16384;  // grid-stride the rest (Guideline 11)
    return dim3((uint32_t)b);
}

__global__ void k_fill_u64(uint64_t *p, uint64_t v, int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        p[i] = v;
}

// ---------------------------------------------------------------------------
// exclusive scan (u64) — 3-kernel recursive device scan
// ---------------------------------------------------------------------------

__device__ inline uint64_t wave_scan_excl(uint64_t x, uint64_t &total) {
    // inclusive shuffle scan over the 64-lane wave, then convert
    uint64_t v = x;
    for (int d = 1; d < WAVE; d <<= 1) {
        uint64_t up = __shfl_up(v, d, WAVE);
        if ((threadIdx.x & (WAVE - 1)) >= d) v += up;
    }
    total = __shfl(v, WAVE - 1, WAVE);
    return v - x;
}

// per-block scan of SORT_TILE elements; writes per-block total
__global__ void k_scan_block(const uint64_t *in, uint64_t *out,
                             uint64_t *block_totals, int64_t n) {
    __shared__ uint64_t wave_tot[BLK / WAVE];
    int64_t base = (int64_t)blockIdx.x * SORT_TILE;
    uint64_t vals[SORT_ITEMS];
    uint64_t thread_sum = 0;
    for (int i = 0; i < SORT_ITEMS; i++) {
        int64_t idx = base + threadIdx.x * SORT_ITEMS + i;
        vals[i] = idx < n ? in[idx] : 0;
        thread_sum += vals[i];
    }
    uint64_t wave_total;
    uint64_t thread_off = wave_scan_excl(thread_sum, wave_total);
    int wid = threadIdx.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == WAVE - 1) wave_tot[wid] = wave_total;
    __syncthreads();
    if (threadIdx.x == 0) {
        uint64_t acc = 0;
        for (int w = 0; w < BLK / WAVE; w++) {
            uint64_t t = wave_tot[w];
            wave_tot[w] = acc;
            acc += t;
        }
        if (block_totals) block_totals[blockIdx.x] = acc;
    }
    __syncthreads();
    uint64_t off = wave_tot[wid] + thread_off;
    for (int i = 0; i < SORT_ITEMS; i++) {
        int64_t idx = base + threadIdx.x * SORT_ITEMS + i;
        if (idx < n) out[idx] = off;
        off += vals[i];
    }
}

__global__ void k_scan_add(uint64_t *out, const uint64_t *block_offsets,
                           int64_t n) {
    int64_t base = (int64_t)blockIdx.x * SORT_TILE;
    uint64_t off = block_offsets[blockIdx.x];
    for (int i = threadIdx.x; i < SORT_TILE; i += BLK) {
        int64_t idx = base + i;
        if (idx < n) out[idx] += off;
    }
}

// host-side recursive exclusive scan; returns total via d_total (device, may be
// null).  in and out may alias.
struct ScanTemp {
    // enough levels for 2048^4 = 1.7e13 elements
    uint64_t *lvl[4] = {nullptr, nullptr, nullptr, nullptr};
    int64_t lvl_n[4] = {0, 0, 0, 0};
};

static dbsp_status scan_exclusive(hipStream_t s, const uint64_t *in,
                                  uint64_t *out, int64_t n, uint64_t *h_total) {
    if (n == 0) {
        if (h_total) *h_total = 0;
        return DBSP_OK;
    }
    // collect level sizes
    int64_t sizes[5];
    int levels = 0;
    int64_t m = n;
    while (true) {
        sizes[levels] = m;
        m = ceil_div(m, SORT_TILE);
        if (sizes[levels] <= SORT_TILE) break;
        levels++;
        if (levels >= 4) return DBSP_ERR_INVALID;
    }
    // allocate block-total arrays per level
    uint64_t *tot[5] = {nullptr};
    for (int l = 0; l <= levels; l++) {
        int64_t nb = ceil_div(sizes[l], SORT_TILE);
        HIP_CHECK(dbspk::cache_malloc((void **)&tot[l], (nb + 1) * sizeof(uint64_t), s));
    }
    // down-sweep: scan each level, producing block totals
    const uint64_t *src = in;
    uint64_t *dst = out;
    for (int l = 0; l <= levels; l++) {
        int64_t nb = ceil_div(sizes[l], SORT_TILE);
        k_scan_block<<<dim3((uint32_t)nb), BLK, 0, s>>>(src, dst, tot[l], sizes[l]);
        src = tot[l];
        dst = tot[l];  // scan totals in place at next level
    }
    // top level: tot[levels] has <= SORT_TILE entries, already scanned in the
    // loop? no — the loop scanned level l's DATA and wrote raw totals to tot[l].
    // The next iteration scans tot[l] (as data) into itself and writes raw
    // totals to tot[l+1].  After the loop, tot[levels] holds RAW totals of the
    // last scanned array; scan it with a single block.
    {
        int64_t nb_last = ceil_div(sizes[levels], SORT_TILE);
        k_scan_block<<<dim3(1), BLK, 0, s>>>(tot[levels], tot[levels], tot[levels] + nb_last,
                                             nb_last);
        // up-sweep: add scanned block offsets back down
        for (int l = levels; l >= 0; l--) {
            int64_t nb = ceil_div(sizes[l], SORT_TILE);
            uint64_t *data = (l == 0) ? out : tot[l - 1];
            if (nb > 1 || l > 0) {
                uint64_t *offs = tot[l];
                k_scan_add<<<dim3((uint32_t)nb), BLK, 0, s>>>(data, offs, sizes[l]);
            }
        }
        // total = tot[levels][nb_last] (one past the scanned entries)
        if (h_total) {
            HIP_CHECK(hipMemcpyAsync(h_total, tot[levels] + nb_last, sizeof(uint64_t),
                                     hipMemcpyDeviceToHost, s));
            HIP_CHECK(hipStreamSynchronize(s));
        }
    }
    for (int l = 0; l <= levels; l++) HIP_CHECK(dbspk::cache_free(tot[l], s));
    return DBSP_OK;
}

// ---------------------------------------------------------------------------
// LSD radix sort of (k, v, w) rows by composite (k major, v minor)
// 8-bit digits; bytes 0..7 = v, 8..15 = k; passes above the detected
// significant byte of max(v)/max(k) are skipped.
// ---------------------------------------------------------------------------

__global__ void k_minmax_u64(const uint64_t *a, const uint64_t *b, int64_t n,
                             uint64_t *out /* [4]: maxA,maxB,minA,minB */) {
    __shared__ uint64_t smax[2];
    __shared__ uint64_t smin[2];
    if (threadIdx.x == 0) {
        smax[0] = 0; smax[1] = 0;
        smin[0] = ~0ull; smin[1] = ~0ull;
    }
    __syncthreads();
    uint64_t ma = 0, mb = 0, na = ~0ull, nb = ~0ull;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        ma = max(ma, a[i]);
        mb = max(mb, b[i]);
        na = min(na, a[i]);
        nb = min(nb, b[i]);
    }
    atomicMax((unsigned long long *)&smax[0], (unsigned long long)ma);
    atomicMax((unsigned long long *)&smax[1], (unsigned long long)mb);
    atomicMin((unsigned long long *)&smin[0], (unsigned long long)na);
    atomicMin((unsigned long long *)&smin[1], (unsigned long long)nb);
    __syncthreads();
    if (threadIdx.x == 0) {
        atomicMax((unsigned long long *)&out[0], (unsigned long long)smax[0]);
        atomicMax((unsigned long long *)&out[1], (unsigned long long)smax[1]);
        atomicMin((unsigned long long *)&out[2], (unsigned long long)smin[0]);
        atomicMin((unsigned long long *)&out[3], (unsigned long long)smin[1]);
    }
}

__device__ inline uint32_t sort_digit(uint64_t k, uint64_t v, int byte,
                                      uint64_t kbase, uint64_t vbase) {
    uint64_t limb = byte < 8 ? v - vbase : k - kbase;
    int sh = (byte & 7) * 8;
    return (uint32_t)((limb >> sh) & 0xFF);
}

// histogram: counts[d * nblocks + blk] (digit-major so one exclusive scan of
// the flat array yields combined digit-base + block-offset)
__global__ void k_radix_hist(const uint64_t *k, const uint64_t *v, int64_t n,
                             int byte, uint64_t kbase, uint64_t vbase,
                             int64_t nblocks, uint64_t *counts) {
    __shared__ uint32_t hist[256];
    for (int i = threadIdx.x; i < 256; i += BLK) hist[i] = 0;
    __syncthreads();
    int64_t base = (int64_t)blockIdx.x * SORT_TILE;
    for (int i = threadIdx.x; i < SORT_TILE; i += BLK) {
        int64_t idx = base + i;
        if (idx < n) atomicAdd(&hist[sort_digit(k[idx], v[idx], byte, kbase, vbase)], 1u);
    }
    __syncthreads();
    for (int d = threadIdx.x; d < 256; d += BLK)
        counts[(int64_t)d * nblocks + blockIdx.x] = hist[d];
}

// stable scatter: in-LDS 8 x 1-bit split of (digit, local idx) pairs, then
// rank within digit run + scanned global base.
__global__ void k_radix_scatter(const uint64_t *k_in, const uint64_t *v_in,
                                const int64_t *w_in, int64_t n, int byte,
                                uint64_t kbase, uint64_t vbase,
                                int64_t nblocks, const uint64_t *scanned,
                                uint64_t *k_out, uint64_t *v_out,
                                int64_t *w_out) {
    __shared__ uint32_t buf_a[SORT_TILE];
    __shared__ uint32_t buf_b[SORT_TILE];
    __shared__ uint32_t run_start[256];
    __shared__ uint64_t wave_tot[BLK / WAVE];

    int64_t base = (int64_t)blockIdx.x * SORT_TILE;
    int64_t tile_n = min((int64_t)SORT_TILE, n - base);
    if (tile_n <= 0) return;

    // load packed (digit << 16 | idx)
    for (int i = threadIdx.x; i < SORT_TILE; i += BLK) {
        uint32_t d = 0xFF;  // pad with max digit so pads sort last
        if (i < tile_n) d = sort_digit(k_in[base + i], v_in[base + i], byte, kbase, vbase);
        buf_a[i] = (d << 16) | (uint32_t)i;
    }
    __syncthreads();

    uint32_t *src = buf_a, *dst = buf_b;
    for (int bit = 0; bit < 8; bit++) {
        // stable 1-bit split: zeros first (keep order), then ones
        // per-thread sequential items for stable ordering
        uint64_t flags[SORT_ITEMS];
        uint64_t tsum = 0;
        for (int i = 0; i < SORT_ITEMS; i++) {
            int idx = threadIdx.x * SORT_ITEMS + i;
            flags[i] = ((src[idx] >> (16 + bit)) & 1) ? 0 : 1;  // 1 == is-zero
            tsum += flags[i];
        }
        uint64_t wtotal;
        uint64_t toff = wave_scan_excl(tsum, wtotal);
        int wid = threadIdx.x / WAVE;
        if ((threadIdx.x & (WAVE - 1)) == WAVE - 1) wave_tot[wid] = wtotal;
        __syncthreads();
        if (threadIdx.x == 0) {
            uint64_t acc = 0;
            for (int w = 0; w < BLK / WAVE; w++) {
                uint64_t t = wave_tot[w];
                wave_tot[w] = acc;
                acc += t;
            }
            run_start[0] = (uint32_t)acc;  // total zeros, reuse smem slot
        }
        __syncthreads();
        uint64_t zeros_before = wave_tot[wid] + toff;
        uint32_t total_zeros = run_start[0];
        __syncthreads();
        for (int i = 0; i < SORT_ITEMS; i++) {
            int idx = threadIdx.x * SORT_ITEMS + i;
            uint32_t e = src[idx];
            uint32_t pos;
            if (flags[i]) {
                pos = (uint32_t)zeros_before;
                zeros_before++;
            } else {
                pos = total_zeros + (uint32_t)(idx - zeros_before);
            }
            dst[pos] = e;
        }
        __syncthreads();
        uint32_t *t = src; src = dst; dst = t;
    }

    // run starts per digit
    for (int i = threadIdx.x; i < 256; i += BLK) run_start[i] = 0xFFFFFFFFu;
    __syncthreads();
    for (int i = threadIdx.x; i < tile_n; i += BLK) {
        uint32_t d = src[i] >> 16;
        if (i == 0 || (src[i - 1] >> 16) != d) run_start[d] = i;
    }
    __syncthreads();

    // scatter rows to scanned global positions
    for (int i = threadIdx.x; i < tile_n; i += BLK) {
        uint32_t e = src[i];
        uint32_t d = e >> 16;
        uint32_t local = (uint32_t)i - run_start[d];
        uint64_t pos = scanned[(int64_t)d * nblocks + blockIdx.x] + local;
        int64_t gi = base + (e & 0xFFFF);
        k_out[pos] = k_in[gi];
        v_out[pos] = v_in[gi];
        w_out[pos] = w_in[gi];
    }
}

// ---------------------------------------------------------------------------
// fused single-workgroup sort+consolidate for small batches (n <= 8192)
//
// A 40k-event tick produces per-stream deltas of a few thousand rows; the
// multi-kernel radix path costs ~30 launches + 3 host syncs there, which
// dominates the tick.  This kernel does the whole consolidate_slice job —
// LSD radix over a permutation held in LDS, then dedup-accumulate and
// zero-drop — in ONE launch with the output length left in device memory.
// 1024 threads = 16 waves on one CU; the permutation entry
// (key16 << 13 | idx, u32) carries the next 16 key bits so digit passes are
// LDS-only, ranks come from bit-sliced wave ballots over strided positions
// (stable by construction), and the entry arrays ping-pong between two
// 32 KiB LDS buffers.  See the kernel's own comment for the details.
// ---------------------------------------------------------------------------

#define FUSE_MAX 8192
#define FUSE_THREADS 1024
#define FUSE_ITEMS (FUSE_MAX / FUSE_THREADS)  // 8
#define FUSE_DIGITS 16                        // 4-bit digits, ranked per pass

// block-wide exclusive scan over FUSE_MAX u32 flags held per-thread
// (16 consecutive items per thread); returns thread's exclusive offset and
// writes the block total to *total (all threads).
__device__ inline uint32_t fuse_scan(uint32_t thread_sum, uint32_t *wave_tot,
                                     uint32_t *total) {
    uint32_t v = thread_sum;
    for (int d = 1; d < WAVE; d <<= 1) {
        uint32_t up = __shfl_up(v, d, WAVE);
        if ((threadIdx.x & (WAVE - 1)) >= d) v += up;
    }
    int wid = threadIdx.x / WAVE;  // 16 waves
    if ((threadIdx.x & (WAVE - 1)) == WAVE - 1) wave_tot[wid] = v;
    __syncthreads();
    if (threadIdx.x < WAVE) {
        // wave 0 scans the 16 wave totals with shuffles (the serial
        // thread-0 loop cost ~1 us of LDS latency per call)
        const int nw = FUSE_THREADS / WAVE;
        uint32_t t = threadIdx.x < nw ? wave_tot[threadIdx.x] : 0;
        uint32_t sc = t;
        for (int d = 1; d < nw; d <<= 1) {
            uint32_t up = __shfl_up(sc, d, WAVE);
            if ((int)threadIdx.x >= d) sc += up;
        }
        if ((int)threadIdx.x < nw) wave_tot[threadIdx.x] = sc - t;
        if ((int)threadIdx.x == nw - 1) wave_tot[nw] = sc;
    }
    __syncthreads();
    uint32_t r = wave_tot[wid] + (v - thread_sum);
    *total = wave_tot[FUSE_THREADS / WAVE];
    __syncthreads();
    return r;
}

template <typename W>
__global__ __launch_bounds__(FUSE_THREADS, 4) void k_sort_cons_small(SortArgs args) {
    // Fused sort+consolidate of one raw batch per workgroup (n <= 8192).
    //
    // LSD radix over 4-bit digits of the concatenated (k - kmin, v - vmin)
    // key, restructured around the 64-wide wavefront:
    //  - positions are STRIDED (row i handled by lane i%64 of wave (i/64)%16
    //    in round i/1024), so every LDS access is conflict-free and ranks
    //    assigned in (digit, round, wave, lane) order equal position order —
    //    i.e. the sort is stable without per-thread counters;
    //  - ranks come from 4 bit-sliced wave ballots per round (no 16x1024
    //    counter array, no per-item LDS counter increments); per-(digit,
    //    round, wave) totals live in a 16*8*16-cell array scanned by the
    //    block scan;
    //  - entries carry the NEXT 16 key bits next to the row index
    //    (entry = key16 << 13 | idx), so digit passes never touch global
    //    memory: one gather per 4-pass group rebuilds the entries.
    // The consolidation reuses the same ballot machinery for head flags and
    // the nonzero compaction.
    const int batch = blockIdx.x;
    const uint64_t *kin = args.kin[batch];
    const uint64_t *vin = args.vin[batch];
    const W *win = (const W *)args.win[batch];
    int64_t n = args.n[batch];
    uint64_t *tk = args.tk[batch];   // scratch (>= n rows)
    uint64_t *tv = args.tv[batch];
    W *tw = (W *)args.tw[batch];
    uint64_t *ok = args.ok[batch];   // output (cap >= n rows)
    uint64_t *ov = args.ov[batch];
    W *ow = (W *)args.ow[batch];
    int64_t *out_len = args.d_len + batch;
    const int64_t n_chain = args.n_dev[batch] ? *args.n_dev[batch] : -1;
    __shared__ alignas(16) uint32_t bufA[FUSE_MAX];
    __shared__ alignas(16) uint32_t bufB[FUSE_MAX];
    __shared__ uint32_t cnt[FUSE_MAX];  // ballot cells / f64 head positions
    __shared__ uint32_t wave_tot[FUSE_THREADS / WAVE + 1];
    __shared__ uint64_t smax[2];
    __shared__ uint64_t smin[2];
    const int tid = threadIdx.x;
    const int wid = tid / WAVE;
    const int lane = tid % WAVE;
    const uint64_t lt = ((uint64_t)1 << lane) - 1;

    if (args.n_dev[batch]) {
        if (n_chain > FUSE_MAX || n_chain < 0) {  // speculation lost
            if (tid == 0) *out_len = -1;
            return;
        }
        n = n_chain;
    }
    if (n == 0) {
        if (tid == 0) *out_len = 0;
        return;
    }

    const int rounds = (int)((n + FUSE_THREADS - 1) / FUSE_THREADS);  // <= 8
    const int nwaves = FUSE_THREADS / WAVE;
    uint32_t *src = bufA, *dst = bufB;

    if (n <= 2048) {
        // ---- bitonic fast path ----
        // Sub-2k tick batches (delta sides, tick outputs) are where the
        // radix path is pass-count-bound (~35 us regardless of n: ~16+
        // digit passes of block-wide ballots and syncs).  A stable LDS
        // bitonic over (k, v, idx) — idx as tiebreak keeps positions
        // identical to the stable radix, so the f64 position-ordered sums
        // stay bit-equal — runs them in ~10-20 us.  Pair-indexed so every
        // thread does a real compare.  bufA/bufB are reused as the u64 key
        // arrays and cnt as the index payload; the result lands in src in
        // the radix entry layout (idx in the low 13 bits) so the
        // consolidation below is shared by both paths.  Beyond 2k the pass
        // count (log^2) overtakes the radix and the radix path stays.
        uint64_t *lk = (uint64_t *)bufA;
        uint64_t *lv = (uint64_t *)bufB;
        uint32_t *lidx = cnt;
        int64_t p2 = 1;
        while (p2 < n) p2 <<= 1;
        for (int64_t i = tid; i < p2; i += FUSE_THREADS) {
            lk[i] = i < n ? kin[i] : ~0ull;
            lv[i] = i < n ? vin[i] : ~0ull;
            lidx[i] = (uint32_t)i;
        }
        __syncthreads();
        const int64_t npairs = p2 >> 1;
        for (int64_t kk = 2; kk <= p2; kk <<= 1) {
            for (int64_t jj = kk >> 1, s = 63 - __clzll(kk >> 1); jj > 0;
                 jj >>= 1, s--) {
                for (int64_t p = tid; p < npairs; p += FUSE_THREADS) {
                    const int64_t i = ((p >> s) << (s + 1)) | (p & (jj - 1));
                    const int64_t l = i + jj;
                    const bool up = (i & kk) == 0;
                    const uint64_t ka = lk[i], kb = lk[l];
                    const uint64_t va = lv[i], vb = lv[l];
                    const uint32_t ia = lidx[i], ib = lidx[l];
                    const bool gt =
                        ka > kb ||
                        (ka == kb && (va > vb || (va == vb && ia > ib)));
                    if (gt == up) {
                        lk[i] = kb; lk[l] = ka;
                        lv[i] = vb; lv[l] = va;
                        lidx[i] = ib; lidx[l] = ia;
                    }
                }
                __syncthreads();
            }
        }
        for (int64_t i = tid; i < n; i += FUSE_THREADS) src[i] = lidx[i];
        __syncthreads();
    } else {
    // ---- significant bits of (max-min) for k and v ----
    if (tid == 0) {
        smax[0] = 0; smax[1] = 0;
        smin[0] = ~0ull; smin[1] = ~0ull;
    }
    __syncthreads();
    {
        uint64_t mk = 0, mv = 0, nk = ~0ull, nv = ~0ull;
        for (int64_t i = tid; i < n; i += FUSE_THREADS) {
            mk = max(mk, kin[i]);
            mv = max(mv, vin[i]);
            nk = min(nk, kin[i]);
            nv = min(nv, vin[i]);
        }
        atomicMax((unsigned long long *)&smax[0], (unsigned long long)mk);
        atomicMax((unsigned long long *)&smax[1], (unsigned long long)mv);
        atomicMin((unsigned long long *)&smin[0], (unsigned long long)nk);
        atomicMin((unsigned long long *)&smin[1], (unsigned long long)nv);
    }
    __syncthreads();
    const uint64_t kbase = smin[0], vbase = smin[1];
    const uint64_t krange = smax[0] - kbase, vrange = smax[1] - vbase;
    int knibs = 0, vnibs = 0;
    while (knibs < 16 && (krange >> (4 * knibs)) != 0) knibs++;
    while (vnibs < 16 && (vrange >> (4 * vnibs)) != 0) vnibs++;
    const int total_nibs = vnibs + knibs;
    const int vbits = 4 * vnibs;

    // bits [s, s+16) of the concatenated key ((k-kbase) << vbits | (v-vbase))
    auto key16_at = [&](uint32_t idx, int s) -> uint32_t {
        const uint64_t vp = vin[idx] - vbase;
        const uint64_t kp = kin[idx] - kbase;
        uint64_t bits = 0;
        if (s < vbits) {
            bits = vp >> s;
            const int up = vbits - s;
            if (up < 64) bits |= kp << up;
        } else {
            bits = kp >> (s - vbits);
        }
        return (uint32_t)(bits & 0xFFFF);
    };

    // ---- digit passes in groups of 4 sharing one entry packing ----
    for (int base = 0; base < total_nibs; base += 4) {
        // (re)pack: entry = next-16-key-bits << 13 | idx  (one global gather)
        for (int64_t i = tid; i < n; i += FUSE_THREADS) {
            const uint32_t idx =
                base == 0 ? (uint32_t)i : (src[i] & 0x1FFFu);
            src[i] = (key16_at(idx, 4 * base) << 13) | idx;
        }
        __syncthreads();
        const int sub_max = min(4, total_nibs - base);
        for (int sub = 0; sub < sub_max; sub++) {
            const int sh = 13 + 4 * sub;
            const int cells_n = FUSE_DIGITS * rounds * nwaves;
            for (int c = tid; c < cells_n; c += FUSE_THREADS) cnt[c] = 0;
            __syncthreads();
            uint32_t myrank[FUSE_ITEMS];
            uint32_t mydig[FUSE_ITEMS];
#pragma unroll
            for (int j = 0; j < FUSE_ITEMS; j++) {
                if (j >= rounds) break;
                const int64_t i = (int64_t)j * FUSE_THREADS + tid;
                const bool act = i < n;
                const uint32_t e = act ? src[i] : 0;
                const uint32_t d = (e >> sh) & 0xF;
                const uint64_t am = __ballot(act);
                const uint64_t b0 = __ballot(d & 1);
                const uint64_t b1 = __ballot(d & 2);
                const uint64_t b2 = __ballot(d & 4);
                const uint64_t b3 = __ballot(d & 8);
                uint64_t m = am;
                m &= (d & 1) ? b0 : ~b0;
                m &= (d & 2) ? b1 : ~b1;
                m &= (d & 4) ? b2 : ~b2;
                m &= (d & 8) ? b3 : ~b3;
                myrank[j] = (uint32_t)__popcll(m & lt);
                mydig[j] = d;
                if (act && (m & lt) == 0)  // lowest lane of this digit group
                    cnt[(d * rounds + j) * nwaves + wid] =
                        (uint32_t)__popcll(m);
                if (!act) myrank[j] = 0xFFFFFFFFu;
            }
            __syncthreads();
            {   // exclusive scan of the (digit, round, wave) cells
                uint32_t local[2];
                const int cpt = (cells_n + FUSE_THREADS - 1) / FUSE_THREADS;
                uint32_t tsum = 0;
                for (int j = 0; j < cpt; j++) {
                    const int f = tid * cpt + j;
                    local[j] = f < cells_n ? cnt[f] : 0;
                    tsum += local[j];
                }
                uint32_t total_unused;
                uint32_t off = fuse_scan(tsum, wave_tot, &total_unused);
                for (int j = 0; j < cpt; j++) {
                    const int f = tid * cpt + j;
                    if (f < cells_n) cnt[f] = off;
                    off += local[j];
                }
            }
            __syncthreads();
#pragma unroll
            for (int j = 0; j < FUSE_ITEMS; j++) {
                if (j >= rounds) break;
                const int64_t i = (int64_t)j * FUSE_THREADS + tid;
                if (i < n) {
                    const uint32_t r =
                        cnt[(mydig[j] * rounds + j) * nwaves + wid] +
                        myrank[j];
                    dst[r] = src[i];
                }
            }
            __syncthreads();
            uint32_t *t = src; src = dst; dst = t;
        }
    }

    if (total_nibs == 0) {  // all rows share one (k,v): identity entries
        for (int64_t i = tid; i < n; i += FUSE_THREADS) src[i] = (uint32_t)i;
        __syncthreads();
    }
    }  // end radix path

    // ---- consolidate: head flags (full-key compares via one gather) ----
    // seg ids into dst (the free buffer); ballot cells per (round, wave)
    uint32_t headmask[FUSE_ITEMS];  // per round: this wave's head ballot
    uint32_t myhead[FUSE_ITEMS];
    {
        const int cells_n = rounds * nwaves;
        for (int c = tid; c < cells_n; c += FUSE_THREADS) cnt[c] = 0;
        __syncthreads();
#pragma unroll
        for (int j = 0; j < FUSE_ITEMS; j++) {
            if (j >= rounds) break;
            const int64_t i = (int64_t)j * FUSE_THREADS + tid;
            uint32_t h = 0;
            if (i < n) {
                const uint32_t idx = src[i] & 0x1FFFu;
                if (i == 0) h = 1;
                else {
                    const uint32_t pidx = src[i - 1] & 0x1FFFu;
                    h = (kin[idx] != kin[pidx]) || (vin[idx] != vin[pidx]);
                }
            }
            myhead[j] = h;
            const uint64_t m = __ballot(h != 0);
            headmask[j] = (uint32_t)__popcll(m & lt);  // rank among heads
            if (i < n && lane == 0)
                cnt[j * nwaves + wid] = (uint32_t)__popcll(m);
        }
        __syncthreads();
    }
    uint32_t nseg;
    {   // scan head cells
        const int cells_n = rounds * nwaves;
        uint32_t local[1];
        uint32_t tsum = 0;
        if (tid < cells_n) {
            local[0] = cnt[tid];
            tsum = local[0];
        }
        uint32_t off = fuse_scan(tsum, wave_tot, &nseg);
        if (tid < cells_n) cnt[tid] = off;
        __syncthreads();
    }
    // seg of row i = heads at-or-before i, minus 1
#pragma unroll
    for (int j = 0; j < FUSE_ITEMS; j++) {
        if (j >= rounds) break;
        const int64_t i = (int64_t)j * FUSE_THREADS + tid;
        if (i < n)
            dst[i] = cnt[j * nwaves + wid] + headmask[j] + myhead[j] - 1;
    }
    __syncthreads();
    if constexpr (sizeof(W) == 8 && (W)0.5 == (W)0) {  // integer weights
        for (uint32_t i = tid; i < nseg; i += FUSE_THREADS) tw[i] = 0;
        __syncthreads();
#pragma unroll
        for (int j = 0; j < FUSE_ITEMS; j++) {
            if (j >= rounds) break;
            const int64_t i = (int64_t)j * FUSE_THREADS + tid;
            if (i < n) {
                const uint32_t idx = src[i] & 0x1FFFu;
                const uint32_t seg = dst[i];
                atomicAdd((unsigned long long *)&tw[seg],
                          (unsigned long long)win[idx]);
                if (myhead[j]) {
                    tk[seg] = kin[idx];
                    tv[seg] = vin[idx];
                }
            }
        }
        __syncthreads();
    } else {
        // f64: deterministic segmented tree sum in sorted-position order
        // (identical order to the stable sort's positions, so results are
        // bit-equal to the previous per-thread-counter kernel).  cnt (8192
        // cells) now holds each segment's head position.
#pragma unroll
        for (int j = 0; j < FUSE_ITEMS; j++) {
            if (j >= rounds) break;
            const int64_t i = (int64_t)j * FUSE_THREADS + tid;
            if (i < n) {
                const uint32_t idx = src[i] & 0x1FFFu;
                tw[i] = win[idx];
                if (myhead[j]) {
                    const uint32_t seg = dst[i];
                    cnt[seg] = (uint32_t)i;
                    tk[seg] = kin[idx];
                    tv[seg] = vin[idx];
                }
            }
        }
        __syncthreads();
        for (int64_t d = 1; d < n; d <<= 1) {
#pragma unroll
            for (int j = 0; j < FUSE_ITEMS; j++) {
                if (j >= rounds) break;
                const int64_t i = (int64_t)j * FUSE_THREADS + tid;
                if (i < n && i + d < n && dst[i] == dst[i + d]) {
                    const uint32_t rel = (uint32_t)i - cnt[dst[i]];
                    if ((rel & (2 * d - 1)) == 0) tw[i] += tw[i + d];
                }
            }
            __syncthreads();
        }
        W my_tot[FUSE_ITEMS];
#pragma unroll
        for (int j = 0; j < FUSE_ITEMS; j++) {
            if (j >= rounds) break;
            const int64_t i = (int64_t)j * FUSE_THREADS + tid;
            my_tot[j] = (i < n && myhead[j]) ? tw[i] : (W)0;
        }
        __syncthreads();
#pragma unroll
        for (int j = 0; j < FUSE_ITEMS; j++) {
            if (j >= rounds) break;
            const int64_t i = (int64_t)j * FUSE_THREADS + tid;
            if (i < n && myhead[j]) tw[dst[i]] = my_tot[j];
        }
        __syncthreads();
    }
    // ---- drop zero-weight segments (ballot compaction over seg ids) ----
    {
        const int seg_rounds = (int)((nseg + FUSE_THREADS - 1) / FUSE_THREADS);
        const int cells_n = seg_rounds * nwaves;
        for (int c = tid; c < cells_n + 1; c += FUSE_THREADS) cnt[c] = 0;
        __syncthreads();
        uint32_t nzrank[FUSE_ITEMS];
        uint32_t mynz[FUSE_ITEMS];
#pragma unroll
        for (int j = 0; j < FUSE_ITEMS; j++) {
            if (j >= seg_rounds) break;
            const uint32_t i = (uint32_t)j * FUSE_THREADS + tid;
            const uint32_t nz = (i < nseg && tw[i] != (W)0) ? 1 : 0;
            mynz[j] = nz;
            const uint64_t m = __ballot(nz != 0);
            nzrank[j] = (uint32_t)__popcll(m & lt);
            if (i < nseg && lane == 0)
                cnt[j * nwaves + wid] = (uint32_t)__popcll(m);
        }
        __syncthreads();
        uint32_t nout;
        {
            uint32_t local = 0;
            if (tid < cells_n) local = cnt[tid];
            uint32_t off = fuse_scan(local, wave_tot, &nout);
            if (tid < cells_n) cnt[tid] = off;
            __syncthreads();
        }
#pragma unroll
        for (int j = 0; j < FUSE_ITEMS; j++) {
            if (j >= seg_rounds) break;
            const uint32_t i = (uint32_t)j * FUSE_THREADS + tid;
            if (i < nseg && mynz[j]) {
                const uint32_t p = cnt[j * nwaves + wid] + nzrank[j];
                ok[p] = tk[i];
                ov[p] = tv[i];
                ow[p] = tw[i];
            }
        }
        if (tid == 0) *out_len = (int64_t)nout;
    }
}




// ---------------------------------------------------------------------------
// dense-range sort+consolidate.  A Nexmark tick's delta usually has a TINY
// key box: q5 bids cover ~4-8 distinct ms timestamps x the ~100-auction
// in-flight window (generator/auctions.rs:112-123), so instead of sorting
// 37k rows we scatter weights into a (krange+1)*(vrange+1) histogram with
// many blocks and emit the nonzero cells in index order — already sorted and
// consolidated.  Integer weights only (atomic accumulation).
// ---------------------------------------------------------------------------

__global__ void k_minmax_rows(const uint64_t *k, const uint64_t *v, int64_t n,
                              unsigned long long *mm /* kmin,vmin,kmax,vmax */,
                              const int64_t *n_dev = nullptr) {
    if (n_dev) n = *n_dev;
    uint64_t mk = 0, mv = 0, nk = ~0ull, nv = ~0ull;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        mk = max(mk, k[i]);
        mv = max(mv, v[i]);
        nk = min(nk, k[i]);
        nv = min(nv, v[i]);
    }
    __shared__ unsigned long long s[4];
    if (threadIdx.x == 0) { s[0] = ~0ull; s[1] = ~0ull; s[2] = 0; s[3] = 0; }
    __syncthreads();
    atomicMin(&s[0], (unsigned long long)nk);
    atomicMin(&s[1], (unsigned long long)nv);
    atomicMax(&s[2], (unsigned long long)mk);
    atomicMax(&s[3], (unsigned long long)mv);
    __syncthreads();
    if (threadIdx.x == 0) {
        atomicMin(&mm[0], s[0]);
        atomicMin(&mm[1], s[1]);
        atomicMax(&mm[2], s[2]);
        atomicMax(&mm[3], s[3]);
    }
}

__global__ void k_hist_add(const uint64_t *k, const uint64_t *v,
                           const int64_t *w, int64_t n, uint64_t kbase,
                           uint64_t vbase, uint64_t vspan,
                           unsigned long long *wbuf) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        atomicAdd(&wbuf[(k[i] - kbase) * vspan + (v[i] - vbase)],
                  (unsigned long long)w[i]);
}

__global__ void k_hist_flags(const unsigned long long *wbuf, int64_t r,
                             uint64_t *flags) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < r;
         i += (int64_t)gridDim.x * blockDim.x)
        flags[i] = wbuf[i] != 0;
}

__global__ void k_hist_total(const unsigned long long *wbuf,
                             const uint64_t *flag_scan, int64_t r,
                             int64_t *out) {
    *out = (int64_t)(flag_scan[r - 1] + (wbuf[r - 1] != 0 ? 1 : 0));
}

__global__ void k_hist_emit(const unsigned long long *wbuf,
                            const uint64_t *flag_scan, int64_t r,
                            uint64_t kbase, uint64_t vbase, uint64_t vspan,
                            uint64_t *ok, uint64_t *ov, int64_t *ow) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < r;
         i += (int64_t)gridDim.x * blockDim.x) {
        unsigned long long wv = wbuf[i];
        if (wv != 0) {
            uint64_t p = flag_scan[i];
            ok[p] = kbase + (uint64_t)i / vspan;
            ov[p] = vbase + (uint64_t)i % vspan;
            ow[p] = (int64_t)wv;
        }
    }
}

// ---------------------------------------------------------------------------
// consolidate sorted rows: head flags -> scan -> segment-sum -> nonzero compact
// ---------------------------------------------------------------------------

__global__ void k_head_flags(const uint64_t *k, const uint64_t *v, int64_t n,
                             uint64_t *flags) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        flags[i] = (i == 0) || k[i] != k[i - 1] || v[i] != v[i - 1];
}

// seg[i] implied by scanned flags; accumulate weights and write unique (k,v)
__global__ void k_seg_accum(const uint64_t *k, const uint64_t *v,
                            const int64_t *w, const uint64_t *flag_scan,
                            const uint64_t *flags, int64_t n, uint64_t *ok,
                            uint64_t *ov, int64_t *ow) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint64_t seg = flags[i] ? flag_scan[i] : flag_scan[i] - 1;
        atomicAdd((unsigned long long *)&ow[seg], (unsigned long long)w[i]);
        if (flags[i]) {
            ok[seg] = k[i];
            ov[seg] = v[i];
        }
    }
}

// f64 big-path consolidate support: deterministic segmented tree sum over the
// sorted weight array (fixed position order; |err| <= 2 ulp * ceil(log2 run)).
__device__ inline uint64_t seg_of(const uint64_t *fscan, const uint64_t *flags,
                                  int64_t i) {
    return flags[i] ? fscan[i] : fscan[i] - 1;
}

__global__ void k_seg_headpos(const uint64_t *flags, const uint64_t *fscan,
                              int64_t n, uint64_t *hp) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        if (flags[i]) hp[fscan[i]] = (uint64_t)i;
}

__global__ void k_seg_tree_round(double *w, const uint64_t *flags,
                                 const uint64_t *fscan, const uint64_t *hp,
                                 int64_t n, int64_t d) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        if (i + d < n) {
            uint64_t sg = seg_of(fscan, flags, i);
            if (sg == seg_of(fscan, flags, i + d) &&
                (((uint64_t)i - hp[sg]) & (uint64_t)(2 * d - 1)) == 0)
                w[i] += w[i + d];
        }
    }
}

__global__ void k_seg_collect_f64(const uint64_t *k, const uint64_t *v,
                                  const double *w, const uint64_t *flags,
                                  const uint64_t *fscan, int64_t n,
                                  uint64_t *ok, uint64_t *ov, double *ow) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        if (flags[i]) {
            uint64_t sg = fscan[i];
            ok[sg] = k[i];
            ov[sg] = v[i];
            ow[sg] = w[i];  // tree total sits at the head position
        }
    }
}

__global__ void k_nonzero_flags_f64(const double *w, int64_t n,
                                    uint64_t *flags) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        flags[i] = w[i] != 0.0;  // -0.0 compares equal to 0.0: dropped
}

__global__ void k_nonzero_flags(const int64_t *w, int64_t n, uint64_t *flags) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        flags[i] = w[i] != 0;
}

__global__ void k_compact(const uint64_t *k, const uint64_t *v, const int64_t *w,
                          const uint64_t *flags, const uint64_t *flag_scan,
                          int64_t n, uint64_t *ok, uint64_t *ov, int64_t *ow) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        if (flags[i]) {
            uint64_t pos = flag_scan[i];
            ok[pos] = k[i];
            ov[pos] = v[i];
            ow[pos] = w[i];
        }
    }
}

// ---------------------------------------------------------------------------
// merge-path merge of two consolidated batches
// (the trace-merge hot kernel; replaces push_merge/merge_step)
// ---------------------------------------------------------------------------

__device__ inline bool row_lt(uint64_t ak, uint64_t av, uint64_t bk, uint64_t bv) {
    return ak != bk ? ak < bk : av < bv;
}
__device__ inline bool row_eq(uint64_t ak, uint64_t av, uint64_t bk, uint64_t bv) {
    return ak == bk && av == bv;
}

// merge-path split: find (ai, bi), ai + bi = diag, such that
// a[0..ai) and b[0..bi) are exactly the first `diag` outputs
// (tie-break: a before b).
__device__ inline void merge_path(const uint64_t *ak, const uint64_t *av,
                                  int64_t na, const uint64_t *bk,
                                  const uint64_t *bv, int64_t nb, int64_t diag,
                                  int64_t &ai, int64_t &bi) {
    int64_t lo = max((int64_t)0, diag - nb);
    int64_t hi = min(diag, na);
    while (lo < hi) {
        int64_t mid = (lo + hi) / 2;         // candidate ai
        int64_t j = diag - mid - 1;          // b index to compare
        // a[mid] vs b[j]: if a[mid] <= b[j] (tie to a), we can take more a
        if (!row_lt(bk[j], bv[j], ak[mid], av[mid]))  // b[j] >= a[mid]
            lo = mid + 1;
        else
            hi = mid;
    }
    ai = lo;
    bi = diag - lo;
}

// adjust a split so an equal (k,v) pair is never separated: with a-before-b
// tie-breaking, the pair (a[ai-1], b[bi]) is the only possible split pair.
__device__ inline void adjust_split(const uint64_t *ak, const uint64_t *av,
                                    const uint64_t *bk, const uint64_t *bv,
                                    int64_t na, int64_t nb, int64_t &ai,
                                    int64_t &bi) {
    if (ai > 0 && bi < nb && row_eq(ak[ai - 1], av[ai - 1], bk[bi], bv[bi])) bi++;
}


// ---------------------------------------------------------------------------
// join: delta x trace with projection (count/emit)
// ---------------------------------------------------------------------------

__device__ inline int64_t lower_bound_k(const uint64_t *k, int64_t n,
                                        uint64_t key) {
    int64_t lo = 0, hi = n;
    while (lo < hi) {
        int64_t mid = (lo + hi) / 2;
        if (k[mid] < key) lo = mid + 1; else hi = mid;
    }
    return lo;
}
__device__ inline int64_t upper_bound_k(const uint64_t *k, int64_t n,
                                        uint64_t key) {
    int64_t lo = 0, hi = n;
    while (lo < hi) {
        int64_t mid = (lo + hi) / 2;
        if (k[mid] <= key) lo = mid + 1; else hi = mid;
    }
    return lo;
}

__global__ void k_join_count(const uint64_t *dk, int64_t nd, const uint64_t *tk,
                             int64_t nt, uint64_t *counts, uint64_t *starts) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nd;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint64_t key = dk[i];
        int64_t lo = lower_bound_k(tk, nt, key);
        int64_t hi = lo;
        // galloping upper bound (advance.rs:11-60 analog): trace runs per key
        // are usually short; probe forward before a full binary search
        if (lo < nt && tk[lo] == key) {
            int64_t step = 1;
            hi = lo;
            while (hi + step < nt && tk[hi + step] == key) { hi += step; step <<= 1; }
            hi = upper_bound_k(tk + hi, min(step, nt - hi), key) + hi;
        }
        counts[i] = hi - lo;
        starts[i] = lo;
    }
}

// Returns whether the pair is a real match (join_func filters, e.g. q4's
// bid-validity window): an invalid pair still occupies its counted output
// slot but emits weight 0, which the tick's consolidate then drops.
__device__ inline bool proj_out(int proj, uint64_t param, uint64_t k,
                                uint64_t v1, uint64_t v2, uint64_t &hi,
                                uint64_t &lo) {
    switch (proj) {
        case DBSP_PROJ_Q4_BID_X_AUC: {
            // delta = bid (v1 = bid_dt<<20 | price), trace = auction
            // (v2 = a_dt<<28 | (expires-a_dt)<<4 | category-10); emit
            // ((auction<<4)|cat, price) when the bid falls in the auction's
            // validity window (q4.rs:58-68)
            const uint64_t bid_dt = v1 >> 27, price = v1 & 0x7FFFFFFull;
            const uint64_t a_dt = v2 >> 28, dur = (v2 >> 4) & 0xFFFFFFull;
            hi = (k << 4) | (v2 & 0xFull);
            lo = price;
            return bid_dt >= a_dt && bid_dt <= a_dt + dur;
        }
        case DBSP_PROJ_Q6_BID_X_AUC: {
            // q6.rs:60-80: delta = bid (v1 = bid_dt<<27|price), trace =
            // auction (v2 = a_dt<<34 | (expires-a_dt)<<20 | seller); emit
            // ((auction<<20)|seller, price) in the validity window
            const uint64_t bid_dt = v1 >> 27, price = v1 & 0x7FFFFFFull;
            const uint64_t a_dt = v2 >> 36, dur = (v2 >> 20) & 0xFFFFull;
            hi = (k << 20) | (v2 & 0xFFFFFull);
            lo = price;
            return bid_dt >= a_dt && bid_dt <= a_dt + dur;
        }
        case DBSP_PROJ_Q6_AUC_X_BID: {
            const uint64_t bid_dt = v2 >> 27, price = v2 & 0x7FFFFFFull;
            const uint64_t a_dt = v1 >> 36, dur = (v1 >> 20) & 0xFFFFull;
            hi = (k << 20) | (v1 & 0xFFFFFull);
            lo = price;
            return bid_dt >= a_dt && bid_dt <= a_dt + dur;
        }
        case DBSP_PROJ_Q4_AUC_X_BID: {  // sides swapped
            const uint64_t bid_dt = v2 >> 27, price = v2 & 0x7FFFFFFull;
            const uint64_t a_dt = v1 >> 28, dur = (v1 >> 4) & 0xFFFFFFull;
            hi = (k << 4) | (v1 & 0xFull);
            lo = price;
            return bid_dt >= a_dt && bid_dt <= a_dt + dur;
        }
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
        default: hi = 0; lo = 0; break;
    }
    return true;
}

// emit over the OUTPUT index space: balanced regardless of per-key skew
// (hot-seller keys have thousands of matches; one-thread-per-delta-row would
// serialize them)
__global__ void k_join_emit(const uint64_t *dk, const uint64_t *dv,
                            const int64_t *dw, int64_t nd, const uint64_t *tv,
                            const int64_t *tw, const uint64_t *offsets,
                            const uint64_t *starts, int64_t n_out, int proj,
                            uint64_t param, uint64_t *ok, uint64_t *ov,
                            int64_t *ow) {
    for (int64_t o = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; o < n_out;
         o += (int64_t)gridDim.x * blockDim.x) {
        // find delta row: offsets is the exclusive scan of counts (length nd)
        int64_t lo = 0, hi = nd;
        while (lo < hi) {  // upper_bound(offsets, o) - 1
            int64_t mid = (lo + hi) / 2;
            if (offsets[mid] <= (uint64_t)o) lo = mid + 1; else hi = mid;
        }
        int64_t i = lo - 1;
        int64_t j = o - (int64_t)offsets[i];
        int64_t t = (int64_t)starts[i] + j;
        uint64_t hi_o, lo_o;
        const bool valid = proj_out(proj, param, dk[i], dv[i], tv[t], hi_o, lo_o);
        ok[o] = hi_o;
        ov[o] = lo_o;
        ow[o] = valid ? dw[i] * tw[t] : 0;
    }
}

// ---------------------------------------------------------------------------
// large-merge path (the HBM roofline kernel): per-BLOCK merge-path partition
// (one global binary search per 4096-row tile, not per thread), then
// LDS-staged count and emit passes — coalesced HBM streams in, per-thread
// merge walks entirely in LDS.  Replaces the naive per-thread global
// partitioning, which at 1B rows spent its time on ~14G random 8B probes
// (measured 53 GB/s).
// ---------------------------------------------------------------------------

#define MP_TILE 1024
#define MP_TILE_OP 2048  // single-pass lookback: wider tiles than the two-pass shorten the chain
#define MP_THREADS 256
#define MP_ITEMS (MP_TILE / MP_THREADS)  // 16 diagonals per thread

__global__ void k_mp_partition(const uint64_t *ak, const uint64_t *av,
                               int64_t na, const uint64_t *bk,
                               const uint64_t *bv, int64_t nb, int64_t nblocks,
                               int64_t tile, int64_t *pa, int64_t *pb) {
    int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
    if (i > nblocks) return;
    int64_t d = min(i * tile, na + nb);
    int64_t ai, bi;
    merge_path(ak, av, na, bk, bv, nb, d, ai, bi);
    adjust_split(ak, av, bk, bv, na, nb, ai, bi);
    pa[i] = ai;
    pb[i] = bi;
}

// local merge-path over the LDS-staged tile (a rows at [0,naL), b rows at
// [naL, naL+nbL) of the lk/lv arrays)
__device__ inline void merge_path_lds(const uint64_t *lk, const uint64_t *lv,
                                      int64_t naL, int64_t nbL, int64_t diag,
                                      int64_t &ai, int64_t &bi) {
    int64_t lo = max((int64_t)0, diag - nbL);
    int64_t hi = min(diag, naL);
    while (lo < hi) {
        int64_t mid = (lo + hi) / 2;
        int64_t j = naL + diag - mid - 1;
        if (!row_lt(lk[j], lv[j], lk[mid], lv[mid])) lo = mid + 1;
        else hi = mid;
    }
    ai = lo;
    bi = diag - lo;
}

__device__ inline void adjust_split_lds(const uint64_t *lk, const uint64_t *lv,
                                        int64_t naL, int64_t nbL, int64_t &ai,
                                        int64_t &bi) {
    if (ai > 0 && bi < nbL &&
        row_eq(lk[ai - 1], lv[ai - 1], lk[naL + bi], lv[naL + bi]))
        bi++;
}


// ---------------------------------------------------------------------------
// single-pass large merge: decoupled-lookback offsets replace the
// count->scan->emit pipeline (removes the 16 B/row count re-read, the device
// scan, and the mid-pipeline host sync).  Blocks take a ticket (atomic) so
// virtual block order equals launch order (forward progress for lookback);
// per-block state is one 8-byte {flag,value} granule written/read with
// agent-scope relaxed atomics (single-granule R2 form of the CDNA4
// inter-workgroup recipe — no fences needed for an 8-byte payload-is-flag).
// State layout: state[0] = ticket; state[1+vb] = granule
//   granule = value<<2 | flag;  flag: 0 invalid, 1 aggregate, 2 prefix
// ---------------------------------------------------------------------------


// stage `n` rows of (k,v[,w]) into LDS at base `off`: pairs of u64 move as
// ulonglong2 when both sides are 16 B aligned; odd head/tail go scalar
template <bool EMIT, typename W>
__device__ inline void stage_run(uint64_t *lk, uint64_t *lv, W *lw,
                                 int64_t off, const uint64_t *gk,
                                 const uint64_t *gv, const W *gw, int64_t n,
                                 int tid) {
    // align the GLOBAL side to 16 B (columns share base alignment, so one
    // parity covers k, v and w)
    const int64_t head = ((uintptr_t)gk & 15) != 0 ? 1 : 0;
    if (head && n > 0 && tid == 0) {
        lk[off] = gk[0];
        lv[off] = gv[0];
        if (EMIT) lw[off] = gw[0];
    }
    const int64_t pairs = n > head ? (n - head) >> 1 : 0;
    const ulonglong2 *gk2 = (const ulonglong2 *)(gk + head);
    const ulonglong2 *gv2 = (const ulonglong2 *)(gv + head);
    for (int64_t p = tid; p < pairs; p += MP_THREADS) {
        ulonglong2 kk = gk2[p];
        ulonglong2 vv = gv2[p];
        int64_t i = off + head + 2 * p;
        lk[i] = kk.x; lk[i + 1] = kk.y;
        lv[i] = vv.x; lv[i + 1] = vv.y;
    }
    if (EMIT) {
        const ulonglong2 *gw2 = (const ulonglong2 *)(gw + head);
        for (int64_t p = tid; p < pairs; p += MP_THREADS) {
            ulonglong2 ww = gw2[p];
            int64_t i = off + head + 2 * p;
            lw[i] = ((const W *)&ww)[0];
            lw[i + 1] = ((const W *)&ww)[1];
        }
    }
    const int64_t tail = head + 2 * pairs;
    if (tail < n && tid == 0) {
        lk[off + tail] = gk[tail];
        lv[off + tail] = gv[tail];
        if (EMIT) lw[off + tail] = gw[tail];
    }
}

typedef __attribute__((address_space(1))) unsigned long long gu64_t;

// ---------------------------------------------------------------------------
// single-pass merge v2.  One FUSED branchless walk per tile (the round-1
// variant walked the tile twice: once to count, once to emit); the walk
// captures per-step decisions (advance-a / equal-pair / keep) as mask bits in
// one register and the emit replays them with pure LDS reads — no compares.
// The decoupled lookback runs WAVE-wide: all 64 lanes of wave 0 load a
// 64-entry predecessor window per round instead of lane 0 walking the chain
// serially (the round-1 chain at ~1M tiles was the measured bottleneck of
// the one-pass variant).  Tile = MP_TILE (1024) -> 24.6 KB LDS -> 6
// blocks/CU, matching the two-pass emit's occupancy.
// State layout: state[0] = ticket; state[1] = poison; state[2+vb] = granule
//   granule = value<<2 | flag;  flag: 0 invalid, 1 aggregate, 2 prefix
// ---------------------------------------------------------------------------

// One merge-tile kernel, three offset modes (MODE):
//   0  single-pass: decoupled-lookback prefix (state = ticket/poison/granules)
//   1  count phase: write the tile's output count to state[vb], no emit
//   2  emit phase: read the tile's pre-scanned output offset from state[vb]
// Modes 1+2 form the default two-pass pipeline (count -> device scan ->
// emit); mode 0 reads each input byte once but its lookback protocol costs
// more than the count re-read at 1B-row scale (measured: granule-load
// throughput-bound — wider windows and preloading both made it slower).
template <typename W, int TILE, int MODE, bool SKIP_LB = false>
__global__ __launch_bounds__(MP_THREADS, 4) void k_mp_merge_onepass(
    const uint64_t *ak, const uint64_t *av, const W *aw, int64_t na,
    const uint64_t *bk, const uint64_t *bv, const W *bw, int64_t nb,
    const int64_t *pa, const int64_t *pb, unsigned long long *state,
    int64_t nblocks, uint64_t *ok, uint64_t *ov, W *ow) {
    // steps per thread: a split adjustment can grow a tile one row past TILE
    constexpr int OP_ITEMS = (TILE + MP_THREADS) / MP_THREADS;
    constexpr bool STAGE_W = MODE != 1;  // count phase: weights only on eq
    extern __shared__ __attribute__((aligned(16))) char smem[];
    uint64_t *lk = (uint64_t *)smem;
    uint64_t *lv = lk + (TILE + 2);
    W *lw = (W *)(lv + (TILE + 2));
    __shared__ uint32_t wt[MP_THREADS / WAVE + 1];
    __shared__ unsigned long long sh_vb;
    __shared__ unsigned long long sh_prefix;
    const int tid = threadIdx.x;
    if (MODE == 0) {
        // ticket: virtual block id in launch order (lookback progress)
        if (tid == 0) sh_vb = atomicAdd(state, 1ull);
        __syncthreads();
    } else if (tid == 0) {
        sh_vb = (unsigned long long)blockIdx.x;
    }
    if (MODE != 0) __syncthreads();
    const int64_t vb = (int64_t)sh_vb;
    const int64_t pa0 = pa[vb], pa1 = pa[vb + 1];
    const int64_t pb0 = pb[vb], pb1 = pb[vb + 1];
    const int naL = (int)(pa1 - pa0), nbL = (int)(pb1 - pb0);
    const int totL = naL + nbL;
    stage_run<STAGE_W>(lk, lv, lw, 0, ak + pa0, av + pa0, aw + pa0, naL, tid);
    stage_run<STAGE_W>(lk, lv, lw, naL, bk + pb0, bv + pb0, bw + pb0, nbL,
                       tid);
    __syncthreads();
    const int items = (totL + MP_THREADS - 1) / MP_THREADS;
    const int d0 = min(tid * items, totL);
    const int d1 = min(d0 + items, totL);
    int64_t ai64, bi64, ae64, be64;
    merge_path_lds(lk, lv, naL, nbL, d0, ai64, bi64);
    adjust_split_lds(lk, lv, naL, nbL, ai64, bi64);
    merge_path_lds(lk, lv, naL, nbL, d1, ae64, be64);
    adjust_split_lds(lk, lv, naL, nbL, ae64, be64);
    const int ai = (int)ai64, bi = (int)bi64, ae = (int)ae64, be = (int)be64;
    // fused walk: predicated (no branches), OUTPUT ROWS captured in
    // registers (OP_ITEMS x 3 VGPR pairs) so the emit below is pure stores —
    // no second index-walk, no re-compares, no replay LDS reads.
    // Steps needed <= items: the range can hold items+1 input rows only when
    // the d1 split adjustment pulled in the b half of an equal pair, and that
    // pair is consumed in one step.
    // register capture fits the VGPR budget only at the small tile (5 steps
    // = 30 capture VGPRs); the 2048 tile (9 steps) spills and keeps the
    // decision-mask + replay form instead
    constexpr bool CAPTURE = MODE != 1 && OP_ITEMS <= 5;
    uint32_t m_take = 0, m_eq = 0, m_keep = 0;
    uint32_t cnt = 0;
    uint64_t cap_k[CAPTURE ? OP_ITEMS : 1], cap_v[CAPTURE ? OP_ITEMS : 1];
    W cap_w[CAPTURE ? OP_ITEMS : 1];
    {
        int i = ai, j = bi;
#pragma unroll
        for (int t = 0; t < OP_ITEMS; t++) {
            const bool a_ok = i < ae, b_ok = j < be;
            const bool act = a_ok | b_ok;
            // all indices stay inside the tile+2 LDS arrays even when a side
            // is exhausted (i <= naL, naL + j <= totL <= MP_TILE + 1)
            const uint64_t ka = lk[i], va = lv[i];
            const uint64_t kb = lk[naL + j], vB = lv[naL + j];
            const bool eq = a_ok & b_ok & row_eq(ka, va, kb, vB);
            const bool take_a = a_ok & ((!b_ok) | row_lt(ka, va, kb, vB) | eq);
            W sum;
            if (MODE == 1) {
                // count phase stages no weights: read both sides only at an
                // equal pair (rare; the cancellation test needs the sum)
                sum = (W)0;
                if (eq) sum = (W)(aw[pa0 + i] + bw[pb0 + j]);
            } else {
                const W wa = lw[i], wb2 = lw[naL + j];
                sum = eq ? (W)(wa + wb2) : (take_a ? wa : wb2);
            }
            const bool keep = act & ((!eq) | (sum != (W)0));
            if (CAPTURE) {
                cap_k[CAPTURE ? t : 0] = take_a ? ka : kb;
                cap_v[CAPTURE ? t : 0] = take_a ? va : vB;
                cap_w[CAPTURE ? t : 0] = sum;
            } else if (MODE != 1) {
                m_take |= (uint32_t)take_a << t;
                m_eq |= (uint32_t)eq << t;
            }
            m_keep |= (uint32_t)(keep & act) << t;
            cnt += keep & act;
            i += (int)(act & (take_a | eq));
            j += (int)(act & ((!take_a) | eq));
        }
    }
    // block-exclusive scan of thread counts
    uint32_t thread_off;
    uint32_t block_cnt;
    {
        uint32_t v = cnt;
        for (int d = 1; d < WAVE; d <<= 1) {
            uint32_t up = __shfl_up(v, d, WAVE);
            if ((tid & (WAVE - 1)) >= d) v += up;
        }
        if ((tid & (WAVE - 1)) == WAVE - 1) wt[tid / WAVE] = v;
        __syncthreads();
        if (tid == 0) {
            uint32_t acc = 0;
            for (int w = 0; w < MP_THREADS / WAVE; w++) {
                uint32_t t = wt[w];
                wt[w] = acc;
                acc += t;
            }
            wt[MP_THREADS / WAVE] = acc;
        }
        __syncthreads();
        thread_off = wt[tid / WAVE] + (v - cnt);
        block_cnt = wt[MP_THREADS / WAVE];
    }
    if (MODE == 1) {
        // count phase: publish the tile's count, nothing else to do
        if (tid == 0) state[vb] = (unsigned long long)block_cnt;
        return;
    }
    if (MODE == 2) {
        // emit phase: the device scan already turned counts into offsets
        if (tid == 0) sh_prefix = state[vb];
    } else if (SKIP_LB) {
        // timing diagnostic ONLY (DBSP_MERGE_NOLB=1): measures the kernel
        // without the lookback protocol; outputs land at uncompacted per-tile
        // offsets (store pattern representative, results wrong by design)
        if (tid == 0) {
            gu64_t *g = (gu64_t *)(state + 2);
            __hip_atomic_store(&g[vb],
                               ((unsigned long long)block_cnt << 2) | 2ull,
                               __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
            sh_prefix = (unsigned long long)(vb * TILE);
        }
    } else if (tid < WAVE) {
        gu64_t *g = (gu64_t *)(state + 2);
        if (tid == 0)
            __hip_atomic_store(&g[vb],
                               ((unsigned long long)block_cnt << 2) | 1ull,
                               __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        // Flat wave-window lookback (64 granules/round).  Measured notes:
        // preloading 8 rounds of granules ahead (one round-trip for 512) and
        // 256-granule windows both came out SLOWER — the protocol is bound
        // by total granule-load throughput against the small hot region, not
        // per-round latency — so the minimal-load sequential window wins.
        unsigned long long running = 0;
        unsigned spins = 0;
        bool done = (vb == 0);
        int64_t wbase = vb - WAVE;
        while (!done) {
            const int64_t p = wbase + tid;
            const unsigned long long e =
                p >= 0 ? __hip_atomic_load(&g[p], __ATOMIC_RELAXED,
                                           __HIP_MEMORY_SCOPE_AGENT)
                       : 2ull;  // virtual predecessor: prefix 0
            const unsigned flag = (unsigned)(e & 3ull);
            const uint64_t pmask = __ballot(flag == 2u);
            const uint64_t imask = __ballot(flag == 0u);
            bool retry;
            if (pmask != 0) {
                const int hi = 63 - __clzll(pmask);  // newest prefix lane
                retry = ((imask >> hi) >> 1) != 0;   // invalid above it
                if (!retry) {
                    unsigned long long c = tid >= hi ? (e >> 2) : 0;
                    for (int d = 32; d; d >>= 1) c += __shfl_xor(c, d, WAVE);
                    running += c;
                    done = true;
                }
            } else {
                retry = imask != 0;
                if (!retry) {
                    unsigned long long c = e >> 2;
                    for (int d = 32; d; d >>= 1) c += __shfl_xor(c, d, WAVE);
                    running += c;
                    wbase -= WAVE;
                }
            }
            if (retry) {
                ++spins;
                if (spins < 8) __builtin_amdgcn_s_sleep(1);
                else __builtin_amdgcn_s_sleep(4);
                if (spins > (1u << 23)) {  // bounded spin: poison, don't hang
                    if (tid == 0)
                        __hip_atomic_store((gu64_t *)(state + 1), 1ull,
                                           __ATOMIC_RELAXED,
                                           __HIP_MEMORY_SCOPE_AGENT);
                    done = true;
                }
            }
        }
        if (tid == 0) {
            __hip_atomic_store(&g[vb],
                               ((running + block_cnt) << 2) | 2ull,
                               __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
            sh_prefix = running;
        }
    }
    __syncthreads();
    // emit: pure stores from the register capture (small tile), or replay of
    // the decision masks with LDS reads (large tile)
    uint64_t gpos = (uint64_t)sh_prefix + thread_off;
    if (CAPTURE) {
#pragma unroll
        for (int t = 0; t < OP_ITEMS; t++) {
            if ((m_keep >> t) & 1u) {
                ok[gpos] = cap_k[CAPTURE ? t : 0];
                ov[gpos] = cap_v[CAPTURE ? t : 0];
                ow[gpos] = cap_w[CAPTURE ? t : 0];
                gpos++;
            }
        }
    } else {
        int i = ai, j = bi;
#pragma unroll
        for (int t = 0; t < OP_ITEMS; t++) {
            const bool take_a = (m_take >> t) & 1u;
            const bool eq = (m_eq >> t) & 1u;
            const bool keep = (m_keep >> t) & 1u;
            const bool act = (i < ae) | (j < be);
            const int sel = take_a ? i : naL + j;
            if (keep) {
                ok[gpos] = lk[sel];
                ov[gpos] = lv[sel];
                ow[gpos] = eq ? (W)(lw[i] + lw[naL + j]) : lw[sel];
                gpos++;
            }
            i += (int)(act & (take_a | eq));
            j += (int)(act & ((!take_a) | eq));
        }
    }
}

// single-workgroup merge for small inputs (na+nb <= FUSE_MAX): merge-path
// thread diagonals + LDS scan; one launch, no host sync, length on device
template <typename W>
__global__ __launch_bounds__(FUSE_THREADS, 4) void k_merge_small(MergeArgs args) {
    __shared__ uint32_t wave_tot[FUSE_THREADS / WAVE + 1];
    const int p = blockIdx.x;
    const uint64_t *ak = args.ak[p], *av = args.av[p];
    const W *aw = (const W *)args.aw[p];
    const int64_t na = args.na[p];
    const uint64_t *bk = args.bk[p], *bv = args.bv[p];
    const W *bw = (const W *)args.bw[p];
    const int64_t nb = args.dnb[p] ? *args.dnb[p] : args.nb[p];
    uint64_t *ok = args.ok[p], *ov = args.ov[p];
    W *ow = (W *)args.ow[p];
    int64_t *out_len = args.d_len + p;
    const int tid = threadIdx.x;
    if (nb < 0 || na < 0) {  // failed upstream speculation: poison the length
        if (tid == 0) *out_len = -1;
        return;
    }
    const int64_t total = na + nb;
    const int64_t items = (total + FUSE_THREADS - 1) / FUSE_THREADS;
    int64_t d0 = min((int64_t)tid * items, total);
    int64_t d1 = min(d0 + items, total);
    int64_t ai, bi, ae, be;
    merge_path(ak, av, na, bk, bv, nb, d0, ai, bi);
    adjust_split(ak, av, bk, bv, na, nb, ai, bi);
    merge_path(ak, av, na, bk, bv, nb, d1, ae, be);
    adjust_split(ak, av, bk, bv, na, nb, ae, be);
    uint32_t cnt = 0;
    {
        int64_t i = ai, j = bi;
        while (i < ae || j < be) {
            if (i < ae && j < be && row_eq(ak[i], av[i], bk[j], bv[j])) {
                if (aw[i] + bw[j] != (W)0) cnt++;
                i++; j++;
            } else if (j >= be || (i < ae && row_lt(ak[i], av[i], bk[j], bv[j]))) {
                cnt++; i++;
            } else {
                cnt++; j++;
            }
        }
    }
    uint32_t tot_out;
    uint32_t off = fuse_scan(cnt, wave_tot, &tot_out);
    {
        int64_t i = ai, j = bi;
        while (i < ae || j < be) {
            if (i < ae && j < be && row_eq(ak[i], av[i], bk[j], bv[j])) {
                W s = aw[i] + bw[j];
                if (s != (W)0) { ok[off] = ak[i]; ov[off] = av[i]; ow[off] = s; off++; }
                i++; j++;
            } else if (j >= be || (i < ae && row_lt(ak[i], av[i], bk[j], bv[j]))) {
                ok[off] = ak[i]; ov[off] = av[i]; ow[off] = aw[i]; off++; i++;
            } else {
                ok[off] = bk[j]; ov[off] = bv[j]; ow[off] = bw[j]; off++; j++;
            }
        }
    }
    if (tid == 0) *out_len = (int64_t)tot_out;
}

// fixed-grid multi-workgroup merge with DEVICE-side b length (the in-train
// accumulator fold: the tick's delta length is only known on device).  Two
// phases over MERGE_MID_WGS workgroups per pair: count writes per-WG output
// counts to scratch; emit recomputes each WG's prefix from the tiny counts
// array (no separate scan launch) and writes its segment.  Equal (k, v)
// pairs never split across WGs (adjust_split), so per-segment consolidation
// is exact.
#define MERGE_MID_WGS 32
static_assert(MERGE_MID_SCRATCH == MERGE_MID_WGS + 1,
              "scratch layout mismatch with kernels_iface.hpp");

template <typename W>
__global__ __launch_bounds__(FUSE_THREADS, 4) void k_merge_mid_count(
        MergeArgs args, int64_t *scratch /* np * (MERGE_MID_WGS+1) */) {
    const int p = blockIdx.y;
    const int wg = blockIdx.x;
    int64_t *cnt_row = scratch + (int64_t)p * (MERGE_MID_WGS + 1);
    const int64_t na = args.na[p];
    const int64_t nb = args.dnb[p] ? *args.dnb[p] : args.nb[p];
    if (na < 0 || nb < 0) {
        if (threadIdx.x == 0) cnt_row[MERGE_MID_WGS] = -1;
        return;
    }
    if (threadIdx.x == 0 && wg == 0) cnt_row[MERGE_MID_WGS] = 0;
    const uint64_t *ak = args.ak[p], *av = args.av[p];
    const W *aw = (const W *)args.aw[p];
    const uint64_t *bk = args.bk[p], *bv = args.bv[p];
    const W *bw = (const W *)args.bw[p];
    const int64_t total = na + nb;
    const int64_t per_wg = (total + MERGE_MID_WGS - 1) / MERGE_MID_WGS;
    const int64_t w0 = min((int64_t)wg * per_wg, total);
    const int64_t w1 = min(w0 + per_wg, total);
    const int64_t items = (w1 - w0 + FUSE_THREADS - 1) / FUSE_THREADS;
    int64_t d0 = min(w0 + (int64_t)threadIdx.x * items, w1);
    int64_t d1 = min(d0 + items, w1);
    int64_t ai, bi, ae, be;
    merge_path(ak, av, na, bk, bv, nb, d0, ai, bi);
    adjust_split(ak, av, bk, bv, na, nb, ai, bi);
    merge_path(ak, av, na, bk, bv, nb, d1, ae, be);
    adjust_split(ak, av, bk, bv, na, nb, ae, be);
    uint32_t cnt = 0;
    {
        int64_t i = ai, j = bi;
        while (i < ae || j < be) {
            if (i < ae && j < be && row_eq(ak[i], av[i], bk[j], bv[j])) {
                if (aw[i] + bw[j] != (W)0) cnt++;
                i++; j++;
            } else if (j >= be || (i < ae && row_lt(ak[i], av[i], bk[j], bv[j]))) {
                cnt++; i++;
            } else {
                cnt++; j++;
            }
        }
    }
    __shared__ uint32_t wave_tot[FUSE_THREADS / WAVE + 1];
    uint32_t wg_tot;
    (void)fuse_scan(cnt, wave_tot, &wg_tot);
    if (threadIdx.x == 0) cnt_row[wg] = (int64_t)wg_tot;
}

template <typename W>
__global__ __launch_bounds__(FUSE_THREADS, 4) void k_merge_mid_emit(
        MergeArgs args, const int64_t *scratch) {
    const int p = blockIdx.y;
    const int wg = blockIdx.x;
    const int64_t *cnt_row = scratch + (int64_t)p * (MERGE_MID_WGS + 1);
    int64_t *out_len = args.d_len + p;
    if (cnt_row[MERGE_MID_WGS] < 0) {
        if (threadIdx.x == 0 && wg == 0) *out_len = -1;
        return;
    }
    const int64_t na = args.na[p];
    const int64_t nb = args.dnb[p] ? *args.dnb[p] : args.nb[p];
    const uint64_t *ak = args.ak[p], *av = args.av[p];
    const W *aw = (const W *)args.aw[p];
    const uint64_t *bk = args.bk[p], *bv = args.bv[p];
    const W *bw = (const W *)args.bw[p];
    uint64_t *ok = args.ok[p], *ov = args.ov[p];
    W *ow = (W *)args.ow[p];
    int64_t base = 0, all = 0;
    for (int g = 0; g < MERGE_MID_WGS; g++) {
        if (g < wg) base += cnt_row[g];
        all += cnt_row[g];
    }
    if (threadIdx.x == 0 && wg == 0) *out_len = all;
    const int64_t total = na + nb;
    const int64_t per_wg = (total + MERGE_MID_WGS - 1) / MERGE_MID_WGS;
    const int64_t w0 = min((int64_t)wg * per_wg, total);
    const int64_t w1 = min(w0 + per_wg, total);
    const int64_t items = (w1 - w0 + FUSE_THREADS - 1) / FUSE_THREADS;
    int64_t d0 = min(w0 + (int64_t)threadIdx.x * items, w1);
    int64_t d1 = min(d0 + items, w1);
    int64_t ai, bi, ae, be;
    merge_path(ak, av, na, bk, bv, nb, d0, ai, bi);
    adjust_split(ak, av, bk, bv, na, nb, ai, bi);
    merge_path(ak, av, na, bk, bv, nb, d1, ae, be);
    adjust_split(ak, av, bk, bv, na, nb, ae, be);
    uint32_t cnt = 0;
    {
        int64_t i = ai, j = bi;
        while (i < ae || j < be) {
            if (i < ae && j < be && row_eq(ak[i], av[i], bk[j], bv[j])) {
                if (aw[i] + bw[j] != (W)0) cnt++;
                i++; j++;
            } else if (j >= be || (i < ae && row_lt(ak[i], av[i], bk[j], bv[j]))) {
                cnt++; i++;
            } else {
                cnt++; j++;
            }
        }
    }
    __shared__ uint32_t wave_tot[FUSE_THREADS / WAVE + 1];
    uint32_t seg_tot;
    uint32_t off32 = fuse_scan(cnt, wave_tot, &seg_tot);
    int64_t off = base + (int64_t)off32;
    {
        int64_t i = ai, j = bi;
        while (i < ae || j < be) {
            if (i < ae && j < be && row_eq(ak[i], av[i], bk[j], bv[j])) {
                W s = aw[i] + bw[j];
                if (s != (W)0) { ok[off] = ak[i]; ov[off] = av[i]; ow[off] = s; off++; }
                i++; j++;
            } else if (j >= be || (i < ae && row_lt(ak[i], av[i], bk[j], bv[j]))) {
                ok[off] = ak[i]; ov[off] = av[i]; ow[off] = aw[i]; off++; i++;
            } else {
                ok[off] = bk[j]; ov[off] = bv[j]; ow[off] = bw[j]; off++; j++;
            }
        }
    }
}

// single-launch variant: count, one-shot cross-WG barrier, emit.  The grid
// is tiny (MERGE_MID_WGS x np <= 128 workgroups, all co-resident), so every
// WG publishes its count tagged with the launch generation and spins for
// its pair's 32 peers — a single barrier between uniform-work peers, NOT a
// serial lookback chain.  Each thread keeps its merge-path bounds in
// registers across the barrier, so the emit phase skips the splits the
// two-launch version recomputes.  Generation tags make the persistent
// scratch self-cleaning (no memset launch); a stale match would need a
// pair row untouched for exactly 2^32 launches.
template <typename W>
__global__ __launch_bounds__(FUSE_THREADS, 4) void k_merge_mid_fused(
        MergeArgs args, int64_t *scratch, uint32_t gen) {
    const int p = blockIdx.y;
    const int wg = blockIdx.x;
    int64_t *cnt_row = scratch + (int64_t)p * (MERGE_MID_WGS + 1);
    int64_t *out_len = args.d_len + p;
    const int tid = threadIdx.x;
    const int64_t na = args.na[p];
    const int64_t nb = args.dnb[p] ? *args.dnb[p] : args.nb[p];
    if (na < 0 || nb < 0) {  // whole pair bails uniformly: no spin to feed
        if (tid == 0 && wg == 0) *out_len = -1;
        return;
    }
    const uint64_t *ak = args.ak[p], *av = args.av[p];
    const W *aw = (const W *)args.aw[p];
    const uint64_t *bk = args.bk[p], *bv = args.bv[p];
    const W *bw = (const W *)args.bw[p];
    uint64_t *ok = args.ok[p], *ov = args.ov[p];
    W *ow = (W *)args.ow[p];
    const int64_t total = na + nb;
    const int64_t per_wg = (total + MERGE_MID_WGS - 1) / MERGE_MID_WGS;
    const int64_t w0 = min((int64_t)wg * per_wg, total);
    const int64_t w1 = min(w0 + per_wg, total);
    const int64_t items = (w1 - w0 + FUSE_THREADS - 1) / FUSE_THREADS;
    int64_t d0 = min(w0 + (int64_t)tid * items, w1);
    int64_t d1 = min(d0 + items, w1);
    int64_t ai, bi, ae, be;
    merge_path(ak, av, na, bk, bv, nb, d0, ai, bi);
    adjust_split(ak, av, bk, bv, na, nb, ai, bi);
    merge_path(ak, av, na, bk, bv, nb, d1, ae, be);
    adjust_split(ak, av, bk, bv, na, nb, ae, be);
    uint32_t cnt = 0;
    {
        int64_t i = ai, j = bi;
        while (i < ae || j < be) {
            if (i < ae && j < be && row_eq(ak[i], av[i], bk[j], bv[j])) {
                if (aw[i] + bw[j] != (W)0) cnt++;
                i++; j++;
            } else if (j >= be || (i < ae && row_lt(ak[i], av[i], bk[j], bv[j]))) {
                cnt++; i++;
            } else {
                cnt++; j++;
            }
        }
    }
    __shared__ uint32_t wave_tot[FUSE_THREADS / WAVE + 1];
    uint32_t wg_tot;
    uint32_t off32 = fuse_scan(cnt, wave_tot, &wg_tot);
    if (tid == 0)
        atomicExch((unsigned long long *)&cnt_row[wg],
                   ((unsigned long long)gen << 32) | wg_tot);
    __shared__ int64_t s_cnt[MERGE_MID_WGS];
    __shared__ int s_dead;
    if (tid == 0) s_dead = 0;
    __syncthreads();
    if (tid < MERGE_MID_WGS) {
        // bounded spin: the grid is co-resident by construction, so this
        // resolves in ~kernel-uniform time; the bound turns a scheduling
        // assumption failure into a poisoned length (host fails loudly /
        // the train replays) instead of a wedged GPU
        unsigned long long v = 0;
        for (long spins = 0;; spins++) {
            v = atomicAdd((unsigned long long *)&cnt_row[tid], 0ull);
            if ((uint32_t)(v >> 32) == (uint32_t)gen) break;
            if (spins > (1l << 26)) { s_dead = 1; break; }
            __builtin_amdgcn_s_sleep(1);
        }
        s_cnt[tid] = (int64_t)(uint32_t)v;
    }
    __syncthreads();
    if (s_dead) {
        if (tid == 0) *out_len = -1;
        return;
    }
    int64_t base = 0, all = 0;
    for (int g = 0; g < MERGE_MID_WGS; g++) {
        if (g < wg) base += s_cnt[g];
        all += s_cnt[g];
    }
    if (wg == 0 && tid == 0) *out_len = all;
    int64_t off = base + (int64_t)off32;
    {
        int64_t i = ai, j = bi;
        while (i < ae || j < be) {
            if (i < ae && j < be && row_eq(ak[i], av[i], bk[j], bv[j])) {
                W s = aw[i] + bw[j];
                if (s != (W)0) { ok[off] = ak[i]; ov[off] = av[i]; ow[off] = s; off++; }
                i++; j++;
            } else if (j >= be || (i < ae && row_lt(ak[i], av[i], bk[j], bv[j]))) {
                ok[off] = ak[i]; ov[off] = av[i]; ow[off] = aw[i]; off++; i++;
            } else {
                ok[off] = bk[j]; ov[off] = bv[j]; ow[off] = bw[j]; off++; j++;
            }
        }
    }
}

// ---------------------------------------------------------------------------
// multi-batch spine join: the reference reads a spine through a k-way
// CursorList (trace/cursor/cursor_list.rs); join is linear in the trace, so
// one kernel probes every spine batch (descriptors passed by value as kernel
// arguments) — one count+emit pair per tick per side regardless of spine depth.
// ---------------------------------------------------------------------------

__device__ inline int64_t gallop_run(const uint64_t *tk, int64_t nt, uint64_t key,
                                     int64_t lo) {
    // length of the key's run starting at lo (gallop + bounded binary search,
    // the advance.rs:11-60 idiom)
    if (lo >= nt || tk[lo] != key) return 0;
    int64_t hi = lo, step = 1;
    while (hi + step < nt && tk[hi + step] == key) { hi += step; step <<= 1; }
    int64_t rem = min(step, nt - hi);
    // upper bound within (hi, hi+rem)
    int64_t a = 0, b = rem;
    while (a < b) {
        int64_t m = (a + b) / 2;
        if (tk[hi + m] <= key) a = m + 1; else b = m;
    }
    return hi + a - lo;
}

__global__ void k_join_count_multi(const uint64_t *dk, int64_t nd, TraceArgs t,
                                   uint32_t *cnts /* nd*nb */,
                                   uint64_t *counts /* nd */) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nd;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint64_t key = dk[i];
        uint64_t total = 0;
        for (int b = 0; b < t.nb; b++) {
            int64_t lo = lower_bound_k(t.k[b], t.n[b], key);
            int64_t c = gallop_run(t.k[b], t.n[b], key, lo);
            cnts[i * t.nb + b] = (uint32_t)c;
            total += c;
        }
        counts[i] = total;
    }
}


// ---------------------------------------------------------------------------
// chained join emit: per-plan output bases from the device-side count totals
// (so the emits and the output consolidate launch before the tick sync), and
// the emit kernel itself reads every size from device memory.  flag != 0
// (a lost speculation upstream, or the combined output exceeding the
// capacity buffer) makes every consumer bail; the host re-emits explicitly.
// ---------------------------------------------------------------------------
// fused variant: bases + all (<=3) plans' emits in ONE launch — each of the
// three separate chained emit launches cost ~5 us of dependent dispatch on
// the q3 tick's critical path; every thread recomputes the 3-entry base
// prefix from the (L2-hot) count totals instead
__global__ void k_join_emit_fused(FusedEmitArgs a) {
    int64_t bases[3];
    int64_t acc = 0;
    int okf = 1;
    for (int i = 0; i < a.np; i++) {
        const int64_t t = a.totals[i];
        if (t < 0) {
            okf = 0;
            break;
        }
        bases[i] = acc;
        acc += t;
    }
    if (acc > a.cap) okf = 0;
    if (blockIdx.x == 0 && threadIdx.x == 0) {
        *a.d_total = okf ? acc : -1;
        *a.d_flag = okf ? 0 : 1;
    }
    if (!okf) return;
    for (int64_t o = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; o < acc;
         o += (int64_t)gridDim.x * blockDim.x) {
        int p = 0;
        while (p + 1 < a.np && o >= bases[p + 1]) p++;
        const int64_t nd = *a.nd_dev[p];
        if (nd <= 0) continue;
        const TraceArgs &t = a.t[p];
        const uint32_t *cnts = a.cnts[p];
        const uint64_t *offsets = a.offsets[p];
        const int64_t tn0 = a.tn_dev[p] ? *a.tn_dev[p] : t.n[0];
        const int64_t oo = o - bases[p];
        int64_t lo = 0, hi = nd;
        while (lo < hi) {
            int64_t mid = (lo + hi) / 2;
            if (offsets[mid] <= (uint64_t)oo) lo = mid + 1; else hi = mid;
        }
        int64_t i = lo - 1;
        int64_t j = oo - (int64_t)offsets[i];
        int b = 0;
        while (j >= (int64_t)cnts[i * t.nb + b]) {
            j -= cnts[i * t.nb + b];
            b++;
        }
        const uint64_t key = a.dk[p][i];
        const int64_t tb_n = b == 0 ? tn0 : t.n[b];
        const int64_t start = lower_bound_k(t.k[b], tb_n, key);
        const int64_t ti = start + j;
        uint64_t hi_o, lo_o;
        const bool valid = proj_out(a.proj[p], 0, key, a.dv[p][i],
                                    t.v[b][ti], hi_o, lo_o);
        a.ok[o] = hi_o;
        a.ov[o] = lo_o;
        a.ow[o] = valid ? a.dw[p][i] * t.w[b][ti] : 0;
    }
}

__global__ void k_emit_bases(const int64_t *totals, int np, int64_t cap,
                             int64_t *bases, int64_t *out_total,
                             int64_t *out_flag) {
    int64_t acc = 0;
    int ok = 1;
    for (int i = 0; i < np; i++) {
        const int64_t t = totals[i];
        if (t < 0) { ok = 0; break; }
        bases[i] = acc;
        acc += t;
    }
    if (acc > cap) ok = 0;
    *out_total = ok ? acc : -1;
    *out_flag = ok ? 0 : 1;
}

__global__ void k_join_emit_chain(const uint64_t *dk, const uint64_t *dv,
                                  const int64_t *dw, const int64_t *nd_dev,
                                  TraceArgs t, const int64_t *tn_dev,
                                  const uint32_t *cnts,
                                  const uint64_t *offsets,
                                  const int64_t *total_dev,
                                  const int64_t *base_dev,
                                  const int64_t *flag_dev, int proj,
                                  uint64_t param, uint64_t *ok, uint64_t *ov,
                                  int64_t *ow) {
    if (*flag_dev != 0) return;
    const int64_t nd = *nd_dev;
    const int64_t n_out = *total_dev;
    if (nd <= 0 || n_out <= 0) return;
    const int64_t base = *base_dev;
    const int64_t tn0 = tn_dev ? *tn_dev : t.n[0];
    for (int64_t o = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; o < n_out;
         o += (int64_t)gridDim.x * blockDim.x) {
        int64_t lo = 0, hi = nd;
        while (lo < hi) {
            int64_t mid = (lo + hi) / 2;
            if (offsets[mid] <= (uint64_t)o) lo = mid + 1; else hi = mid;
        }
        int64_t i = lo - 1;
        int64_t j = o - (int64_t)offsets[i];
        int b = 0;
        while (j >= (int64_t)cnts[i * t.nb + b]) {
            j -= cnts[i * t.nb + b];
            b++;
        }
        uint64_t key = dk[i];
        const int64_t tb_n = b == 0 ? tn0 : t.n[b];
        int64_t start = lower_bound_k(t.k[b], tb_n, key);
        int64_t ti = start + j;
        uint64_t hi_o, lo_o;
        const bool valid =
            proj_out(proj, param, key, dv[i], t.v[b][ti], hi_o, lo_o);
        ok[base + o] = hi_o;
        ov[base + o] = lo_o;
        ow[base + o] = valid ? dw[i] * t.w[b][ti] : 0;
    }
}

__global__ void k_join_emit_multi(const uint64_t *dk, const uint64_t *dv,
                                  const int64_t *dw, int64_t nd, TraceArgs t,
                                  const uint32_t *cnts, const uint64_t *offsets,
                                  int64_t n_out, int proj, uint64_t param,
                                  uint64_t *ok, uint64_t *ov, int64_t *ow) {
    for (int64_t o = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; o < n_out;
         o += (int64_t)gridDim.x * blockDim.x) {
        int64_t lo = 0, hi = nd;
        while (lo < hi) {
            int64_t mid = (lo + hi) / 2;
            if (offsets[mid] <= (uint64_t)o) lo = mid + 1; else hi = mid;
        }
        int64_t i = lo - 1;
        int64_t j = o - (int64_t)offsets[i];
        int b = 0;
        while (j >= (int64_t)cnts[i * t.nb + b]) {
            j -= cnts[i * t.nb + b];
            b++;
        }
        uint64_t key = dk[i];
        int64_t start = lower_bound_k(t.k[b], t.n[b], key);
        int64_t ti = start + j;
        uint64_t hi_o, lo_o;
        const bool valid =
            proj_out(proj, param, key, dv[i], t.v[b][ti], hi_o, lo_o);
        ok[o] = hi_o;
        ov[o] = lo_o;
        ow[o] = valid ? dw[i] * t.w[b][ti] : 0;
    }
}

// small join count + scan (nd <= 8192), split in two chained launches: a
// multi-block probe phase (the round-1 single-WG version serialized ~2M
// dependent probe loads on ONE CU — 44 us of a 280 us q3 tick) and a
// single-WG scan that re-sums the per-row counts from cnts.
#define JPROBE_BLOCKS 8

__global__ __launch_bounds__(BLK, 8) void k_join_probe_multi(
    JoinCountArgs args) {
    const int plan = blockIdx.x / JPROBE_BLOCKS;
    const int slice = blockIdx.x % JPROBE_BLOCKS;
    const uint64_t *dk = args.dk[plan];
    const int64_t nd =
        args.nd_dev[plan] ? *args.nd_dev[plan] : args.nd[plan];
    const TraceArgs &t = args.t[plan];
    const int64_t tn0 = args.tn_dev[plan] ? *args.tn_dev[plan] : t.n[0];
    if (nd < 0 || nd > FUSE_MAX || tn0 < 0) return;  // scan flags the total
    uint32_t *cnts = args.cnts[plan];
    for (int64_t i = (int64_t)slice * BLK + threadIdx.x; i < nd;
         i += (int64_t)JPROBE_BLOCKS * BLK) {
        const uint64_t key = dk[i];
        for (int b = 0; b < t.nb; b++) {
            const int64_t tb_n = b == 0 ? tn0 : t.n[b];
            int64_t lo = lower_bound_k(t.k[b], tb_n, key);
            cnts[(int64_t)i * t.nb + b] =
                (uint32_t)gallop_run(t.k[b], tb_n, key, lo);
        }
    }
}

__global__ __launch_bounds__(FUSE_THREADS, 4) void k_join_scan_small(
    JoinCountArgs args) {
    __shared__ uint32_t wave_tot[FUSE_THREADS / WAVE + 1];
    const int plan = blockIdx.x;
    const int64_t nd =
        args.nd_dev[plan] ? *args.nd_dev[plan] : args.nd[plan];
    const TraceArgs &t = args.t[plan];
    const int64_t tn0 = args.tn_dev[plan] ? *args.tn_dev[plan] : t.n[0];
    if (nd < 0 || nd > FUSE_MAX || tn0 < 0) {
        if (threadIdx.x == 0) args.d_total[plan] = -1;
        return;
    }
    uint32_t *cnts = args.cnts[plan];
    uint64_t *offsets = args.offsets[plan];
    int64_t *d_total = args.d_total + plan;
    const int tid = threadIdx.x;
    uint32_t row_tot[FUSE_ITEMS];
    uint32_t tsum = 0;
    for (int j = 0; j < FUSE_ITEMS; j++) {
        int i = tid * FUSE_ITEMS + j;
        uint32_t rt = 0;
        if (i < nd)
            for (int b = 0; b < t.nb; b++)
                rt += cnts[(int64_t)i * t.nb + b];
        row_tot[j] = rt;
        tsum += rt;
    }
    uint32_t total;
    uint32_t off = fuse_scan(tsum, wave_tot, &total);
    for (int j = 0; j < FUSE_ITEMS; j++) {
        int i = tid * FUSE_ITEMS + j;
        if (i < nd) offsets[i] = off;
        off += row_tot[j];
    }
    if (tid == 0) *d_total = (int64_t)total;
}

// ---------------------------------------------------------------------------
// incremental distinct (operator/distinct.rs:273,404-462 at root scope):
// for each delta pair (k,v):  out = [w_before + dw > 0] - [w_before > 0]
// where w_before is the pair's total weight in the delayed integral.  NOT
// linear in the trace (the indicator needs the total), so one kernel probes
// every spine batch.  Output positions via flags+scan keep delta order, so
// the result is consolidated by construction.
// ---------------------------------------------------------------------------

__device__ inline int64_t find_pair_weight(const uint64_t *tk,
                                           const uint64_t *tv,
                                           const int64_t *tw, int64_t n,
                                           uint64_t key, uint64_t val) {
    int64_t lo = lower_bound_k(tk, n, key);
    int64_t hi = lo + gallop_run(tk, n, key, lo);
    // binary search the val within the key's run
    while (lo < hi) {
        int64_t mid = (lo + hi) / 2;
        if (tv[mid] < val) lo = mid + 1;
        else hi = mid;
    }
    if (lo < n && tk[lo] == key && tv[lo] == val) return tw[lo];
    return 0;
}

__global__ void k_distinct_count(const uint64_t *dk, const uint64_t *dv,
                                 const int64_t *dw, int64_t nd, TraceArgs t,
                                 uint64_t *flags, int64_t *outw) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nd;
         i += (int64_t)gridDim.x * blockDim.x) {
        int64_t before = 0;
        for (int b = 0; b < t.nb; b++)
            before += find_pair_weight(t.k[b], t.v[b], t.w[b], t.n[b], dk[i],
                                       dv[i]);
        int64_t after = before + dw[i];
        int64_t o = (int64_t)(after > 0) - (int64_t)(before > 0);
        flags[i] = o != 0;
        outw[i] = o;
    }
}

// ---------------------------------------------------------------------------
// aggregate (linear / max) + upsert  (count/emit over delta keys)
// ---------------------------------------------------------------------------

// MODE: 0 = linear weight-sum (WeightedCount, aggregate/mod.rs:129-156),
// 1 = Max (max.rs:36-55), 2 = q6's fold: average of the last <= 10 vals in
// the key's run (queries/q6.rs:96-110 VecDeque fold in cursor order; the
// price lives in the low 20 bits of the val)
template <int MODE>
__global__ void k_agg_count(const uint64_t *keys, int64_t nd,
                            const uint64_t *ik, const int64_t *iw, int64_t ni,
                            const uint64_t *ok_, int64_t no,
                            uint64_t *counts, uint64_t *istart, uint64_t *ostart,
                            uint64_t *newval, uint64_t *has_new) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nd;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint64_t key = keys[i];
        int64_t ilo = lower_bound_k(ik, ni, key);
        int64_t ihi = upper_bound_k(ik, ni, key);
        uint64_t nv = 0;
        uint64_t hn = 0;
        if (MODE != 0) {
            // consolidated trace: any present val has w != 0; max = last val
            if (ihi > ilo) { nv = 1; hn = 1; }
            if (hn) nv = 0;  // placeholder; real val read in emit via istart/iend
        } else {
            int64_t s = 0;
            for (int64_t t = ilo; t < ihi; t++) s += iw[t];
            if (s != 0) { nv = (uint64_t)s; hn = 1; }
        }
        int64_t olo = lower_bound_k(ok_, no, key);
        int64_t ohi = upper_bound_k(ok_, no, key);
        counts[i] = (hn ? 1 : 0) + (ohi - olo);
        istart[i] = (uint64_t)ilo;
        ostart[i] = (uint64_t)olo;
        newval[i] = MODE != 0 ? (ihi > ilo ? /* iend */ (uint64_t)ihi : 0) : nv;
        has_new[i] = hn;
    }
}

template <int MODE>
__global__ void k_agg_emit(const uint64_t *keys, int64_t nd,
                           const uint64_t *iv, const uint64_t *ov_,
                           const int64_t *ow_, const uint64_t *offsets,
                           const uint64_t *counts, const uint64_t *istart,
                           const uint64_t *ostart, const uint64_t *newval,
                           const uint64_t *has_new, uint64_t *rk, uint64_t *rv,
                           int64_t *rw) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nd;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint64_t key = keys[i];
        uint64_t pos = offsets[i];
        uint64_t cnt = counts[i];
        uint64_t emitted = 0;
        if (has_new[i]) {
            uint64_t nv;
            if (MODE == 1) {
                nv = iv[newval[i] - 1];  // last val of key's run (max.rs:40-52)
            } else if (MODE == 2) {
                // q6.rs:96-110: the VecDeque fold keeps the last <= 10 vals
                // of the seller's cursor-ordered run; avg of their prices
                // (val low 20 bits), integer division
                const uint64_t ihi = newval[i], ilo = istart[i];
                const uint64_t n10 =
                    ihi - ilo < 10 ? ihi - ilo : (uint64_t)10;
                uint64_t sum = 0;
                for (uint64_t t = ihi - n10; t < ihi; t++)
                    sum += iv[t] & 0x7FFFFFFull;
                nv = sum / n10;
            } else {
                nv = newval[i];
            }
            rk[pos] = key; rv[pos] = nv; rw[pos] = 1;
            pos++; emitted++;
        }
        // retractions from the output trace (upsert.rs:180-195)
        for (uint64_t t = ostart[i]; emitted < cnt; t++, emitted++, pos++) {
            rk[pos] = key; rv[pos] = ov_[t]; rw[pos] = -ow_[t];
        }
    }
}

// ---------------------------------------------------------------------------
// window ranges (window.rs:144-220): 4 slice copies with sign
// ---------------------------------------------------------------------------

// compute the 4 region ranges on device: [r0lo,r0hi) retract, [r1lo,r1hi)
// retract-shrink, [r2lo,r2hi) insert (trace), [r3lo,r3hi) insert (batch)
__global__ void k_window_ranges(const uint64_t *tk, int64_t nt,
                                const uint64_t *bk, int64_t nb, int have_prev,
                                uint64_t s0, uint64_t e0, uint64_t s1,
                                uint64_t e1, int64_t *ranges /* [8] */) {
    if (threadIdx.x != 0 || blockIdx.x != 0) return;
    int64_t r0lo = 0, r0hi = 0, r1lo = 0, r1hi = 0, r2lo = 0, r2hi = 0;
    if (have_prev) {
        uint64_t r0end = min(s1, e0);
        r0lo = lower_bound_k(tk, nt, s0);
        r0hi = max(r0lo, (int64_t)lower_bound_k(tk, nt, r0end));
        if (e1 < e0) {
            r1lo = lower_bound_k(tk, nt, e1);
            r1hi = max(r1lo, (int64_t)lower_bound_k(tk, nt, e0));
        }
        uint64_t r2start = max(e0, s1);
        r2lo = lower_bound_k(tk, nt, r2start);
        r2hi = max(r2lo, (int64_t)lower_bound_k(tk, nt, e1));
    }
    int64_t r3lo = lower_bound_k(bk, nb, s1);
    int64_t r3hi = max(r3lo, (int64_t)lower_bound_k(bk, nb, e1));
    ranges[0] = r0lo; ranges[1] = r0hi; ranges[2] = r1lo; ranges[3] = r1hi;
    ranges[4] = r2lo; ranges[5] = r2hi; ranges[6] = r3lo; ranges[7] = r3hi;
}

__global__ void k_window_emit(const uint64_t *tk, const uint64_t *tv,
                              const int64_t *tw, const uint64_t *bk,
                              const uint64_t *bv, const int64_t *bw,
                              const int64_t *ranges, uint64_t *ok, uint64_t *ov,
                              int64_t *ow) {
    int64_t len0 = ranges[1] - ranges[0];
    int64_t len1 = ranges[3] - ranges[2];
    int64_t len2 = ranges[5] - ranges[4];
    int64_t len3 = ranges[7] - ranges[6];
    int64_t total = len0 + len1 + len2 + len3;
    for (int64_t o = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; o < total;
         o += (int64_t)gridDim.x * blockDim.x) {
        int64_t src;
        int64_t sign;
        const uint64_t *k;
        const uint64_t *v;
        const int64_t *w;
        if (o < len0) { src = ranges[0] + o; sign = -1; k = tk; v = tv; w = tw; }
        else if (o < len0 + len1) { src = ranges[2] + (o - len0); sign = -1; k = tk; v = tv; w = tw; }
        else if (o < len0 + len1 + len2) { src = ranges[4] + (o - len0 - len1); sign = 1; k = tk; v = tv; w = tw; }
        else { src = ranges[6] + (o - len0 - len1 - len2); sign = 1; k = bk; v = bv; w = bw; }
        ok[o] = k[src];
        ov[o] = v[src];
        ow[o] = sign * w[src];
    }
}

// multi-batch window: all three regions of every spine batch plus the tick's
// batch region in one ranges launch + one emit launch (window.rs:144-220
// evaluated per spine batch; the trace excludes the current tick).
// Region table rows: [src_batch(-1 = tick batch), lo, len, sign, goff]

// ---------------------------------------------------------------------------
// device-resident watermark (q5.rs:85-90 / q8.rs:63-65 waterline): single
// thread reads the sorted delta's last key (the tick's max event-time),
// advances the persistent watermark state and publishes the window bounds
// for the chained ranges launch.  state = {wm, s0, e0, have_prev};
// bounds = {s0, e0, s1, e1, have_prev, err}.  err=1 when the delta length is
// the speculative-sort overflow sentinel — the host then recomputes on its
// explicit path (the state is left untouched).
// ---------------------------------------------------------------------------
__device__ void wm_update_body(const uint64_t *ak, int64_t n, uint64_t width,
                               uint64_t tumble, uint64_t lag,
                               unsigned long long *state,
                               unsigned long long *bounds);

// sharded variant: the tick's global max event-time arrives REDUCED in
// *gmax_dev (local last key -> ncclAllReduce(max) on the stream), so the
// whole watermark/window-bounds update stays device-side across ranks
__global__ void k_wm_update_g(const unsigned long long *gmax_dev,
                              uint64_t width, uint64_t tumble, uint64_t lag,
                              unsigned long long *state,
                              unsigned long long *bounds) {
    const uint64_t g = *gmax_dev;
    uint64_t wm = state[0];
    if (g > 0) wm = max(wm, (uint64_t)(g - lag));
    uint64_t rounded = wm - wm % tumble;
    uint64_t s1 = rounded >= width ? rounded - width : 0;
    uint64_t e1 = rounded;
    bounds[0] = state[1];
    bounds[1] = state[2];
    bounds[2] = s1;
    bounds[3] = e1;
    bounds[4] = state[3];
    bounds[5] = 0;
    state[0] = wm;
    state[1] = s1;
    state[2] = e1;
    state[3] = 1;
}

__global__ void k_wm_update_n(const uint64_t *ak, int64_t n, uint64_t width,
                              uint64_t tumble, uint64_t lag,
                              unsigned long long *state,
                              unsigned long long *bounds) {
    wm_update_body(ak, n, width, tumble, lag, state, bounds);
}

__global__ void k_wm_update(const uint64_t *ak, const int64_t *n_dev,
                            uint64_t width, uint64_t tumble, uint64_t lag,
                            unsigned long long *state,
                            unsigned long long *bounds) {
    wm_update_body(ak, *n_dev, width, tumble, lag, state, bounds);
}

__device__ void wm_update_body(const uint64_t *ak, int64_t n, uint64_t width,
                               uint64_t tumble, uint64_t lag,
                               unsigned long long *state,
                               unsigned long long *bounds) {
    if (n < 0) {
        bounds[5] = 1;
        return;
    }
    uint64_t wm = state[0];
    if (n > 0) {
        uint64_t gmax = ak[n - 1];
        if (gmax > 0) wm = max(wm, gmax - lag);
    }
    uint64_t rounded = wm - wm % tumble;
    uint64_t s1 = rounded >= width ? rounded - width : 0;
    uint64_t e1 = rounded;
    bounds[0] = state[1];
    bounds[1] = state[2];
    bounds[2] = s1;
    bounds[3] = e1;
    bounds[4] = state[3];
    bounds[5] = 0;
    state[0] = wm;
    state[1] = s1;
    state[2] = e1;
    state[3] = 1;
}

__global__ void k_window_ranges_multi(TraceArgs t, const uint64_t *bk,
                                      int64_t bn, int have_prev, uint64_t s0,
                                      uint64_t e0, uint64_t s1, uint64_t e1,
                                      int64_t *table, int64_t *d_total,
                                      const int64_t *bn_dev,
                                      const unsigned long long *bounds) {
    if (bounds) {  // chained: bounds from k_wm_update, delta length from sort
        if (bounds[5] != 0 || (bn_dev && *bn_dev < 0)) {
            if (threadIdx.x == 0) *d_total = -1;
            return;
        }
        s0 = bounds[0];
        e0 = bounds[1];
        s1 = bounds[2];
        e1 = bounds[3];
        have_prev = (int)bounds[4];
        if (bn_dev) bn = *bn_dev;
    }
    const int nreg = 3 * t.nb + 1;
    for (int r = threadIdx.x; r < nreg; r += blockDim.x) {
        int64_t src = -1, lo = 0, len = 0, sign = 1;
        if (r < 3 * t.nb) {
            int b = r / 3, which = r % 3;
            const uint64_t *k = t.k[b];
            int64_t n = t.n[b];
            src = b;
            if (have_prev) {
                if (which == 0) {  // region 1: [s0, min(s1,e0)) retract
                    uint64_t end = min(s1, e0);
                    lo = lower_bound_k(k, n, s0);
                    len = max((int64_t)0, lower_bound_k(k, n, end) - lo);
                    sign = -1;
                } else if (which == 1) {  // shrink: [e1, e0) retract
                    if (e1 < e0) {
                        lo = lower_bound_k(k, n, e1);
                        len = max((int64_t)0, lower_bound_k(k, n, e0) - lo);
                    }
                    sign = -1;
                } else {  // region 3: [max(e0,s1), e1) insert
                    uint64_t st = max(e0, s1);
                    lo = lower_bound_k(k, n, st);
                    len = max((int64_t)0, lower_bound_k(k, n, e1) - lo);
                }
            }
        } else {  // batch region: [s1, e1) insert
            lo = lower_bound_k(bk, bn, s1);
            len = max((int64_t)0, lower_bound_k(bk, bn, e1) - lo);
        }
        table[r * 5 + 0] = src;
        table[r * 5 + 1] = lo;
        table[r * 5 + 2] = len;
        table[r * 5 + 3] = sign;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
        int64_t acc = 0;
        for (int r = 0; r < nreg; r++) {
            table[r * 5 + 4] = acc;
            acc += table[r * 5 + 2];
        }
        *d_total = acc;
    }
}

__global__ void k_window_emit_multi(TraceArgs t, const uint64_t *bk,
                                    const uint64_t *bv, const int64_t *bw,
                                    const int64_t *table, int nreg,
                                    int64_t total, uint64_t *ok, uint64_t *ov,
                                    int64_t *ow) {
    for (int64_t o = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; o < total;
         o += (int64_t)gridDim.x * blockDim.x) {
        // binary search the region by goff
        int lo = 0, hi = nreg;
        while (lo < hi) {
            int mid = (lo + hi) / 2;
            if (table[mid * 5 + 4] <= o) lo = mid + 1; else hi = mid;
        }
        int r = lo - 1;
        int64_t src = table[r * 5 + 0];
        int64_t i = table[r * 5 + 1] + (o - table[r * 5 + 4]);
        int64_t sign = table[r * 5 + 3];
        const uint64_t *k = src < 0 ? bk : t.k[src];
        const uint64_t *v = src < 0 ? bv : t.v[src];
        const int64_t *w = src < 0 ? bw : t.w[src];
        ok[o] = k[i];
        ov[o] = v[i];
        ow[o] = sign * w[i];
    }
}

// ---------------------------------------------------------------------------
// shard partition: xxh3(key) % nshards (shard.rs:165-199, hash.rs:9-13)
// ---------------------------------------------------------------------------

__device__ inline uint64_t dev_xxh3_u64(uint64_t key, uint64_t seed) {
    const uint64_t SEC8_16 = 0x1cad21f72c81017cull ^ 0xdb979083e96dd4deull;
    // read64(kSecret+8) = 0x1cad21f72c81017c, read64(kSecret+16) = 0xdb979083e96dd4de
    uint64_t s = seed ^ ((uint64_t)__builtin_bswap32((uint32_t)seed) << 32);
    uint32_t in_lo = (uint32_t)key;
    uint32_t in_hi = (uint32_t)(key >> 32);
    uint64_t bitflip = SEC8_16 - s;
    uint64_t input64 = (uint64_t)in_hi + ((uint64_t)in_lo << 32);
    uint64_t h = input64 ^ bitflip;
    uint64_t r49 = (h << 49) | (h >> 15);
    uint64_t r24 = (h << 24) | (h >> 40);
    h ^= r49 ^ r24;
    h *= 0x9FB21C651E98DF25ull;
    h ^= (h >> 35) + 8;
    h *= 0x9FB21C651E98DF25ull;
    h ^= h >> 28;
    return h;
}

#define DBSP_HASH_SEED 0x7f95ef85be33c337ull  /* hash.rs:6 */

__global__ void k_shard_hist(const uint64_t *k, int64_t n, int nshards,
                             uint64_t *hist) {
    __shared__ uint64_t sh[64];
    for (int i = threadIdx.x; i < nshards; i += BLK) sh[i] = 0;
    __syncthreads();
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        atomicAdd((unsigned long long *)&sh[dev_xxh3_u64(k[i], DBSP_HASH_SEED) % nshards], 1ull);
    __syncthreads();
    for (int i = threadIdx.x; i < nshards; i += BLK)
        atomicAdd((unsigned long long *)&hist[i], (unsigned long long)sh[i]);
}

// unstable within a shard (receivers re-consolidate; Z-set semantics are
// order-invariant — shard.rs re-assembles via Spine + consolidate anyway)
__global__ void k_shard_scatter(const uint64_t *k, const uint64_t *v,
                                const int64_t *w, int64_t n, int nshards,
                                uint64_t *cursor, uint64_t *ok, uint64_t *ov,
                                int64_t *ow) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        int s = (int)(dev_xxh3_u64(k[i], DBSP_HASH_SEED) % nshards);
        uint64_t pos = atomicAdd((unsigned long long *)&cursor[s], 1ull);
        ok[pos] = k[i];
        ov[pos] = v[i];
        ow[pos] = w[i];
    }
}

// ---------------------------------------------------------------------------
// Nexmark flat_map_index kernels (ingress: input.rs:591-721 + per-query
// flat_map_index in queries/q{3,5,8}.rs); ticket-append (order-free, the
// consolidate that follows sorts)
// ---------------------------------------------------------------------------

// wave-aggregated ticket append: one atomicAdd per wave per predicate
// instead of one per matching row (the flatmap's counters were the hottest
// atomics in the tick)
__device__ inline uint64_t wave_append(unsigned long long *ctr, bool pred) {
    const int lane = threadIdx.x & (WAVE - 1);
    const uint64_t m = __ballot(pred);
    if (m == 0) return 0;
    const uint64_t lt = ((uint64_t)1 << lane) - 1;
    unsigned long long base = 0;
    const int leader = __ffsll((unsigned long long)m) - 1;
    if (lane == leader)
        base = atomicAdd(ctr, (unsigned long long)__popcll(m));
    base = (unsigned long long)__shfl((long long)base, leader, WAVE);
    return base + (uint64_t)__popcll(m & lt);
}

__global__ void k_columnize(const dbsp_event *ev, int64_t n, uint64_t *kind,
                            uint64_t *f0, uint64_t *f1, uint64_t *f2,
                            uint64_t *f3, uint64_t *f4, int64_t *w) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        const dbsp_event e = ev[i];
        kind[i] = e.kind;
        f0[i] = e.f0;
        f1[i] = e.f1;
        f2[i] = e.f2;
        f3[i] = e.f3;
        f4[i] = e.f4;
        w[i] = e.w;
    }
}

__global__ void k_flatmap(const dbsp_event *ev, dbspk::EventCols cols, int64_t n,
                          int query,
                          uint64_t *c0, uint64_t *k0, uint64_t *v0, int64_t *w0,
                          uint64_t *c1, uint64_t *k1, uint64_t *v1, int64_t *w1) {
    // wave-uniform outer loop (wb is the wave's first row) so the ballots in
    // wave_append see every lane
    const int lane = threadIdx.x & (WAVE - 1);
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t wb = blockIdx.x * (int64_t)blockDim.x +
                      (threadIdx.x & ~(int64_t)(WAVE - 1));
         wb < n; wb += stride) {
        const int64_t i = wb + lane;
        dbsp_event e{};
        const bool act = i < n;
        if (act) {
            if (cols.kind) {  // columnized staging: 7 coalesced 8 B streams
                e.kind = cols.kind[i]; e.f0 = cols.f0[i]; e.f1 = cols.f1[i];
                e.f2 = cols.f2[i]; e.f3 = cols.f3[i]; e.f4 = cols.f4[i];
                e.w = cols.w[i];
            } else {
                e = ev[i];
            }
        }
        if (query == 3) {
            // q3.rs:37-49
            const bool p0 = act && e.kind == 1 && e.f2 == 10;
            const uint64_t pos0 = wave_append((unsigned long long *)c0, p0);
            if (p0) {
                k0[pos0] = e.f1; v0[pos0] = e.f0; w0[pos0] = e.w;
            }
            const bool p1 = act && e.kind == 0 &&
                            (e.f3 == 1 || e.f3 == 2 || e.f3 == 3);  // CA,ID,OR
            const uint64_t pos1 = wave_append((unsigned long long *)c1, p1);
            if (p1) {
                k1[pos1] = e.f0;
                v1[pos1] = (e.f1 << 8) | ((e.f2 & 0xF) << 4) | (e.f3 & 0xF);
                w1[pos1] = e.w;
            }
        } else if (query == 4) {
            // q4.rs:45-56: auctions by id (validity window + category packed
            // into the val), bids by auction (dt + price packed)
            const bool p0 = act && e.kind == 1;
            const uint64_t pos0 = wave_append((unsigned long long *)c0, p0);
            if (p0) {
                k0[pos0] = e.f0;
                v0[pos0] = (e.f3 << 28) | (((e.f4 - e.f3) & 0xFFFFFFull) << 4) |
                           (e.f2 & 0xFull);
                w0[pos0] = e.w;
            }
            const bool p1 = act && e.kind == 2;
            const uint64_t pos1 = wave_append((unsigned long long *)c1, p1);
            if (p1) {
                k1[pos1] = e.f0;
                v1[pos1] = (e.f3 << 27) | (e.f2 & 0x7FFFFFFull);
                w1[pos1] = e.w;
            }
        } else if (query == 6) {
            // q6.rs:46-57: auctions by id (seller + validity window packed:
            // dt 28 bits from bit 36, duration 16 bits, seller 20 bits),
            // bids by auction as q4
            const bool p0 = act && e.kind == 1;
            const uint64_t pos0 = wave_append((unsigned long long *)c0, p0);
            if (p0) {
                k0[pos0] = e.f0;
                v0[pos0] = (e.f3 << 36) | (((e.f4 - e.f3) & 0xFFFFull) << 20) |
                           (e.f1 & 0xFFFFFull);
                w0[pos0] = e.w;
            }
            const bool p1 = act && e.kind == 2;
            const uint64_t pos1 = wave_append((unsigned long long *)c1, p1);
            if (p1) {
                k1[pos1] = e.f0;
                v1[pos1] = (e.f3 << 27) | (e.f2 & 0x7FFFFFFull);
                w1[pos1] = e.w;
            }
        } else if (query == 5) {
            // q5.rs:79-83: bids by time
            const bool p0 = act && e.kind == 2;
            const uint64_t pos0 = wave_append((unsigned long long *)c0, p0);
            if (p0) {
                k0[pos0] = e.f3; v0[pos0] = e.f0; w0[pos0] = e.w;
            }
        } else if (query == 8) {
            // q8.rs:50-60
            const bool p0 = act && e.kind == 0;
            const uint64_t pos0 = wave_append((unsigned long long *)c0, p0);
            if (p0) {
                // (id, name) packed ORDER-PRESERVING as id*1024+name: the
                // name dictionary is the generator's ~1000-value space
                // (people.rs name draw), so 10 bits suffice — and the packed
                // range stays ~2^20 per tick instead of id<<32 spanning 2^42,
                // which halves the delta sort's digit passes
                k0[pos0] = e.f4; v0[pos0] = (e.f0 << 10) | (e.f1 & 0x3FFull);
                w0[pos0] = e.w;
            }
            const bool p1 = act && e.kind == 1;
            const uint64_t pos1 = wave_append((unsigned long long *)c1, p1);
            if (p1) {
                k1[pos1] = e.f3; v1[pos1] = e.f1; w1[pos1] = e.w;
            }
        }
    }
}

// generic per-row map for the small derived streams (q5/q8):
// mode 0: (k,v) -> (v>>10, (v&0x3FF)<<32 | (k&lo32))  [q8 people_by_id map_index]
// mode 1: (k,v) -> (v, 0)                             [q8 auctions map / q5 windowed-bids map]
// mode 2: (k,v) -> (0, v)                             [q5 map_index ((),count)]
// mode 3: (k,v) -> (v, k)                             [q5 by_count map_index]
// mode 5: (k=auction<<4|cat, v=price) -> (cat, 0) with w' = w*(price<<18|1)
//         [q4 average weigh: one linear pass accumulates (sum<<20)+count
//          exactly — price < 2^27 (price.rs) and count < 2^18]
// mode 6: (k=cat, v=(sum<<18)|count) -> (cat, sum/count)  [q4 average output,
//          integer division as the reference's isize avg (average.rs)]
__global__ void k_map(const uint64_t *k, const uint64_t *v, const int64_t *w,
                      int64_t n, int mode, uint64_t *ok, uint64_t *ov,
                      int64_t *ow) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint64_t kk = k[i], vv = v[i];
        uint64_t rk, rv;
        int64_t rw = w[i];
        switch (mode) {
            case 0: rk = vv >> 10; rv = ((vv & 0x3FFull) << 32) | (kk & 0xFFFFFFFFull); break;
            case 1: rk = vv; rv = 0; break;
            case 2: rk = 0; rv = vv; break;
            case 4: {  // weigh (aggregate/mod.rs:297-323): f(k,v)=f64(v); w' = f(k,v)*w
                rk = kk; rv = 0;
                double f = *(const double *)&vv;
                double wp = f * (double)w[i];
                rw = *(const int64_t *)&wp;
                break;
            }
            case 5:
                // (sum<<18)+count packed weight: price <= 10^8 (price.rs),
                // per-category sum <= ~1.2e13 -> sum<<18 < 2^62; count
                // (auctions per category) < 2^18
                rk = kk & 0xFull;
                rv = 0;
                rw = w[i] * (int64_t)((vv << 18) | 1ull);
                break;
            case 6: {
                const int64_t cnt = (int64_t)(vv & 0x3FFFFull);
                rk = kk;
                rv = cnt > 0 ? (vv >> 18) / (uint64_t)cnt : 0;
                break;
            }
            case 7:  // q6 map_index (q6.rs:92-94): winning bids keyed by
                     // seller, val (auction<<20)|price keeps cursor order
                     // by auction id for the last-10 fold
                rk = kk & 0xFFFFFull;
                rv = ((kk >> 20) << 27) | (vv & 0x7FFFFFFull);
                break;
            default: rk = vv; rv = kk; break;
        }
        ok[i] = rk;
        ov[i] = rv;
        ow[i] = rw;
    }
}

// ===========================================================================
// host-side primitive layer (exported via engine.cpp which owns dbsp_ctx);
// the functions below are internal helpers shared with engine.cpp
// ===========================================================================

#include "kernels_iface.hpp"

namespace dbspk {

dbsp_status scan_excl(hipStream_t s, const uint64_t *in, uint64_t *out,
                      int64_t n, uint64_t *h_total) {
    return scan_exclusive(s, in, out, n, h_total);
}

void fill_u64(hipStream_t s, uint64_t *p, uint64_t v, int64_t n) {
    k_fill_u64<<<grid_for(n), BLK, 0, s>>>(p, v, n);
}

// ---------------------------------------------------------------------------
// C5 synthetic operand generator (BASELINE configs[4]: 1B-row OrdIndexedZSet
// x 10M-row delta).  Counter-based splitmix64 keeps keys strictly increasing
// by construction (k_i = stride*i + h(i) % jitter, jitter < stride), so the
// generated batch is sorted-unique with no device scan or sort — the trace
// never stages through the host (24 GB at 1B rows).
// ---------------------------------------------------------------------------

__device__ __host__ inline uint64_t c5_mix(uint64_t x) {
    uint64_t z = x + 0x9E3779B97F4A7C15ull;
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
    return z ^ (z >> 31);
}

__global__ void k_c5_gen(int64_t n, uint64_t stride, uint64_t jitter,
                         uint64_t seed, int val_mode, uint64_t *k, uint64_t *v,
                         int64_t *w) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        const uint64_t h = c5_mix(seed + (uint64_t)i * 0x9E3779B97F4A7C15ull);
        k[i] = stride * (uint64_t)i + (jitter ? h % jitter : 0);
        if (val_mode == 0) {
            // uniform f64 in [0,1) as bit pattern (f64 vals)
            const double u = (double)(c5_mix(h) >> 11) * 0x1.0p-53;
            v[i] = __double_as_longlong(u);
        } else {
            v[i] = 0;
        }
        w[i] = 1;
    }
}

dbsp_status c5_gen_rows(hipStream_t s, int64_t n, uint64_t stride,
                        uint64_t jitter, uint64_t seed, int val_mode,
                        uint64_t *k, uint64_t *v, int64_t *w) {
    if (n <= 0) return DBSP_OK;
    k_c5_gen<<<grid_for(n), BLK, 0, s>>>(n, stride, jitter, seed, val_mode, k,
                                         v, w);
    return DBSP_OK;
}

// sort rows by (k major, v minor); ping-pong scratch must hold n rows
// (kk2/vv2/ww2).  Skips byte passes above the significant bytes of max(k)/max(v).
dbsp_status sort_rows(hipStream_t s, uint64_t *kk, uint64_t *vv, int64_t *ww,
                      int64_t n, uint64_t *kk2, uint64_t *vv2, int64_t *ww2,
                      bool *result_in_scratch) {
    *result_in_scratch = false;
    if (n <= 1) return DBSP_OK;
    // significant bytes from max values
    uint64_t *d_max;
    HIP_CHECK(dbspk::cache_malloc((void **)&d_max, 4 * sizeof(uint64_t), s));
    HIP_CHECK(hipMemsetAsync(d_max, 0, 2 * sizeof(uint64_t), s));
    HIP_CHECK(hipMemsetAsync(d_max + 2, 0xFF, 2 * sizeof(uint64_t), s));
    k_minmax_u64<<<grid_for(n), BLK, 0, s>>>(kk, vv, n, d_max);
    uint64_t h_max[4];
    HIP_CHECK(hipMemcpyAsync(h_max, d_max, 4 * sizeof(uint64_t),
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    HIP_CHECK(dbspk::cache_free(d_max, s));
    const uint64_t kbase = h_max[2], vbase = h_max[3];
    const uint64_t krange = h_max[0] - kbase, vrange = h_max[1] - vbase;
    int kbytes = 0, vbytes = 0;
    while (kbytes < 8 && (krange >> (8 * kbytes)) != 0) kbytes++;
    while (vbytes < 8 && (vrange >> (8 * vbytes)) != 0) vbytes++;

    int64_t nblocks = ceil_div(n, SORT_TILE);
    uint64_t *counts;
    HIP_CHECK(dbspk::cache_malloc((void **)&counts, (int64_t)256 * nblocks * sizeof(uint64_t), s));

    uint64_t *src_k = kk, *src_v = vv; int64_t *src_w = ww;
    uint64_t *dst_k = kk2, *dst_v = vv2; int64_t *dst_w = ww2;
    for (int byte = 0; byte < 16; byte++) {
        bool is_v = byte < 8;
        if (is_v && (byte & 7) >= vbytes) continue;
        if (!is_v && (byte & 7) >= kbytes) continue;
        k_radix_hist<<<dim3((uint32_t)nblocks), BLK, 0, s>>>(
            src_k, src_v, n, byte, kbase, vbase, nblocks, counts);
        dbsp_status st = scan_exclusive(s, counts, counts, 256 * nblocks, nullptr);
        if (st != DBSP_OK) return st;
        k_radix_scatter<<<dim3((uint32_t)nblocks), BLK, 0, s>>>(
            src_k, src_v, src_w, n, byte, kbase, vbase, nblocks, counts, dst_k,
            dst_v, dst_w);
        uint64_t *t;
        int64_t *tw;
        t = src_k; src_k = dst_k; dst_k = t;
        t = src_v; src_v = dst_v; dst_v = t;
        tw = src_w; src_w = dst_w; dst_w = tw;
    }
    HIP_CHECK(dbspk::cache_free(counts, s));
    *result_in_scratch = (src_k != kk);
    return DBSP_OK;
}

// consolidate SORTED rows into freshly allocated output; returns exact length
dbsp_status consolidate_sorted(hipStream_t s, const uint64_t *kk,
                               const uint64_t *vv, const int64_t *ww, int64_t n,
                               uint64_t **ok, uint64_t **ov, int64_t **ow,
                               int64_t *out_n) {
    if (n == 0) {
        *ok = nullptr; *ov = nullptr; *ow = nullptr; *out_n = 0;
        return DBSP_OK;
    }
    uint64_t *flags, *fscan;
    HIP_CHECK(dbspk::cache_malloc((void **)&flags, n * sizeof(uint64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&fscan, n * sizeof(uint64_t), s));
    k_head_flags<<<grid_for(n), BLK, 0, s>>>(kk, vv, n, flags);
    uint64_t nseg = 0;
    dbsp_status st = scan_exclusive(s, flags, fscan, n, &nseg);
    if (st != DBSP_OK) return st;
    uint64_t *sk, *sv; int64_t *sw;
    HIP_CHECK(dbspk::cache_malloc((void **)&sk, nseg * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&sv, nseg * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&sw, nseg * sizeof(int64_t) + 8, s));
    HIP_CHECK(hipMemsetAsync(sw, 0, nseg * sizeof(int64_t), s));
    k_seg_accum<<<grid_for(n), BLK, 0, s>>>(kk, vv, ww, fscan, flags, n, sk, sv, sw);
    // drop zero-weight segments
    uint64_t *nzflags, *nzscan;
    HIP_CHECK(dbspk::cache_malloc((void **)&nzflags, nseg * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&nzscan, nseg * sizeof(uint64_t) + 8, s));
    k_nonzero_flags<<<grid_for(nseg), BLK, 0, s>>>(sw, nseg, nzflags);
    uint64_t nout = 0;
    st = scan_exclusive(s, nzflags, nzscan, nseg, &nout);
    if (st != DBSP_OK) return st;
    uint64_t *rk, *rv; int64_t *rw;
    HIP_CHECK(dbspk::cache_malloc((void **)&rk, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rv, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rw, nout * sizeof(int64_t) + 8, s));
    k_compact<<<grid_for(nseg), BLK, 0, s>>>(sk, sv, sw, nzflags, nzscan, nseg,
                                             rk, rv, rw);
    HIP_CHECK(dbspk::cache_free(flags, s));
    HIP_CHECK(dbspk::cache_free(fscan, s));
    HIP_CHECK(dbspk::cache_free(sk, s));
    HIP_CHECK(dbspk::cache_free(sv, s));
    HIP_CHECK(dbspk::cache_free(sw, s));
    HIP_CHECK(dbspk::cache_free(nzflags, s));
    HIP_CHECK(dbspk::cache_free(nzscan, s));
    *ok = rk; *ov = rv; *ow = rw; *out_n = (int64_t)nout;
    return DBSP_OK;
}

template <typename W>
static dbsp_status merge_rows_t(hipStream_t s, const uint64_t *ak,
                                const uint64_t *av, const W *aw, int64_t na,
                                const uint64_t *bk, const uint64_t *bv,
                                const W *bw, int64_t nb, uint64_t **ok,
                                uint64_t **ov, W **ow, int64_t *out_n) {
    int64_t total = na + nb;
    if (total == 0) {
        *ok = nullptr; *ov = nullptr; *ow = nullptr; *out_n = 0;
        return DBSP_OK;
    }
    // Default: two-pass count -> device scan -> emit (k_mp_merge_onepass
    // modes 1+2 — branchless capture walk in both phases).  The single-pass
    // decoupled-lookback variant (mode 0) reads each input byte once but its
    // lookback protocol measures SLOWER than the count re-read at 1B rows
    // (granule-load throughput-bound; see the kernel comment) — opt in with
    // DBSP_MERGE_ONEPASS=1; a poisoned lookback falls back to two-pass.
    static const bool twopass = []() {
        const char *e = getenv("DBSP_MERGE_ONEPASS");
        return !(e && e[0] == '1');
    }();
    static const int op_tile = []() {
        const char *e = getenv("DBSP_MERGE_OP_TILE");
        return (e && atoi(e) == 2048) ? 2048 : MP_TILE;
    }();
    const int64_t tile = op_tile;
    int64_t nblocks = ceil_div(total, tile);
    int64_t *pa, *pb;
    HIP_CHECK(dbspk::cache_malloc((void **)&pa, (nblocks + 1) * sizeof(int64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&pb, (nblocks + 1) * sizeof(int64_t), s));
    k_mp_partition<<<grid_for(nblocks + 1), BLK, 0, s>>>(ak, av, na, bk, bv, nb,
                                                         nblocks, tile, pa, pb);
    uint64_t *rk, *rv;
    W *rw;
    const dim3 g((uint32_t)nblocks);
    if (!twopass) {
        unsigned long long *state;
        HIP_CHECK(dbspk::cache_malloc((void **)&state, (nblocks + 2) * sizeof(uint64_t), s));
        HIP_CHECK(hipMemsetAsync(state, 0, (nblocks + 2) * sizeof(uint64_t), s));
        HIP_CHECK(dbspk::cache_malloc((void **)&rk, total * sizeof(uint64_t) + 8, s));
        HIP_CHECK(dbspk::cache_malloc((void **)&rv, total * sizeof(uint64_t) + 8, s));
        HIP_CHECK(dbspk::cache_malloc((void **)&rw, total * sizeof(W) + 8, s));
        static const bool nolb = []() {  // timing diagnostic (wrong results)
            const char *e = getenv("DBSP_MERGE_NOLB");
            return e && e[0] == '1';
        }();
        const size_t smem = 3 * (tile + 2) * sizeof(uint64_t);
        if (nolb && tile == 2048)
            k_mp_merge_onepass<W, 2048, 0, true><<<g, MP_THREADS, smem, s>>>(
                ak, av, aw, na, bk, bv, bw, nb, pa, pb, state, nblocks, rk, rv, rw);
        else if (nolb)
            k_mp_merge_onepass<W, MP_TILE, 0, true><<<g, MP_THREADS, smem, s>>>(
                ak, av, aw, na, bk, bv, bw, nb, pa, pb, state, nblocks, rk, rv, rw);
        else if (tile == 2048)
            k_mp_merge_onepass<W, 2048, 0><<<g, MP_THREADS, smem, s>>>(
                ak, av, aw, na, bk, bv, bw, nb, pa, pb, state, nblocks, rk, rv, rw);
        else
            k_mp_merge_onepass<W, MP_TILE, 0><<<g, MP_THREADS, smem, s>>>(
                ak, av, aw, na, bk, bv, bw, nb, pa, pb, state, nblocks, rk, rv, rw);
        unsigned long long h_state[2];
        HIP_CHECK(hipMemcpyAsync(&h_state[0], state + 1, sizeof(uint64_t),
                                 hipMemcpyDeviceToHost, s));
        HIP_CHECK(hipMemcpyAsync(&h_state[1], state + 2 + (nblocks - 1),
                                 sizeof(uint64_t), hipMemcpyDeviceToHost, s));
        HIP_CHECK(hipStreamSynchronize(s));
        HIP_CHECK(dbspk::cache_free(state, s));
        if (h_state[0] == 0 && (h_state[1] & 3ull) == 2ull) {
            HIP_CHECK(dbspk::cache_free(pa, s));
            HIP_CHECK(dbspk::cache_free(pb, s));
            *ok = rk; *ov = rv; *ow = rw; *out_n = (int64_t)(h_state[1] >> 2);
            return DBSP_OK;
        }
        // poisoned lookback: free the optimistic buffers and fall through to
        // the two-pass pipeline on the partitions already computed
        (void)dbspk::cache_free(rk, s);
        (void)dbspk::cache_free(rv, s);
        (void)dbspk::cache_free(rw, s);
    }
    uint64_t *counts;
    HIP_CHECK(dbspk::cache_malloc((void **)&counts, (nblocks + 1) * sizeof(uint64_t), s));
    const size_t smem_count = 2 * (tile + 2) * sizeof(uint64_t);
    if (tile == 2048)
        k_mp_merge_onepass<W, 2048, 1><<<g, MP_THREADS, smem_count, s>>>(
            ak, av, aw, na, bk, bv, bw, nb, pa, pb,
            (unsigned long long *)counts, nblocks, nullptr, nullptr, nullptr);
    else
        k_mp_merge_onepass<W, MP_TILE, 1><<<g, MP_THREADS, smem_count, s>>>(
            ak, av, aw, na, bk, bv, bw, nb, pa, pb,
            (unsigned long long *)counts, nblocks, nullptr, nullptr, nullptr);
    uint64_t nout = 0;
    dbsp_status st = scan_exclusive(s, counts, counts, nblocks, &nout);
    if (st != DBSP_OK) return st;
    HIP_CHECK(dbspk::cache_malloc((void **)&rk, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rv, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rw, nout * sizeof(W) + 8, s));
    const size_t smem_emit = 3 * (tile + 2) * sizeof(uint64_t);
    if (nout > 0) {
        if (tile == 2048)
            k_mp_merge_onepass<W, 2048, 2><<<g, MP_THREADS, smem_emit, s>>>(
                ak, av, aw, na, bk, bv, bw, nb, pa, pb,
                (unsigned long long *)counts, nblocks, rk, rv, rw);
        else
            k_mp_merge_onepass<W, MP_TILE, 2><<<g, MP_THREADS, smem_emit, s>>>(
                ak, av, aw, na, bk, bv, bw, nb, pa, pb,
                (unsigned long long *)counts, nblocks, rk, rv, rw);
    }
    HIP_CHECK(dbspk::cache_free(pa, s));
    HIP_CHECK(dbspk::cache_free(pb, s));
    HIP_CHECK(dbspk::cache_free(counts, s));
    *ok = rk; *ov = rv; *ow = rw; *out_n = (int64_t)nout;
    return DBSP_OK;
}

dbsp_status merge_rows(hipStream_t s, const uint64_t *ak, const uint64_t *av,
                       const int64_t *aw, int64_t na, const uint64_t *bk,
                       const uint64_t *bv, const int64_t *bw, int64_t nb,
                       uint64_t **ok, uint64_t **ov, int64_t **ow,
                       int64_t *out_n) {
    return merge_rows_t<int64_t>(s, ak, av, aw, na, bk, bv, bw, nb, ok, ov, ow,
                                 out_n);
}

dbsp_status merge_rows_f64(hipStream_t s, const uint64_t *ak, const uint64_t *av,
                           const double *aw, int64_t na, const uint64_t *bk,
                           const uint64_t *bv, const double *bw, int64_t nb,
                           uint64_t **ok, uint64_t **ov, double **ow,
                           int64_t *out_n) {
    return merge_rows_t<double>(s, ak, av, aw, na, bk, bv, bw, nb, ok, ov, ow,
                                out_n);
}


dbsp_status minmax_rows(hipStream_t s, const uint64_t *k, const uint64_t *v,
                        int64_t n, uint64_t mm[4]) {
    unsigned long long *d;
    HIP_CHECK(dbspk::cache_malloc((void **)&d, 4 * sizeof(uint64_t), s));
    const unsigned long long init[4] = {~0ull, ~0ull, 0ull, 0ull};
    HIP_CHECK(hipMemcpyAsync(d, init, sizeof(init), hipMemcpyHostToDevice, s));
    // cap the grid: each block ends with 8 global atomics on 4 hot words,
    // and a full grid_for() grid serializes ~15 us on them alone
    k_minmax_rows<<<dim3(std::min(grid_for(n).x, 64u)), BLK, 0, s>>>(k, v, n,
                                                                     d);
    HIP_CHECK(hipMemcpyAsync(mm, d, 4 * sizeof(uint64_t),
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    HIP_CHECK(dbspk::cache_free(d, s));
    return DBSP_OK;
}

// chained variant: n read from the device, results left in mm_dev[0..3]
// (caller reads them at its own sync); grid sized by the capacity bound
dbsp_status minmax_rows_chain(hipStream_t s, const uint64_t *k,
                              const uint64_t *v, int64_t cap,
                              const int64_t *n_dev,
                              unsigned long long *mm_dev) {
    const unsigned long long init[4] = {~0ull, ~0ull, 0ull, 0ull};
    HIP_CHECK(hipMemcpyAsync(mm_dev, init, sizeof(init), hipMemcpyHostToDevice,
                             s));
    k_minmax_rows<<<dim3(std::min(grid_for(cap).x, 64u)), BLK, 0, s>>>(
        k, v, 0, mm_dev, n_dev);
    return DBSP_OK;
}

// chained variant: no host sync — the consolidated length lands in
// *out_n_dev for the caller's one tick sync (the emit grid covers the whole
// histogram range, so nothing here needs the total on the host)
dbsp_status sort_cons_dense_chain(hipStream_t s, const uint64_t *k,
                                  const uint64_t *v, const int64_t *w,
                                  int64_t n, uint64_t kbase, uint64_t vbase,
                                  int64_t kspan, int64_t vspan, uint64_t *ok,
                                  uint64_t *ov, int64_t *ow,
                                  int64_t *out_n_dev) {
    const int64_t r = kspan * vspan;
    unsigned long long *wbuf;
    uint64_t *flags;
    HIP_CHECK(dbspk::cache_malloc((void **)&wbuf, r * sizeof(uint64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&flags, (r + 1) * sizeof(uint64_t), s));
    HIP_CHECK(hipMemsetAsync(wbuf, 0, r * sizeof(uint64_t), s));
    k_hist_add<<<grid_for(n), BLK, 0, s>>>(k, v, w, n, kbase, vbase,
                                           (uint64_t)vspan, wbuf);
    k_hist_flags<<<grid_for(r), BLK, 0, s>>>(wbuf, r, flags);
    dbsp_status st = scan_exclusive(s, flags, flags, r, nullptr);
    if (st != DBSP_OK) return st;
    k_hist_total<<<1, 1, 0, s>>>(wbuf, flags, r, out_n_dev);
    k_hist_emit<<<grid_for(r), BLK, 0, s>>>(wbuf, flags, r, kbase, vbase,
                                            (uint64_t)vspan, ok, ov, ow);
    HIP_CHECK(dbspk::cache_free(wbuf, s));
    HIP_CHECK(dbspk::cache_free(flags, s));
    return DBSP_OK;
}

dbsp_status sort_cons_dense(hipStream_t s, const uint64_t *k, const uint64_t *v,
                            const int64_t *w, int64_t n, uint64_t kbase,
                            uint64_t vbase, int64_t kspan, int64_t vspan,
                            uint64_t *ok, uint64_t *ov, int64_t *ow,
                            int64_t *out_n) {
    const int64_t r = kspan * vspan;
    unsigned long long *wbuf;
    uint64_t *flags;
    HIP_CHECK(dbspk::cache_malloc((void **)&wbuf, r * sizeof(uint64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&flags, (r + 1) * sizeof(uint64_t), s));
    HIP_CHECK(hipMemsetAsync(wbuf, 0, r * sizeof(uint64_t), s));
    k_hist_add<<<grid_for(n), BLK, 0, s>>>(k, v, w, n, kbase, vbase,
                                           (uint64_t)vspan, wbuf);
    k_hist_flags<<<grid_for(r), BLK, 0, s>>>(wbuf, r, flags);
    uint64_t nout = 0;
    dbsp_status st = scan_exclusive(s, flags, flags, r, &nout);
    if (st != DBSP_OK) return st;
    if (nout > 0)
        k_hist_emit<<<grid_for(r), BLK, 0, s>>>(wbuf, flags, r, kbase, vbase,
                                                (uint64_t)vspan, ok, ov, ow);
    HIP_CHECK(dbspk::cache_free(wbuf, s));
    HIP_CHECK(dbspk::cache_free(flags, s));
    *out_n = (int64_t)nout;
    return DBSP_OK;
}

dbsp_status sort_cons_small_batch(hipStream_t s, const SortArgs &args) {
    for (int b = 0; b < args.nb; b++)
        if (args.n[b] > FUSE_MAX) return DBSP_ERR_INVALID;
    if (args.nb == 0) return DBSP_OK;
    k_sort_cons_small<int64_t>
        <<<dim3((uint32_t)args.nb), FUSE_THREADS, 0, s>>>(args);
    return DBSP_OK;
}

dbsp_status sort_cons_small_batch_f64(hipStream_t s, const SortArgs &args) {
    for (int b = 0; b < args.nb; b++)
        if (args.n[b] > FUSE_MAX) return DBSP_ERR_INVALID;
    if (args.nb == 0) return DBSP_OK;
    k_sort_cons_small<double>
        <<<dim3((uint32_t)args.nb), FUSE_THREADS, 0, s>>>(args);
    return DBSP_OK;
}

dbsp_status merge_small_batch(hipStream_t s, const MergeArgs &args) {
    if (args.np == 0) return DBSP_OK;
    for (int i = 0; i < args.np; i++)
        if (args.na[i] + args.nb[i] > 4 * FUSE_MAX) return DBSP_ERR_INVALID;
    k_merge_small<int64_t><<<dim3((uint32_t)args.np), FUSE_THREADS, 0, s>>>(args);
    return DBSP_OK;
}

dbsp_status merge_small_batch_f64(hipStream_t s, const MergeArgs &args) {
    if (args.np == 0) return DBSP_OK;
    for (int i = 0; i < args.np; i++)
        if (args.na[i] + args.nb[i] > 4 * FUSE_MAX) return DBSP_ERR_INVALID;
    k_merge_small<double><<<dim3((uint32_t)args.np), FUSE_THREADS, 0, s>>>(args);
    return DBSP_OK;
}

// launch generation for the fused barrier's tagged counts — shared across
// the int64/f64 instantiations so two launchers can never reuse a tag the
// other just published into the same scratch
static std::atomic<uint32_t> g_mid_gen{0};

dbsp_status merge_mid_batch(hipStream_t s, const MergeArgs &args,
                            int64_t *scratch) {
    if (args.np == 0) return DBSP_OK;
    const uint32_t gen = ++g_mid_gen;
    dim3 grid(MERGE_MID_WGS, (uint32_t)args.np);
    k_merge_mid_fused<int64_t><<<grid, FUSE_THREADS, 0, s>>>(args, scratch,
                                                             gen);
    return DBSP_OK;
}

dbsp_status merge_mid_batch_f64(hipStream_t s, const MergeArgs &args,
                                int64_t *scratch) {
    if (args.np == 0) return DBSP_OK;
    const uint32_t gen = ++g_mid_gen;
    dim3 grid(MERGE_MID_WGS, (uint32_t)args.np);
    k_merge_mid_fused<double><<<grid, FUSE_THREADS, 0, s>>>(args, scratch,
                                                            gen);
    return DBSP_OK;
}

dbsp_status join_spine_rows(hipStream_t s, const uint64_t *dk,
                            const uint64_t *dv, const int64_t *dw, int64_t nd,
                            const TraceArgs &t, int proj, uint64_t param,
                            uint64_t **ok, uint64_t **ov, int64_t **ow,
                            int64_t *out_n) {
    if (nd == 0 || t.nb == 0) {
        *ok = nullptr; *ov = nullptr; *ow = nullptr; *out_n = 0;
        return DBSP_OK;
    }
    uint32_t *cnts;
    uint64_t *counts;
    HIP_CHECK(dbspk::cache_malloc((void **)&cnts, (int64_t)nd * t.nb * sizeof(uint32_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&counts, (nd + 1) * sizeof(uint64_t), s));
    k_join_count_multi<<<grid_for(nd), BLK, 0, s>>>(dk, nd, t, cnts, counts);
    uint64_t nout = 0;
    dbsp_status st = scan_exclusive(s, counts, counts, nd, &nout);
    if (st != DBSP_OK) return st;
    uint64_t *rk, *rv; int64_t *rw;
    HIP_CHECK(dbspk::cache_malloc((void **)&rk, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rv, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rw, nout * sizeof(int64_t) + 8, s));
    if (nout > 0)
        k_join_emit_multi<<<grid_for((int64_t)nout), BLK, 0, s>>>(
            dk, dv, dw, nd, t, cnts, counts, (int64_t)nout, proj, param, rk, rv,
            rw);
    HIP_CHECK(dbspk::cache_free(cnts, s));
    HIP_CHECK(dbspk::cache_free(counts, s));
    *ok = rk; *ov = rv; *ow = rw; *out_n = (int64_t)nout;
    return DBSP_OK;
}

dbsp_status join_count_scan_batch(hipStream_t s, const JoinCountArgs &args) {
    if (args.np == 0) return DBSP_OK;
    for (int i = 0; i < args.np; i++)
        if (args.nd[i] > FUSE_MAX) return DBSP_ERR_INVALID;
    k_join_probe_multi<<<dim3((uint32_t)(args.np * JPROBE_BLOCKS)), BLK, 0,
                         s>>>(args);
    k_join_scan_small<<<dim3((uint32_t)args.np), FUSE_THREADS, 0, s>>>(args);
    return DBSP_OK;
}


dbsp_status emit_bases(hipStream_t s, const int64_t *totals, int np,
                       int64_t cap, int64_t *bases, int64_t *out_total,
                       int64_t *out_flag) {
    k_emit_bases<<<1, 1, 0, s>>>(totals, np, cap, bases, out_total, out_flag);
    return DBSP_OK;
}

dbsp_status join_emit_chain(hipStream_t s, const uint64_t *dk,
                            const uint64_t *dv, const int64_t *dw,
                            const int64_t *nd_dev, const TraceArgs &t,
                            const int64_t *tn_dev, const uint32_t *cnts,
                            const uint64_t *offsets, const int64_t *total_dev,
                            const int64_t *base_dev, const int64_t *flag_dev,
                            int64_t grid_cap, int proj, uint64_t param,
                            uint64_t *ok, uint64_t *ov, int64_t *ow) {
    k_join_emit_chain<<<grid_for(grid_cap), BLK, 0, s>>>(
        dk, dv, dw, nd_dev, t, tn_dev, cnts, offsets, total_dev, base_dev,
        flag_dev, proj, param, ok, ov, ow);
    return DBSP_OK;
}

dbsp_status join_emit_fused(hipStream_t s, const FusedEmitArgs &a) {
    k_join_emit_fused<<<grid_for(a.cap), BLK, 0, s>>>(a);
    return DBSP_OK;
}

dbsp_status join_emit_prepared(hipStream_t s, const uint64_t *dk,
                               const uint64_t *dv, const int64_t *dw,
                               int64_t nd, const TraceArgs &t,
                               const uint32_t *cnts, const uint64_t *offsets,
                               int64_t n_out, int proj, uint64_t param,
                               uint64_t *ok, uint64_t *ov, int64_t *ow) {
    if (n_out > 0)
        k_join_emit_multi<<<grid_for(n_out), BLK, 0, s>>>(
            dk, dv, dw, nd, t, cnts, offsets, n_out, proj, param, ok, ov, ow);
    return DBSP_OK;
}

dbsp_status join_rows(hipStream_t s, const uint64_t *dk, const uint64_t *dv,
                      const int64_t *dw, int64_t nd, const uint64_t *tk,
                      const uint64_t *tv, const int64_t *tw, int64_t nt,
                      int proj, uint64_t param, uint64_t **ok, uint64_t **ov,
                      int64_t **ow, int64_t *out_n) {
    if (nd == 0 || nt == 0) {
        *ok = nullptr; *ov = nullptr; *ow = nullptr; *out_n = 0;
        return DBSP_OK;
    }
    uint64_t *counts, *starts;
    HIP_CHECK(dbspk::cache_malloc((void **)&counts, (nd + 1) * sizeof(uint64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&starts, nd * sizeof(uint64_t), s));
    k_join_count<<<grid_for(nd), BLK, 0, s>>>(dk, nd, tk, nt, counts, starts);
    uint64_t nout = 0;
    dbsp_status st = scan_exclusive(s, counts, counts, nd, &nout);
    if (st != DBSP_OK) return st;
    uint64_t *rk, *rv; int64_t *rw;
    HIP_CHECK(dbspk::cache_malloc((void **)&rk, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rv, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rw, nout * sizeof(int64_t) + 8, s));
    if (nout > 0)
        k_join_emit<<<grid_for((int64_t)nout), BLK, 0, s>>>(
            dk, dv, dw, nd, tv, tw, counts, starts, (int64_t)nout, proj, param,
            rk, rv, rw);
    HIP_CHECK(dbspk::cache_free(counts, s));
    HIP_CHECK(dbspk::cache_free(starts, s));
    *ok = rk; *ov = rv; *ow = rw; *out_n = (int64_t)nout;
    return DBSP_OK;
}

template <int MODE>
static dbsp_status agg_upsert_impl(hipStream_t s, const uint64_t *keys,
                                   int64_t nd, const uint64_t *ik,
                                   const uint64_t *iv, const int64_t *iw,
                                   int64_t ni, const uint64_t *tok,
                                   const uint64_t *tov, const int64_t *tow,
                                   int64_t no, uint64_t **ok, uint64_t **ov,
                                   int64_t **ow, int64_t *out_n) {
    if (nd == 0) {
        *ok = nullptr; *ov = nullptr; *ow = nullptr; *out_n = 0;
        return DBSP_OK;
    }
    uint64_t *counts, *istart, *ostart, *newval, *hasnew, *offsets;
    HIP_CHECK(dbspk::cache_malloc((void **)&counts, nd * sizeof(uint64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&istart, nd * sizeof(uint64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&ostart, nd * sizeof(uint64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&newval, nd * sizeof(uint64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&hasnew, nd * sizeof(uint64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&offsets, nd * sizeof(uint64_t), s));
    k_agg_count<MODE><<<grid_for(nd), BLK, 0, s>>>(keys, nd, ik, iw, ni, tok, no,
                                                  counts, istart, ostart, newval,
                                                  hasnew);
    uint64_t nout = 0;
    dbsp_status st = scan_exclusive(s, counts, offsets, nd, &nout);
    if (st != DBSP_OK) return st;
    uint64_t *rk, *rv; int64_t *rw;
    HIP_CHECK(dbspk::cache_malloc((void **)&rk, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rv, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rw, nout * sizeof(int64_t) + 8, s));
    if (nout > 0)
        k_agg_emit<MODE><<<grid_for(nd), BLK, 0, s>>>(keys, nd, iv, tov, tow,
                                                     offsets, counts, istart,
                                                     ostart, newval, hasnew, rk,
                                                     rv, rw);
    HIP_CHECK(dbspk::cache_free(counts, s));
    HIP_CHECK(dbspk::cache_free(istart, s));
    HIP_CHECK(dbspk::cache_free(ostart, s));
    HIP_CHECK(dbspk::cache_free(newval, s));
    HIP_CHECK(dbspk::cache_free(hasnew, s));
    HIP_CHECK(dbspk::cache_free(offsets, s));
    *ok = rk; *ov = rv; *ow = rw; *out_n = (int64_t)nout;
    return DBSP_OK;
}

dbsp_status distinct_inc_rows(hipStream_t s, const uint64_t *dk,
                              const uint64_t *dv, const int64_t *dw,
                              int64_t nd, const TraceArgs &t, uint64_t **ok,
                              uint64_t **ov, int64_t **ow, int64_t *out_n) {
    if (nd == 0) {
        *ok = nullptr; *ov = nullptr; *ow = nullptr; *out_n = 0;
        return DBSP_OK;
    }
    uint64_t *flags, *fscan;
    int64_t *outw;
    HIP_CHECK(dbspk::cache_malloc((void **)&flags, nd * sizeof(uint64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&fscan, nd * sizeof(uint64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&outw, nd * sizeof(int64_t), s));
    k_distinct_count<<<grid_for(nd), BLK, 0, s>>>(dk, dv, dw, nd, t, flags, outw);
    uint64_t nout = 0;
    dbsp_status st = scan_exclusive(s, flags, fscan, nd, &nout);
    if (st != DBSP_OK) return st;
    uint64_t *rk, *rv;
    int64_t *rw;
    HIP_CHECK(dbspk::cache_malloc((void **)&rk, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rv, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rw, nout * sizeof(int64_t) + 8, s));
    k_compact<<<grid_for(nd), BLK, 0, s>>>(dk, dv, outw, flags, fscan, nd, rk,
                                           rv, rw);
    HIP_CHECK(dbspk::cache_free(flags, s));
    HIP_CHECK(dbspk::cache_free(fscan, s));
    HIP_CHECK(dbspk::cache_free(outw, s));
    *ok = rk; *ov = rv; *ow = rw; *out_n = (int64_t)nout;
    return DBSP_OK;
}

dbsp_status agg_linear_upsert_rows(hipStream_t s, const uint64_t *keys,
                                   int64_t nd, const uint64_t *ik,
                                   const uint64_t *iv, const int64_t *iw,
                                   int64_t ni, const uint64_t *tok,
                                   const uint64_t *tov, const int64_t *tow,
                                   int64_t no, uint64_t **ok, uint64_t **ov,
                                   int64_t **ow, int64_t *out_n) {
    return agg_upsert_impl<0>(s, keys, nd, ik, iv, iw, ni, tok, tov, tow, no,
                              ok, ov, ow, out_n);
}

dbsp_status agg_max_upsert_rows(hipStream_t s, const uint64_t *keys, int64_t nd,
                                const uint64_t *ik, const uint64_t *iv,
                                const int64_t *iw, int64_t ni,
                                const uint64_t *tok, const uint64_t *tov,
                                const int64_t *tow, int64_t no, uint64_t **ok,
                                uint64_t **ov, int64_t **ow, int64_t *out_n) {
    return agg_upsert_impl<1>(s, keys, nd, ik, iv, iw, ni, tok, tov, tow, no,
                              ok, ov, ow, out_n);
}

dbsp_status agg_last10_upsert_rows(hipStream_t s, const uint64_t *keys,
                                   int64_t nd, const uint64_t *ik,
                                   const uint64_t *iv, const int64_t *iw,
                                   int64_t ni, const uint64_t *tok,
                                   const uint64_t *tov, const int64_t *tow,
                                   int64_t no, uint64_t **ok, uint64_t **ov,
                                   int64_t **ow, int64_t *out_n) {
    return agg_upsert_impl<2>(s, keys, nd, ik, iv, iw, ni, tok, tov, tow, no,
                              ok, ov, ow, out_n);
}

dbsp_status window_rows(hipStream_t s, const uint64_t *tk, const uint64_t *tv,
                        const int64_t *tw, int64_t nt, const uint64_t *bk,
                        const uint64_t *bv, const int64_t *bw, int64_t nb,
                        int have_prev, uint64_t s0, uint64_t e0, uint64_t s1,
                        uint64_t e1, uint64_t **ok, uint64_t **ov, int64_t **ow,
                        int64_t *out_n) {
    int64_t *d_ranges;
    HIP_CHECK(dbspk::cache_malloc((void **)&d_ranges, 8 * sizeof(int64_t), s));
    k_window_ranges<<<1, 1, 0, s>>>(tk, nt, bk, nb, have_prev, s0, e0, s1, e1,
                                    d_ranges);
    int64_t h_ranges[8];
    HIP_CHECK(hipMemcpyAsync(h_ranges, d_ranges, sizeof(h_ranges),
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    int64_t total = (h_ranges[1] - h_ranges[0]) + (h_ranges[3] - h_ranges[2]) +
                    (h_ranges[5] - h_ranges[4]) + (h_ranges[7] - h_ranges[6]);
    uint64_t *rk, *rv; int64_t *rw;
    HIP_CHECK(dbspk::cache_malloc((void **)&rk, total * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rv, total * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rw, total * sizeof(int64_t) + 8, s));
    if (total > 0)
        k_window_emit<<<grid_for(total), BLK, 0, s>>>(tk, tv, tw, bk, bv, bw,
                                                      d_ranges, rk, rv, rw);
    HIP_CHECK(dbspk::cache_free(d_ranges, s));
    *ok = rk; *ov = rv; *ow = rw; *out_n = total;
    return DBSP_OK;
}

dbsp_status window_ranges_multi(hipStream_t s, const TraceArgs &t,
                                const uint64_t *bk, int64_t bn, int have_prev,
                                uint64_t s0, uint64_t e0, uint64_t s1,
                                uint64_t e1, int64_t *table, int64_t *d_total) {
    k_window_ranges_multi<<<1, BLK, 0, s>>>(t, bk, bn, have_prev, s0, e0, s1,
                                            e1, table, d_total, nullptr,
                                            nullptr);
    return DBSP_OK;
}

dbsp_status wm_update(hipStream_t s, const uint64_t *ak, const int64_t *n_dev,
                      uint64_t width, uint64_t tumble, uint64_t lag,
                      unsigned long long *state, unsigned long long *bounds) {
    k_wm_update<<<1, 1, 0, s>>>(ak, n_dev, width, tumble, lag, state, bounds);
    return DBSP_OK;
}

dbsp_status wm_update_n(hipStream_t s, const uint64_t *ak, int64_t n,
                        uint64_t width, uint64_t tumble, uint64_t lag,
                        unsigned long long *state,
                        unsigned long long *bounds) {
    k_wm_update_n<<<1, 1, 0, s>>>(ak, n, width, tumble, lag, state, bounds);
    return DBSP_OK;
}

dbsp_status wm_update_g(hipStream_t s, const unsigned long long *gmax_dev,
                        uint64_t width, uint64_t tumble, uint64_t lag,
                        unsigned long long *state,
                        unsigned long long *bounds) {
    k_wm_update_g<<<1, 1, 0, s>>>(gmax_dev, width, tumble, lag, state,
                                  bounds);
    return DBSP_OK;
}

dbsp_status window_ranges_chain(hipStream_t s, const TraceArgs &t,
                                const uint64_t *bk, int64_t bn,
                                const int64_t *bn_dev,
                                const unsigned long long *bounds,
                                int64_t *table, int64_t *d_total) {
    k_window_ranges_multi<<<1, BLK, 0, s>>>(t, bk, bn, 0, 0, 0, 0, 0, table,
                                            d_total, bn_dev, bounds);
    return DBSP_OK;
}

dbsp_status window_emit_multi(hipStream_t s, const TraceArgs &t,
                              const uint64_t *bk, const uint64_t *bv,
                              const int64_t *bw, const int64_t *table,
                              int nreg, int64_t total, uint64_t *ok,
                              uint64_t *ov, int64_t *ow) {
    if (total > 0)
        k_window_emit_multi<<<grid_for(total), BLK, 0, s>>>(t, bk, bv, bw,
                                                            table, nreg, total,
                                                            ok, ov, ow);
    return DBSP_OK;
}

dbsp_status shard_rows(hipStream_t s, const uint64_t *k, const uint64_t *v,
                       const int64_t *w, int64_t n, int nshards, uint64_t *ok,
                       uint64_t *ov, int64_t *ow, int64_t *h_offsets) {
    uint64_t *hist;
    HIP_CHECK(dbspk::cache_malloc((void **)&hist, (nshards + 1) * sizeof(uint64_t), s));
    HIP_CHECK(hipMemsetAsync(hist, 0, (nshards + 1) * sizeof(uint64_t), s));
    if (n > 0) k_shard_hist<<<grid_for(n), BLK, 0, s>>>(k, n, nshards, hist);
    uint64_t h_hist[64];
    HIP_CHECK(hipMemcpyAsync(h_hist, hist, nshards * sizeof(uint64_t),
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    int64_t acc = 0;
    for (int i = 0; i < nshards; i++) {
        h_offsets[i] = acc;
        acc += (int64_t)h_hist[i];
    }
    h_offsets[nshards] = acc;
    // cursor initialised to shard starts
    HIP_CHECK(hipMemcpyAsync(hist, h_offsets, nshards * sizeof(int64_t),
                             hipMemcpyHostToDevice, s));
    if (n > 0)
        k_shard_scatter<<<grid_for(n), BLK, 0, s>>>(k, v, w, n, nshards, hist,
                                                    ok, ov, ow);
    HIP_CHECK(dbspk::cache_free(hist, s));
    return DBSP_OK;
}

// pair variant: both deltas' histograms land in one readback, so the
// exchange pays ONE partition sync instead of two
dbsp_status shard_rows_pair(hipStream_t s, const uint64_t *k0,
                            const uint64_t *v0, const int64_t *w0, int64_t n0,
                            const uint64_t *k1, const uint64_t *v1,
                            const int64_t *w1, int64_t n1, int nshards,
                            uint64_t *ok0, uint64_t *ov0, int64_t *ow0,
                            uint64_t *ok1, uint64_t *ov1, int64_t *ow1,
                            int64_t *h_off0, int64_t *h_off1) {
    uint64_t *hist;
    HIP_CHECK(dbspk::cache_malloc((void **)&hist,
                                  2 * (nshards + 1) * sizeof(uint64_t), s));
    HIP_CHECK(hipMemsetAsync(hist, 0, 2 * (nshards + 1) * sizeof(uint64_t), s));
    uint64_t *hist1 = hist + nshards + 1;
    if (n0 > 0) k_shard_hist<<<grid_for(n0), BLK, 0, s>>>(k0, n0, nshards, hist);
    if (n1 > 0)
        k_shard_hist<<<grid_for(n1), BLK, 0, s>>>(k1, n1, nshards, hist1);
    uint64_t h_hist[130];
    HIP_CHECK(hipMemcpyAsync(h_hist, hist, sizeof(h_hist[0]) * (nshards + 1),
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipMemcpyAsync(h_hist + 65, hist1,
                             sizeof(h_hist[0]) * (nshards + 1),
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    int64_t acc0 = 0, acc1 = 0;
    int64_t cur[130];
    for (int i = 0; i < nshards; i++) {
        h_off0[i] = acc0;
        h_off1[i] = acc1;
        cur[i] = acc0;
        cur[65 + i] = acc1;
        acc0 += (int64_t)h_hist[i];
        acc1 += (int64_t)h_hist[65 + i];
    }
    h_off0[nshards] = acc0;
    h_off1[nshards] = acc1;
    HIP_CHECK(hipMemcpyAsync(hist, cur, nshards * sizeof(int64_t),
                             hipMemcpyHostToDevice, s));
    HIP_CHECK(hipMemcpyAsync(hist1, cur + 65, nshards * sizeof(int64_t),
                             hipMemcpyHostToDevice, s));
    if (n0 > 0)
        k_shard_scatter<<<grid_for(n0), BLK, 0, s>>>(k0, v0, w0, n0, nshards,
                                                     hist, ok0, ov0, ow0);
    if (n1 > 0)
        k_shard_scatter<<<grid_for(n1), BLK, 0, s>>>(k1, v1, w1, n1, nshards,
                                                     hist1, ok1, ov1, ow1);
    HIP_CHECK(dbspk::cache_free(hist, s));
    return DBSP_OK;
}

// ---------------------------------------------------------------------------
// fixed-frame pair exchange (see kernels_iface.hpp FramePairArgs): ONE pack
// kernel + ONE equal-count ncclAllToAll + ONE unpack kernel replace the
// counts-allgather, its host sync, and the ~6*world-call grouped send/recv
// mesh of the dynamic path (r01_q3_shard_kernel_trace.txt attributed the 2x
// sharded-tick cost to exactly that host latency).
// ---------------------------------------------------------------------------

__global__ void k_frames_pack_pair(FramePairArgs a) {
    const int64_t S = 2 + 3 * a.P0 + 3 * a.P1;  // segment words
    const int64_t total = (int64_t)a.world * S;
    for (int64_t x = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         x < total; x += gridDim.x * (int64_t)blockDim.x) {
        const int r = (int)(x / S);
        const int64_t q = x % S;
        const int64_t n0 = a.off0[r + 1] - a.off0[r];
        const int64_t n1 = a.off1[r + 1] - a.off1[r];
        uint64_t val = 0;
        if (q == 0) {
            val = (uint64_t)n0;  // true count: > P0 flags sender overflow
        } else if (q == 1) {
            val = (uint64_t)n1;
        } else if (q - 2 < 3 * a.P0) {
            const int64_t p = q - 2;
            const int col = (int)(p / a.P0);
            const int64_t i = p % a.P0;
            if (i < n0 && i < a.P0) {
                const uint64_t *src = col == 0   ? a.p0k
                                      : col == 1 ? a.p0v
                                                 : (const uint64_t *)a.p0w;
                val = src[a.off0[r] + i];
            }
        } else {
            const int64_t p = q - 2 - 3 * a.P0;
            const int col = (int)(p / a.P1);
            const int64_t i = p % a.P1;
            if (i < n1 && i < a.P1) {
                const uint64_t *src = col == 0   ? a.p1k
                                      : col == 1 ? a.p1v
                                                 : (const uint64_t *)a.p1w;
                val = src[a.off1[r] + i];
            }
        }
        a.frame[x] = val;
    }
}

__global__ void k_frames_unpack_pair(const uint64_t *frame, int world,
                                     int64_t P0, int64_t P1, uint64_t *r0k,
                                     uint64_t *r0v, int64_t *r0w,
                                     uint64_t *r1k, uint64_t *r1v,
                                     int64_t *r1w, int64_t *d_tot0,
                                     int64_t *d_tot1) {
    const int64_t S = 2 + 3 * P0 + 3 * P1;
    // per-thread recompute of the <=8 headers (cached) + prefixes
    int64_t c0[8], c1[8], pre0[9], pre1[9];
    bool lost = false;
    pre0[0] = 0;
    pre1[0] = 0;
    for (int r = 0; r < world; r++) {
        c0[r] = (int64_t)frame[(int64_t)r * S];
        c1[r] = (int64_t)frame[(int64_t)r * S + 1];
        if (c0[r] > P0 || c1[r] > P1 || c0[r] < 0 || c1[r] < 0) lost = true;
        pre0[r + 1] = pre0[r] + (lost ? 0 : c0[r]);
        pre1[r + 1] = pre1[r] + (lost ? 0 : c1[r]);
    }
    if (lost) {
        // sender overflow somewhere: publish the sentinel; every rank sees
        // its own copy at the next sync and replays the dynamic exchange
        if (blockIdx.x == 0 && threadIdx.x == 0) {
            *d_tot0 = -1;
            *d_tot1 = -1;
        }
        return;
    }
    if (blockIdx.x == 0 && threadIdx.x == 0) {
        *d_tot0 = pre0[world];
        *d_tot1 = pre1[world];
    }
    const int64_t total = (int64_t)world * S;
    for (int64_t x = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         x < total; x += gridDim.x * (int64_t)blockDim.x) {
        const int r = (int)(x / S);
        const int64_t q = x % S;
        if (q < 2) continue;
        if (q - 2 < 3 * P0) {
            const int64_t p = q - 2;
            const int col = (int)(p / P0);
            const int64_t i = p % P0;
            if (i < c0[r]) {
                const uint64_t v = frame[x];
                if (col == 0) r0k[pre0[r] + i] = v;
                else if (col == 1) r0v[pre0[r] + i] = v;
                else r0w[pre0[r] + i] = (int64_t)v;
            }
        } else {
            const int64_t p = q - 2 - 3 * P0;
            const int col = (int)(p / P1);
            const int64_t i = p % P1;
            if (i < c1[r]) {
                const uint64_t v = frame[x];
                if (col == 0) r1k[pre1[r] + i] = v;
                else if (col == 1) r1v[pre1[r] + i] = v;
                else r1w[pre1[r] + i] = (int64_t)v;
            }
        }
    }
}

dbsp_status frames_pack_pair(hipStream_t s, const FramePairArgs &a) {
    const int64_t total = (int64_t)a.world * (2 + 3 * a.P0 + 3 * a.P1);
    k_frames_pack_pair<<<grid_for(total), BLK, 0, s>>>(a);
    return DBSP_OK;
}

// chained shard-to-frames: hash-partition BOTH raw streams straight into the
// frame segments with atomic header tickets — no host offsets, no separate
// partition/pack kernels, lengths read from the device (the chained tick's
// flatmap counters).  A ticket beyond capacity skips the write; the header
// keeps the true count, which the receiving unpack flags as the overflow
// sentinel.
__global__ void k_frames_init_headers(uint64_t *frame, int world, int64_t S) {
    const int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i < 2 * world) frame[(int64_t)(i / 2) * S + (i & 1)] = 0;
}

__global__ void k_shard_frames_chain(const uint64_t *k0, const uint64_t *v0,
                                     const int64_t *w0, const int64_t *n0_dev,
                                     const uint64_t *k1, const uint64_t *v1,
                                     const int64_t *w1, const int64_t *n1_dev,
                                     int world, int64_t P0, int64_t P1,
                                     uint64_t *frame) {
    const int64_t S = 2 + 3 * P0 + 3 * P1;
    const int64_t n0 = *n0_dev, n1 = *n1_dev;
    const int64_t total = n0 + n1;
    for (int64_t x = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         x < total; x += gridDim.x * (int64_t)blockDim.x) {
        const bool s1 = x >= n0;
        const int64_t i = s1 ? x - n0 : x;
        const uint64_t key = s1 ? k1[i] : k0[i];
        const int peer = (int)(dev_xxh3_u64(key, DBSP_HASH_SEED) % world);
        uint64_t *seg = frame + (int64_t)peer * S;
        const int64_t P = s1 ? P1 : P0;
        const int64_t t = (int64_t)atomicAdd(
            (unsigned long long *)(seg + (s1 ? 1 : 0)), 1ull);
        if (t < P) {
            uint64_t *base = seg + 2 + (s1 ? 3 * P0 : 0);
            base[t] = key;
            base[P + t] = s1 ? v1[i] : v0[i];
            base[2 * P + t] = (uint64_t)(s1 ? w1[i] : w0[i]);
        }
    }
}

dbsp_status shard_frames_chain(hipStream_t s, const uint64_t *k0,
                               const uint64_t *v0, const int64_t *w0,
                               const int64_t *n0_dev, const uint64_t *k1,
                               const uint64_t *v1, const int64_t *w1,
                               const int64_t *n1_dev, int world, int64_t P0,
                               int64_t P1, int64_t n_cap, uint64_t *frame) {
    const int64_t S = 2 + 3 * P0 + 3 * P1;
    k_frames_init_headers<<<dim3(1), dim3(64), 0, s>>>(frame, world, S);
    k_shard_frames_chain<<<grid_for(n_cap > 0 ? n_cap : 1), BLK, 0, s>>>(
        k0, v0, w0, n0_dev, k1, v1, w1, n1_dev, world, P0, P1, frame);
    return DBSP_OK;
}

dbsp_status frames_unpack_pair(hipStream_t s, const uint64_t *frame,
                               int world, int64_t P0, int64_t P1,
                               uint64_t *r0k, uint64_t *r0v, int64_t *r0w,
                               uint64_t *r1k, uint64_t *r1v, int64_t *r1w,
                               int64_t *d_tot0, int64_t *d_tot1) {
    const int64_t total = (int64_t)world * (2 + 3 * P0 + 3 * P1);
    k_frames_unpack_pair<<<grid_for(total), BLK, 0, s>>>(
        frame, world, P0, P1, r0k, r0v, r0w, r1k, r1v, r1w, d_tot0, d_tot1);
    return DBSP_OK;
}

dbsp_status columnize_events(hipStream_t s, const dbsp_event *ev, int64_t n,
                             uint64_t *kind, uint64_t *f0, uint64_t *f1,
                             uint64_t *f2, uint64_t *f3, uint64_t *f4,
                             int64_t *w) {
    if (n > 0)
        k_columnize<<<grid_for(n), BLK, 0, s>>>(ev, n, kind, f0, f1, f2, f3,
                                                f4, w);
    return DBSP_OK;
}

dbsp_status flatmap_events_chain(hipStream_t s, const dbsp_event *ev,
                                 const EventCols *cols, int64_t n, int query,
                                 uint64_t *k0, uint64_t *v0, int64_t *w0,
                                 uint64_t *k1, uint64_t *v1, int64_t *w1,
                                 uint64_t *ctr /* 2 device slots */) {
    HIP_CHECK(hipMemsetAsync(ctr, 0, 2 * sizeof(uint64_t), s));
    EventCols c{};
    if (cols) c = *cols;
    if (n > 0)
        k_flatmap<<<grid_for(n), BLK, 0, s>>>(ev, c, n, query, ctr, k0, v0,
                                              w0, ctr + 1, k1, v1, w1);
    return DBSP_OK;
}

dbsp_status flatmap_events(hipStream_t s, const dbsp_event *ev,
                           const EventCols *cols, int64_t n, int query, uint64_t *k0, uint64_t *v0, int64_t *w0,
                           int64_t *n0, uint64_t *k1, uint64_t *v1, int64_t *w1,
                           int64_t *n1) {
    EventCols c{};
    if (cols) c = *cols;
    uint64_t *ctr;
    HIP_CHECK(dbspk::cache_malloc((void **)&ctr, 2 * sizeof(uint64_t), s));
    HIP_CHECK(hipMemsetAsync(ctr, 0, 2 * sizeof(uint64_t), s));
    if (n > 0)
        k_flatmap<<<grid_for(n), BLK, 0, s>>>(ev, c, n, query, ctr, k0, v0, w0,
                                              ctr + 1, k1, v1, w1);
    uint64_t h_ctr[2];
    HIP_CHECK(hipMemcpyAsync(h_ctr, ctr, sizeof(h_ctr), hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    HIP_CHECK(dbspk::cache_free(ctr, s));
    *n0 = (int64_t)h_ctr[0];
    if (n1) *n1 = (int64_t)h_ctr[1];
    return DBSP_OK;
}

dbsp_status map_rows(hipStream_t s, const uint64_t *k, const uint64_t *v,
                     const int64_t *w, int64_t n, int mode, uint64_t *ok,
                     uint64_t *ov, int64_t *ow) {
    if (n > 0)
        k_map<<<grid_for(n), BLK, 0, s>>>(k, v, w, n, mode, ok, ov, ow);
    return DBSP_OK;
}

}  // namespace dbspk (reopened below after the extra kernels)

// ---- multi-batch linear aggregate (spine batches summed; linear op) ----

// f64 linear aggregate: per-key sums in batch-position order (deterministic)
__global__ void k_agg_sum_f64(const uint64_t *keys, int64_t nd,
                              const uint64_t *ik, const double *iw, int64_t ni,
                              double *acc) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nd;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint64_t key = keys[i];
        int64_t lo = lower_bound_k(ik, ni, key);
        int64_t hi = upper_bound_k(ik, ni, key);
        double sum = 0.0;
        for (int64_t t = lo; t < hi; t++) sum += iw[t];
        acc[i] += sum;
    }
}

// emit (key, bits(sum), +1) for sum != 0.0 (WeightedCount returns None for a
// zero f64 aggregate — aggregate/mod.rs:129-156 with R = F64)
__global__ void k_emit_nonzero_f64(const uint64_t *keys, const double *acc,
                                   int64_t nd, uint64_t *ctr, uint64_t *ok,
                                   uint64_t *ov, int64_t *ow) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nd;
         i += (int64_t)gridDim.x * blockDim.x) {
        if (acc[i] != 0.0) {
            uint64_t p = atomicAdd((unsigned long long *)ctr, 1ull);
            ok[p] = keys[i];
            double a = acc[i];
            ov[p] = *(const uint64_t *)&a;
            ow[p] = 1;
        }
    }
}

__global__ void k_agg_sum(const uint64_t *keys, int64_t nd, const uint64_t *ik,
                          const int64_t *iw, int64_t ni, int64_t *acc) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nd;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint64_t key = keys[i];
        int64_t lo = lower_bound_k(ik, ni, key);
        int64_t hi = upper_bound_k(ik, ni, key);
        int64_t s = 0;
        for (int64_t t = lo; t < hi; t++) s += iw[t];
        acc[i] += s;
    }
}

__global__ void k_emit_nonzero(const uint64_t *keys, const int64_t *acc,
                               int64_t nd, uint64_t *ctr, uint64_t *ok,
                               uint64_t *ov, int64_t *ow) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nd;
         i += (int64_t)gridDim.x * blockDim.x) {
        if (acc[i] != 0) {
            uint64_t p = atomicAdd((unsigned long long *)ctr, 1ull);
            ok[p] = keys[i];
            ov[p] = (uint64_t)acc[i];
            ow[p] = 1;
        }
    }
}

__global__ void k_key_head_flags(const uint64_t *k, int64_t n, uint64_t *flags) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        flags[i] = (i == 0) || k[i] != k[i - 1];
}

__global__ void k_compact_keys(const uint64_t *k, const uint64_t *flags,
                               const uint64_t *fscan, int64_t n, uint64_t *ok) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        if (flags[i]) ok[fscan[i]] = k[i];
}

namespace dbspk {

// consolidate SORTED rows with f64 weights: deterministic segmented tree
// (position-fixed order; tolerance 2 ulp * ceil(log2 run) vs sequential)
dbsp_status consolidate_sorted_f64(hipStream_t s, const uint64_t *kk,
                                   const uint64_t *vv, double *ww, int64_t n,
                                   uint64_t **ok, uint64_t **ov, double **ow,
                                   int64_t *out_n) {
    if (n == 0) {
        *ok = nullptr; *ov = nullptr; *ow = nullptr; *out_n = 0;
        return DBSP_OK;
    }
    uint64_t *flags, *fscan, *hp;
    HIP_CHECK(dbspk::cache_malloc((void **)&flags, n * sizeof(uint64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&fscan, n * sizeof(uint64_t), s));
    k_head_flags<<<grid_for(n), BLK, 0, s>>>(kk, vv, n, flags);
    uint64_t nseg = 0;
    dbsp_status st = scan_exclusive(s, flags, fscan, n, &nseg);
    if (st != DBSP_OK) return st;
    HIP_CHECK(dbspk::cache_malloc((void **)&hp, nseg * sizeof(uint64_t) + 8, s));
    k_seg_headpos<<<grid_for(n), BLK, 0, s>>>(flags, fscan, n, hp);
    for (int64_t d = 1; d < n; d <<= 1)
        k_seg_tree_round<<<grid_for(n), BLK, 0, s>>>(ww, flags, fscan, hp, n, d);
    uint64_t *sk, *sv;
    double *sw;
    HIP_CHECK(dbspk::cache_malloc((void **)&sk, nseg * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&sv, nseg * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&sw, nseg * sizeof(double) + 8, s));
    k_seg_collect_f64<<<grid_for(n), BLK, 0, s>>>(kk, vv, ww, flags, fscan, n,
                                                  sk, sv, sw);
    uint64_t *nzflags, *nzscan;
    HIP_CHECK(dbspk::cache_malloc((void **)&nzflags, nseg * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&nzscan, nseg * sizeof(uint64_t) + 8, s));
    k_nonzero_flags_f64<<<grid_for(nseg), BLK, 0, s>>>(sw, nseg, nzflags);
    uint64_t nout = 0;
    st = scan_exclusive(s, nzflags, nzscan, nseg, &nout);
    if (st != DBSP_OK) return st;
    uint64_t *rk, *rv;
    double *rw;
    HIP_CHECK(dbspk::cache_malloc((void **)&rk, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rv, nout * sizeof(uint64_t) + 8, s));
    HIP_CHECK(dbspk::cache_malloc((void **)&rw, nout * sizeof(double) + 8, s));
    // k_compact copies the weight column bitwise (8 B) — valid for f64
    k_compact<<<grid_for(nseg), BLK, 0, s>>>(sk, sv, (const int64_t *)sw,
                                             nzflags, nzscan, nseg, rk, rv,
                                             (int64_t *)rw);
    HIP_CHECK(dbspk::cache_free(flags, s));
    HIP_CHECK(dbspk::cache_free(fscan, s));
    HIP_CHECK(dbspk::cache_free(hp, s));
    HIP_CHECK(dbspk::cache_free(sk, s));
    HIP_CHECK(dbspk::cache_free(sv, s));
    HIP_CHECK(dbspk::cache_free(sw, s));
    HIP_CHECK(dbspk::cache_free(nzflags, s));
    HIP_CHECK(dbspk::cache_free(nzscan, s));
    *ok = rk; *ov = rv; *ow = rw; *out_n = (int64_t)nout;
    return DBSP_OK;
}

dbsp_status agg_sum_batch_f64(hipStream_t s, const uint64_t *keys, int64_t nd,
                              const uint64_t *ik, const double *iw, int64_t ni,
                              double *acc) {
    if (nd > 0 && ni > 0)
        k_agg_sum_f64<<<grid_for(nd), BLK, 0, s>>>(keys, nd, ik, iw, ni, acc);
    return DBSP_OK;
}

dbsp_status emit_nonzero_f64(hipStream_t s, const uint64_t *keys,
                             const double *acc, int64_t nd, uint64_t *ok,
                             uint64_t *ov, int64_t *ow, int64_t *h_count) {
    uint64_t *ctr;
    HIP_CHECK(dbspk::cache_malloc((void **)&ctr, sizeof(uint64_t), s));
    HIP_CHECK(hipMemsetAsync(ctr, 0, sizeof(uint64_t), s));
    if (nd > 0)
        k_emit_nonzero_f64<<<grid_for(nd), BLK, 0, s>>>(keys, acc, nd, ctr, ok,
                                                        ov, ow);
    uint64_t h = 0;
    HIP_CHECK(hipMemcpyAsync(&h, ctr, sizeof(uint64_t), hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    HIP_CHECK(dbspk::cache_free(ctr, s));
    *h_count = (int64_t)h;
    return DBSP_OK;
}

dbsp_status agg_sum_batch(hipStream_t s, const uint64_t *keys, int64_t nd,
                          const uint64_t *ik, const int64_t *iw, int64_t ni,
                          int64_t *acc) {
    if (nd > 0 && ni > 0)
        k_agg_sum<<<grid_for(nd), BLK, 0, s>>>(keys, nd, ik, iw, ni, acc);
    return DBSP_OK;
}

dbsp_status emit_nonzero(hipStream_t s, const uint64_t *keys,
                         const int64_t *acc, int64_t nd, uint64_t *ok,
                         uint64_t *ov, int64_t *ow, int64_t *h_count) {
    uint64_t *ctr;
    HIP_CHECK(dbspk::cache_malloc((void **)&ctr, sizeof(uint64_t), s));
    HIP_CHECK(hipMemsetAsync(ctr, 0, sizeof(uint64_t), s));
    if (nd > 0)
        k_emit_nonzero<<<grid_for(nd), BLK, 0, s>>>(keys, acc, nd, ctr, ok, ov, ow);
    uint64_t h = 0;
    HIP_CHECK(hipMemcpyAsync(&h, ctr, sizeof(uint64_t), hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    HIP_CHECK(dbspk::cache_free(ctr, s));
    *h_count = (int64_t)h;
    return DBSP_OK;
}

dbsp_status unique_keys(hipStream_t s, const uint64_t *kk, int64_t n,
                        uint64_t **okeys, int64_t *out_n) {
    if (n == 0) {
        *okeys = nullptr;
        *out_n = 0;
        return DBSP_OK;
    }
    uint64_t *flags, *fscan;
    HIP_CHECK(dbspk::cache_malloc((void **)&flags, n * sizeof(uint64_t), s));
    HIP_CHECK(dbspk::cache_malloc((void **)&fscan, n * sizeof(uint64_t), s));
    k_key_head_flags<<<grid_for(n), BLK, 0, s>>>(kk, n, flags);
    uint64_t nk = 0;
    dbsp_status st = scan_exclusive(s, flags, fscan, n, &nk);
    if (st != DBSP_OK) return st;
    uint64_t *out;
    HIP_CHECK(dbspk::cache_malloc((void **)&out, nk * sizeof(uint64_t) + 8, s));
    k_compact_keys<<<grid_for(n), BLK, 0, s>>>(kk, flags, fscan, n, out);
    HIP_CHECK(dbspk::cache_free(flags, s));
    HIP_CHECK(dbspk::cache_free(fscan, s));
    *okeys = out;
    *out_n = (int64_t)nk;
    return DBSP_OK;
}


// ---------------------------------------------------------------------------
// Radix-tree rolling aggregate (operator/time_series/radix_tree/mod.rs:1-75,
// rolling_aggregate.rs:235-280, range.rs:76-110 — SURVEY.md §8f4).  The
// reference covers the timeline with an adaptive radix tree of per-prefix
// aggregates, stored as indexed Z-sets, so a range aggregate visits O(log n)
// nodes.  The MI355X-native restatement: the batch is already (partition,
// time)-sorted, so the same O(log) range queries come from a FLAT radix-16
// prefix-aggregate tree over the row array (levels of 16:1 weight-sum
// reductions — "prefix-sum trees map well to GPU scans") and
// a row's time range maps to a row range by binary search.  One query
// thread per input row computes the rolling aggregate
// range_of(ts) = [ts - width, ts] (RelRange::range_of, range.rs:93-110,
// saturating at 0) within its partition — the per-row output of
// partitioned_rolling_aggregate for the linear (weight-sum) aggregate.
// ---------------------------------------------------------------------------

#define RT_RADIX 16
#define RT_MAX_LEVELS 16

struct RtLevels {
    int nl;
    const int64_t *lv[RT_MAX_LEVELS];
    int64_t n[RT_MAX_LEVELS];
};

__global__ void k_rt_reduce(const int64_t *in, int64_t n, int64_t *out,
                            int64_t n_out) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_out;
         i += (int64_t)gridDim.x * blockDim.x) {
        int64_t s = 0;
        const int64_t lo = i * RT_RADIX;
        const int64_t hi = lo + RT_RADIX < n ? lo + RT_RADIX : n;
        for (int64_t j = lo; j < hi; j++) s += in[j];
        out[i] = s;
    }
}

// range weight-sum over [a, b) using the tree: peel unaligned edges at each
// level (<= 15 adds per level per side), then ascend — O(log16 n) work
__device__ inline int64_t rt_range_sum(const int64_t *w, int64_t nw,
                                       const RtLevels &t, int64_t a,
                                       int64_t b) {
    int64_t sum = 0;
    const int64_t *cur = w;
    int lvl = -1;
    while (a < b) {
        while ((a % RT_RADIX) != 0 && a < b) sum += cur[a++];
        while ((b % RT_RADIX) != 0 && a < b) sum += cur[--b];
        if (a >= b) break;
        a /= RT_RADIX;
        b /= RT_RADIX;
        lvl++;
        cur = t.lv[lvl];
    }
    return sum;
}

__global__ void k_rolling_agg(const uint64_t *k, const uint64_t *v,
                              const int64_t *w, int64_t n, RtLevels t,
                              uint64_t width, uint64_t *ok, uint64_t *ov,
                              int64_t *ow) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        const uint64_t p = k[i], ts = v[i];
        // partition row range
        const int64_t plo = lower_bound_k(k, n, p);
        int64_t phi = plo;
        {
            int64_t step = 1;
            while (phi + step < n && k[phi + step] == p) { phi += step; step <<= 1; }
            phi = upper_bound_k(k + phi, min(step, n - phi), p) + phi;
        }
        // time range [ts - width, ts] inclusive (range.rs:93-110, saturating)
        const uint64_t t0 = ts >= width ? ts - width : 0;
        int64_t qlo = plo, qhi = plo;
        {   // lower_bound of t0 and upper_bound of ts within [plo, phi)
            int64_t lo = plo, hi = phi;
            while (lo < hi) {
                int64_t mid = (lo + hi) / 2;
                if (v[mid] < t0) lo = mid + 1; else hi = mid;
            }
            qlo = lo;
            hi = phi;
            while (lo < hi) {
                int64_t mid = (lo + hi) / 2;
                if (v[mid] <= ts) lo = mid + 1; else hi = mid;
            }
            qhi = lo;
        }
        ok[i] = p;
        ov[i] = ts;
        ow[i] = rt_range_sum(w, n, t, qlo, qhi);
    }
}

dbsp_status rolling_agg_rows(hipStream_t s, const uint64_t *k,
                             const uint64_t *v, const int64_t *w, int64_t n,
                             uint64_t width, uint64_t *ok, uint64_t *ov,
                             int64_t *ow) {
    if (n <= 0) return DBSP_OK;
    RtLevels t{};
    int64_t cur_n = n;
    const int64_t *cur = w;
    int64_t *bufs[RT_MAX_LEVELS] = {};
    while (cur_n > 1 && t.nl < RT_MAX_LEVELS) {
        const int64_t nn = (cur_n + RT_RADIX - 1) / RT_RADIX;
        HIP_CHECK(dbspk::cache_malloc((void **)&bufs[t.nl],
                                      nn * sizeof(int64_t) + 8, s));
        k_rt_reduce<<<grid_for(nn), BLK, 0, s>>>(cur, cur_n, bufs[t.nl], nn);
        t.lv[t.nl] = bufs[t.nl];
        t.n[t.nl] = nn;
        cur = bufs[t.nl];
        cur_n = nn;
        t.nl++;
    }
    k_rolling_agg<<<grid_for(n), BLK, 0, s>>>(k, v, w, n, t, width, ok, ov,
                                              ow);
    for (int i = 0; i < t.nl; i++)
        HIP_CHECK(dbspk::cache_free(bufs[i], s));
    return DBSP_OK;
}

uint64_t host_xxh3_u64(uint64_t key, uint64_t seed) {
    // host copy of dev_xxh3_u64 (kept in sync; parity-tested against the oracle)
    const uint64_t SEC8_16 = 0x1cad21f72c81017cull ^ 0xdb979083e96dd4deull;
    uint64_t s = seed ^ ((uint64_t)__builtin_bswap32((uint32_t)seed) << 32);
    uint64_t input64 = (uint64_t)(uint32_t)(key >> 32) + ((uint64_t)(uint32_t)key << 32);
    uint64_t h = input64 ^ (SEC8_16 - s);
    uint64_t r49 = (h << 49) | (h >> 15);
    uint64_t r24 = (h << 24) | (h >> 40);
    h ^= r49 ^ r24;
    h *= 0x9FB21C651E98DF25ull;
    h ^= (h >> 35) + 8;
    h *= 0x9FB21C651E98DF25ull;
    h ^= h >> 28;
    return h;
}

}  // namespace dbspk
