// gen_lib.cpp — host-only shared library exposing the deterministic Nexmark
// generator (nexmark_gen.hpp) to Python (tests, bench pre-staging).  CPU-only:
// no HIP dependency, loadable in the GPU-less build container.
#include "nexmark_gen.hpp"

extern "C" {

// Opaque generator handle for chunked generation of long streams.
void *dbsp_gen_new(uint64_t seed, uint64_t base_time_ms, double rate) {
    return new nexgen::Generator(seed, base_time_ms, rate);
}

void dbsp_gen_free(void *h) { delete (nexgen::Generator *)h; }

int64_t dbsp_gen_next(void *h, dbsp_event *out, int64_t n) {
    auto *g = (nexgen::Generator *)h;
    for (int64_t i = 0; i < n; i++) out[i] = g->next();
    return n;
}

// One-shot convenience.
int64_t dbsp_gen_events(uint64_t seed, uint64_t base_time_ms, double rate,
                        int64_t n, dbsp_event *out) {
    nexgen::Generator g(seed, base_time_ms, rate);
    for (int64_t i = 0; i < n; i++) out[i] = g.next();
    return n;
}

}  // extern "C"
