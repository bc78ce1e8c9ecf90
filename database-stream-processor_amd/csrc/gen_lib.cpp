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

// Reference-test replay entry (generator/mod.rs:147-159 make_test_generator):
// a fresh generator with the reference tests' StepRng(0,1) and base_time 0,
// calling next_person / next_auction / next_bid directly so the reference's
// own generator unit tests (people.rs/auctions.rs/bids.rs) can be asserted
// verbatim on the emitted fields.  kind: 0 person(event_id, ts),
// 1 auction(events_count=event_id? no: events_count, event_id, ts uses
// events_count = 0 as the tests do), 2 bid(event_id, ts).
int64_t dbsp_gen_unit(int kind, uint64_t event_id, uint64_t ts,
                      dbsp_event *out) {
    nexgen::Generator g(0, 0, nexgen::DEFAULT_RATE);
    g.base_time = 0;
    g.rng.s = 0;
    g.rng.step = true;
    if (kind == 0)
        *out = g.next_person(event_id, ts);
    else if (kind == 1)
        *out = g.next_auction(0, event_id, ts);
    else
        *out = g.next_bid(event_id, ts);
    return 0;
}

}  // extern "C"
