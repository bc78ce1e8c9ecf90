// nexmark_gen.hpp — deterministic Nexmark event generator (host side).
//
// Mirrors the structure of the reference generator
// (crates/nexmark/src/generator/{mod,auctions,people,bids,config}.rs and
// crates/nexmark/src/config.rs), which is itself a port of the Java Nexmark
// generator.  The DETERMINISTIC parts — event-kind schedule, person/auction id
// arithmetic, timestamps — follow the reference exactly:
//   - event mix person:auction:bid = 1:3:46 (config.rs:128-143)
//   - last/next person and auction id arithmetic
//     (generator/people.rs:93-118, generator/auctions.rs:85-123)
//   - timestamp_for_event = base + n * inter_event_delay_us/1000
//     (generator/config.rs:118-120; default first_event_rate = 10M events/s,
//      config.rs:50)
// The RANDOM draws (hot-seller/bidder/auction choice, category, state, city,
// name, price) come from a documented splitmix64 stream seeded per run:
// event-stream parity with the reference's Rust SmallRng is unpinned
// (SURVEY.md §8c), so the oracle and the GPU engine are fed IDENTICAL streams
// from this generator and parity is asserted between them (plus against the
// reference's own golden query vectors for hand-written inputs).
//
// Strings are dictionary ids end-to-end: state ids index the reference's
// US_STATES table (generator/people.rs:18-25: AZ,CA,ID,OR,WA,WY = 0..5), city
// ids its US_CITIES table (people.rs:27-38, Phoenix=0), name ids are opaque
// u32s standing for the generated first+last name combination.
#pragma once
#include <cmath>
#include <cstdint>
#include "../../include/dbsp_hip.h"

namespace nexgen {

// config.rs:119-143 defaults
constexpr uint64_t PERSON_PROP = 1, AUCTION_PROP = 3, BID_PROP = 46;
constexpr uint64_t TOTAL_PROP = PERSON_PROP + AUCTION_PROP + BID_PROP;  // 50
constexpr uint64_t FIRST_PERSON_ID = 1000;    // generator/config.rs:5
constexpr uint64_t FIRST_AUCTION_ID = 1000;   // generator/config.rs:6
constexpr uint64_t FIRST_CATEGORY_ID = 10;    // generator/config.rs:7
constexpr uint64_t NUM_CATEGORIES = 5;        // generator/auctions.rs:19
constexpr uint64_t HOT_SELLER_RATIO_CONST = 100;   // auctions.rs:23
constexpr uint64_t HOT_AUCTION_RATIO_CONST = 100;  // bids.rs:17
constexpr uint64_t HOT_BIDDER_RATIO_CONST = 100;   // bids.rs:18
constexpr uint64_t CFG_HOT_SELLERS_RATIO = 4;   // config.rs:137
constexpr uint64_t CFG_HOT_AUCTION_RATIO = 2;   // config.rs:135
constexpr uint64_t CFG_HOT_BIDDERS_RATIO = 4;   // config.rs:136
constexpr uint64_t NUM_ACTIVE_PEOPLE = 1000;    // config.rs:139
constexpr uint64_t PERSON_ID_LEAD = 10;         // config.rs:11
constexpr uint64_t NUM_IN_FLIGHT_AUCTIONS = 100;  // config.rs:141
constexpr uint64_t NUM_US_STATES = 6, NUM_US_CITIES = 10;
constexpr double DEFAULT_RATE = 10'000'000.0;   // events/s, config.rs:50

struct Rng {
    // Raw stream: splitmix64 (documented; stands in for the reference bench's
    // UNSEEDED ThreadRng — lib.rs:198 `create_generators_for_config::<ThreadRng>`
    // — so no reproducible reference event stream exists to pin against).
    // step=true switches to the reference generator tests' StepRng(0,1)
    // (rand::rngs::mock::StepRng: raw values 0,1,2,...) so those tests'
    // expected events can be asserted verbatim.
    uint64_t s;
    bool step = false;
    uint64_t next() {
        if (step) return s++;
        uint64_t z = (s += 0x9E3779B97F4A7C15ull);
        z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
        z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
        return z ^ (z >> 31);
    }
    // rand 0.8.5 UniformInt::sample_single (uniform_int_impl!, the
    // widening-multiply + zone-rejection method) — the exact arithmetic the
    // reference's rng.gen_range(0..n) performs on the raw draw
    uint64_t range(uint64_t n) {
        if (n == 0) return 0;
        const uint64_t zone = (n << __builtin_clzll(n)) - 1;
        for (;;) {
            const uint64_t v = next();
            const unsigned __int128 m = (unsigned __int128)v * n;
            if ((uint64_t)m <= zone) return (uint64_t)(m >> 64);
        }
    }
    // rand's f32 range sample on [0,1): 24 high mantissa bits of next_u32
    // (xoshiro-style next_u32 = upper half of next_u64)
    float f01() {
        return (float)((uint32_t)(next() >> 32) >> 8) * (1.0f / 16777216.0f);
    }
};

struct Generator {
    uint64_t seed;
    uint64_t base_time;     // ms; reference benches use wallclock; we fix it
    double inter_delay_us;  // generator/config.rs:62
    uint64_t count = 0;     // events generated so far
    Rng rng;

    explicit Generator(uint64_t seed_ = 1, uint64_t base_time_ms = 10'000'000,
                       double rate = DEFAULT_RATE)
        : seed(seed_), base_time(base_time_ms),
          inter_delay_us(1'000'000.0 / rate), rng{seed_ * 0x9E3779B97F4A7C15ull + 1} {}

    // generator/config.rs:118-120
    uint64_t timestamp_for(uint64_t event_number) const {
        return base_time + (uint64_t)(inter_delay_us * (double)event_number) / 1000;
    }

    // generator/people.rs:105-118
    static uint64_t last_base0_person_id(uint64_t event_id) {
        uint64_t epoch = event_id / TOTAL_PROP;
        uint64_t offset = event_id % TOTAL_PROP;
        if (offset >= PERSON_PROP) offset = PERSON_PROP - 1;
        return epoch * PERSON_PROP + offset;
    }

    // generator/people.rs:93-103
    uint64_t next_base0_person_id(uint64_t event_id) {
        uint64_t num_people = last_base0_person_id(event_id) + 1;
        uint64_t active = num_people < NUM_ACTIVE_PEOPLE ? num_people : NUM_ACTIVE_PEOPLE;
        uint64_t n = rng.range(active + PERSON_ID_LEAD);
        return num_people - active + n;
    }

    // generator/auctions.rs:85-110
    static uint64_t last_base0_auction_id(uint64_t event_id) {
        uint64_t epoch = event_id / TOTAL_PROP;
        uint64_t offset = event_id % TOTAL_PROP;
        if (offset < PERSON_PROP) {
            if (epoch == 0) return 0;
            epoch -= 1;
            offset = AUCTION_PROP - 1;
        } else if (offset >= PERSON_PROP + AUCTION_PROP) {
            offset = AUCTION_PROP - 1;
        } else {
            offset -= PERSON_PROP;
        }
        return epoch * AUCTION_PROP + offset;
    }

    // generator/auctions.rs:112-123
    uint64_t next_base0_auction_id(uint64_t event_id) {
        uint64_t last = last_base0_auction_id(event_id);
        uint64_t min_a = last > NUM_IN_FLIGHT_AUCTIONS ? last - NUM_IN_FLIGHT_AUCTIONS : 0;
        return min_a + rng.range(last - min_a + 1);
    }

    // generator/price.rs:9-11: ceil(10^(U(0,1)*6) * 100) — log-uniform in
    // [100, 10^8]
    uint64_t next_price() {
        return (uint64_t)std::ceil(std::pow(10.0f, rng.f01() * 6.0f) * 100.0f);
    }

    // generator/auctions.rs:125-148: auction length with the in-flight
    // horizon (events for NUM_IN_FLIGHT_AUCTIONS more auctions)
    uint64_t next_auction_length_ms(uint64_t events_count, uint64_t ts) {
        const uint64_t num_events_for_auctions =
            NUM_IN_FLIGHT_AUCTIONS * TOTAL_PROP / AUCTION_PROP;  // 1666
        const uint64_t future = timestamp_for(events_count + num_events_for_auctions);
        const uint64_t horizon = future > ts ? future - ts : 0;
        const uint64_t cap = horizon * 2 > 1 ? horizon * 2 : 1;
        return 1 + rng.range(cap);
    }

    // One event.  kind/f* layout per include/dbsp_hip.h.
    dbsp_event next() {
        uint64_t event_id = count;  // single generator: first_event_id = 0
        uint64_t ts = timestamp_for(event_id);
        uint64_t rem = event_id % TOTAL_PROP;  // generator/mod.rs:78-88
        dbsp_event e{};
        e.w = 1;
        if (rem < PERSON_PROP) {
            e = next_person(event_id, ts);
        } else if (rem < PERSON_PROP + AUCTION_PROP) {
            e = next_auction(count, event_id, ts);
        } else {
            e = next_bid(event_id, ts);
        }
        count++;
        return e;
    }

    // generator/people.rs:50-79.  Draw order follows the reference for the
    // fields carried here (name, city, state); the email / credit-card /
    // extra string draws are not emitted and are skipped — with the test
    // StepRng their gen_range results are all 0, so the reference's own
    // generator tests remain assertable verbatim.
    dbsp_event next_person(uint64_t event_id, uint64_t ts) {
        dbsp_event e{};
        e.w = 1;
        e.kind = 0;
        e.f0 = last_base0_person_id(event_id) + FIRST_PERSON_ID;
        const uint64_t first = rng.range(11);  // FIRST_NAMES (people.rs:40-42)
        const uint64_t last = rng.range(10);   // LAST_NAMES (people.rs:44-46)
        e.f1 = first * 16 + last;              // name id = (first, last) pair
        e.f2 = rng.range(NUM_US_CITIES);       // city id (Phoenix = 0)
        e.f3 = rng.range(NUM_US_STATES);       // state id (AZ = 0)
        e.f4 = ts;
        return e;
    }

    // generator/auctions.rs:26-82 (draw order: seller, category, initial_bid
    // — drawn, not emitted — then the length horizon draw)
    dbsp_event next_auction(uint64_t events_count, uint64_t event_id,
                            uint64_t ts) {
        dbsp_event e{};
        e.w = 1;
        e.kind = 1;
        e.f0 = last_base0_auction_id(event_id) + FIRST_AUCTION_ID;
        uint64_t seller;
        if (rng.range(CFG_HOT_SELLERS_RATIO) == 0)
            seller = next_base0_person_id(event_id);
        else
            seller = (last_base0_person_id(event_id) / HOT_SELLER_RATIO_CONST) *
                     HOT_SELLER_RATIO_CONST;
        e.f1 = seller + FIRST_PERSON_ID;
        e.f2 = FIRST_CATEGORY_ID + rng.range(NUM_CATEGORIES);
        (void)next_price();  // initial_bid: drawn by the reference, not emitted
        e.f3 = ts;
        e.f4 = ts + next_auction_length_ms(events_count, ts);  // expires
        return e;
    }

    // generator/bids.rs:60-100 (channel/url draws not emitted, skipped)
    dbsp_event next_bid(uint64_t event_id, uint64_t ts) {
        dbsp_event e{};
        e.w = 1;
        e.kind = 2;
        uint64_t auction;
        if (rng.range(CFG_HOT_AUCTION_RATIO) == 0)
            auction = next_base0_auction_id(event_id);
        else
            auction = (last_base0_auction_id(event_id) / HOT_AUCTION_RATIO_CONST) *
                      HOT_AUCTION_RATIO_CONST;
        e.f0 = auction + FIRST_AUCTION_ID;
        uint64_t bidder;
        if (rng.range(CFG_HOT_BIDDERS_RATIO) == 0)
            bidder = next_base0_person_id(event_id);
        else
            bidder = (last_base0_person_id(event_id) / HOT_BIDDER_RATIO_CONST) *
                         HOT_BIDDER_RATIO_CONST + 1;
        e.f1 = bidder + FIRST_PERSON_ID;
        e.f2 = next_price();
        e.f3 = ts;
        e.f4 = 0;
        return e;
    }
};

}  // namespace nexgen
