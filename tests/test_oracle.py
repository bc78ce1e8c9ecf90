"""CPU oracle vs the reference's own golden vectors (tests/golden/) plus
randomized model tests following the reference's TestBatch pattern
(reference trace/test_batch.rs:1-60)."""
import numpy as np
import pytest

from dbsp_amd import ROW_DT
from dbsp_amd import oracle
from helpers import (CITY_IDS, Intern, auction_event, bid_event, events,
                     load_golden, np_consolidate, pack_person, person_event,
                     rows_of, state_id, zset)

# ---------------------------------------------------------------------------
# golden: join_test (reference operator/join.rs:886-1017)
# ---------------------------------------------------------------------------

PROJ_HI_K_LO_V1V2 = 2
PROJ_HI_K_LO_V2V1 = 7


def _tick_rows(tick, intern):
    return rows_of([(k, intern(s), w) for k, s, w in tick])


def test_golden_stream_join():
    g = load_golden("join_test.json")
    intern = Intern()
    for t1, t2, exp in zip(g["input1"], g["input2"], g["stream_join"]):
        a = oracle.consolidate(_tick_rows(t1, intern))
        b = oracle.consolidate(_tick_rows(t2, intern))
        out = oracle.consolidate(oracle.join_raw(a, b, PROJ_HI_K_LO_V1V2))
        expected = zset(rows_of([
            (k, (intern(s1) << 32) | intern(s2), w) for k, s1, s2, w in exp]))
        assert zset(out) == expected


def test_golden_incremental_join():
    """Incremental join decomposition dA ⋈ B_prev + A_cur ⋈ dB vs the
    reference's expected inc_outputs (join.rs:938-962)."""
    g = load_golden("join_test.json")
    intern = Intern()
    a_int = np.empty(0, dtype=ROW_DT)
    b_int = np.empty(0, dtype=ROW_DT)
    for t1, t2, exp in zip(g["input1"], g["input2"], g["incremental"]):
        da = oracle.consolidate(_tick_rows(t1, intern))
        db = oracle.consolidate(_tick_rows(t2, intern))
        out1 = oracle.join_raw(da, b_int, PROJ_HI_K_LO_V1V2)
        a_int = oracle.merge(a_int, da)
        out2 = oracle.join_raw(db, a_int, PROJ_HI_K_LO_V2V1)
        b_int = oracle.merge(b_int, db)
        out = oracle.consolidate(np.concatenate([out1, out2]))
        expected = zset(rows_of([
            (k, (intern(s1) << 32) | intern(s2), w) for k, s1, s2, w in exp]))
        assert zset(out) == expected


# ---------------------------------------------------------------------------
# golden: q3 (reference nexmark queries/q3.rs tests)
# ---------------------------------------------------------------------------

def test_golden_q3():
    g = load_golden("q3_people.json")
    intern = Intern()
    q = oracle.Query(3)
    for tick in g["ticks"]:
        evs = []
        for p in tick["persons"]:
            evs.append(person_event(p["id"], intern(p["name"]),
                                    CITY_IDS[p["city"]], state_id(p["state"], intern)))
        for a in tick["auctions"]:
            evs.append(auction_event(a["id"], a["seller"], a["category"]))
        out = q.step(events(*evs))
        expected = zset(rows_of([
            (pack_person(intern(name), CITY_IDS[city], state_id(st, intern)), aid, w)
            for name, city, st, aid, w in tick["expected"]]))
        assert zset(out) == expected


# ---------------------------------------------------------------------------
# golden: q5 (reference nexmark queries/q5.rs tests)
# ---------------------------------------------------------------------------

def test_golden_q5():
    g = load_golden("q5_hot_items.json")
    for case in g["cases"]:
        q = oracle.Query(5)
        for b1, b2, exp in zip(case["auction1_batches"], case["auction2_batches"],
                               case["expected"]):
            evs = [bid_event(1, dt) for dt in b1] + [bid_event(2, dt) for dt in b2]
            out = q.step(events(*evs))
            expected = zset(rows_of([(a, n, w) for a, n, w in exp]))
            assert zset(out) == expected, case["name"]


# ---------------------------------------------------------------------------
# golden: q8 (reference nexmark queries/q8.rs tests)
# ---------------------------------------------------------------------------

def test_golden_q8():
    g = load_golden("q8_monitor_new_users.json")
    for case in g["cases"]:
        intern = Intern()
        q = oracle.Query(8)
        for pb, ab, exp in zip(case["people_batches"], case["auction_batches"],
                               case["expected"]):
            evs = [person_event(pid, intern(name), 0, 3, dt=dt)
                   for pid, name, dt in pb]
            evs += [auction_event(1, seller, 1, dt=dt) for seller, dt in ab]
            out = q.step(events(*evs))
            expected = zset(rows_of([
                (pid, (intern(name) << 32) | stime, w)
                for pid, name, stime, w in exp]))
            assert zset(out) == expected, case["name"]


# ---------------------------------------------------------------------------
# model tests (reference TestBatch pattern: random inputs vs naive model)
# ---------------------------------------------------------------------------

def _random_rows(rng, n, key_range=50, val_range=8, w_range=3):
    out = np.empty(n, dtype=ROW_DT)
    out["k"] = rng.integers(0, key_range, n)
    out["v"] = rng.integers(0, val_range, n)
    out["w"] = rng.integers(-w_range, w_range + 1, n)
    return out


def test_consolidate_model():
    rng = np.random.default_rng(7)
    for n in [0, 1, 2, 17, 1000, 20000]:
        r = _random_rows(rng, n)
        got = oracle.consolidate(r)
        exp = np_consolidate(r)
        assert np.array_equal(got, exp)
        # output invariants: sorted by (k,v), unique, no zero weights
        if len(got):
            kv = np.stack([got["k"], got["v"]], axis=1)
            assert (np.lexsort((got["v"], got["k"])) == np.arange(len(got))).all()
            assert (kv[1:] != kv[:-1]).any(axis=1).all()
            assert (got["w"] != 0).all()


def test_merge_model():
    rng = np.random.default_rng(8)
    for n in [0, 1, 5, 300, 5000]:
        a = oracle.consolidate(_random_rows(rng, n))
        b = oracle.consolidate(_random_rows(rng, n))
        got = oracle.merge(a, b)
        exp = np_consolidate(np.concatenate([a, b]))
        assert np.array_equal(got, exp)
        # adversarial: disjoint, identical, cancelling
        c = a.copy()
        c["w"] = -c["w"]
        assert len(oracle.merge(a, c)) == 0


def test_join_model():
    rng = np.random.default_rng(9)
    for n in [0, 3, 100, 2000]:
        d = oracle.consolidate(_random_rows(rng, n, key_range=20))
        t = oracle.consolidate(_random_rows(rng, n, key_range=20))
        got = zset(oracle.join_raw(d, t, PROJ_HI_K_LO_V1V2))
        exp = {}
        for rd in d:
            for rt in t:
                if rd["k"] == rt["k"]:
                    key = (int(rd["k"]), (int(rd["v"]) << 32) | int(rt["v"]))
                    exp[key] = exp.get(key, 0) + int(rd["w"]) * int(rt["w"])
        exp = {k: w for k, w in exp.items() if w != 0}
        assert got == exp


def test_agg_linear_model():
    rng = np.random.default_rng(10)
    in_trace = oracle.consolidate(_random_rows(rng, 500, key_range=30, val_range=1))
    out_trace = oracle.consolidate(_random_rows(rng, 60, key_range=30, val_range=4,
                                                w_range=1))
    keys = np.unique(in_trace["k"])[:10]
    got = oracle.agg_linear_upsert(keys, in_trace, out_trace)
    exp = []
    for k in keys:
        upd = {}
        s = int(in_trace["w"][in_trace["k"] == k].sum())
        if s != 0:
            upd[(int(k), s)] = upd.get((int(k), s), 0) + 1
        for r in out_trace[out_trace["k"] == k]:
            if r["w"] != 0:
                kk = (int(k), int(r["v"]))
                upd[kk] = upd.get(kk, 0) - int(r["w"])
        exp.extend((k_, v_, w_) for (k_, v_), w_ in upd.items() if w_ != 0)
    assert zset(got) == zset(rows_of(exp))


def test_window_model():
    rng = np.random.default_rng(11)
    trace = oracle.consolidate(_random_rows(rng, 400, key_range=100, w_range=2))
    batch = oracle.consolidate(_random_rows(rng, 50, key_range=100, w_range=2))
    s0, e0, s1, e1 = 10, 50, 30, 80
    got = zset(oracle.window(trace, batch, True, s0, e0, s1, e1))
    exp = {}

    def add(r, sign):
        key = (int(r["k"]), int(r["v"]))
        exp[key] = exp.get(key, 0) + sign * int(r["w"])

    for r in trace:
        k = int(r["k"])
        if s0 <= k < min(s1, e0):
            add(r, -1)
        if e1 < e0 and e1 <= k < e0:
            add(r, -1)
        if max(e0, s1) <= k < e1:
            add(r, 1)
    for r in batch:
        if s1 <= int(r["k"]) < e1:
            add(r, 1)
    exp = {k: w for k, w in exp.items() if w != 0}
    assert got == exp


def test_xxh3_against_reference_library():
    """Pin the xxh3 restatement to the real xxhash library (same algorithm the
    reference's xxhash-rust crate implements) with the reference's seed
    (hash.rs:6)."""
    xxhash = pytest.importorskip("xxhash")
    seed = 0x7F95EF85BE33C337
    rng = np.random.default_rng(12)
    keys = [0, 1, 2**63, 2**64 - 1] + [int(x) for x in
                                       rng.integers(0, 2**63, 50)]
    for k in keys:
        expected = xxhash.xxh3_64_intdigest(int(k).to_bytes(8, "little"), seed=seed)
        assert oracle.xxh3_u64(k) == expected


def test_q0_consolidates_events():
    e1 = auction_event(1, 99, 1, dt=0, expires=10000)
    e2 = bid_event(1, 1000, price=80)
    evs = events(e1, e2, e2)
    out = oracle.q0_step(evs)
    assert len(out) == 2
    assert out["w"].tolist() == [2, 1] or out["w"].tolist() == [1, 2]


def test_oracle_f64_model():
    """f64-weight oracle vs a numpy restatement (sequential sums, ==0.0 drop)."""
    rng = np.random.default_rng(21)
    n = 5000
    r = np.empty(n, dtype=ROW_DT)
    r["k"] = rng.integers(0, 100, n)
    r["v"] = rng.integers(0, 4, n)
    r["w"] = (rng.integers(-8, 9, n).astype(np.float64) * 0.25).view(np.int64)
    got = oracle.consolidate_f64(r)
    d = {}
    order = np.lexsort((r["v"], r["k"]))
    for i in order:
        key = (int(r["k"][i]), int(r["v"][i]))
        d[key] = d.get(key, 0.0) + float(np.int64(r["w"][i]).view(np.float64))
    exp = {k: w for k, w in d.items() if w != 0.0}
    gotd = {(int(x["k"]), int(x["v"])): float(np.int64(x["w"]).view(np.float64))
            for x in got}
    assert gotd == exp
    # merge_f64 == consolidate_f64 of concat on consolidated inputs
    a = oracle.consolidate_f64(r[: n // 2])
    b = oracle.consolidate_f64(r[n // 2:])
    m = oracle.merge_f64(a, b)
    c = oracle.consolidate_f64(np.concatenate([a, b]))
    assert np.array_equal(m, c)
    # -0.0 elimination
    z = np.zeros(2, dtype=ROW_DT)
    z["k"] = [1, 1]
    z["w"] = np.array([0.5, -0.5]).view(np.int64)
    assert len(oracle.consolidate_f64(z)) == 0


def test_oracle_distinct_model():
    """Distinct indicator-difference vs a brute-force model
    (reference distinct.rs proptests pattern)."""
    rng = np.random.default_rng(42)
    trace = oracle.consolidate(_random_rows(rng, 500, key_range=30, val_range=3))
    delta = oracle.consolidate(_random_rows(rng, 200, key_range=30, val_range=3))
    got = zset(oracle.distinct_inc(delta, trace))
    exp = {}
    tr = {(int(r["k"]), int(r["v"])): int(r["w"]) for r in trace}
    for r in delta:
        key = (int(r["k"]), int(r["v"]))
        before = tr.get(key, 0)
        after = before + int(r["w"])
        w = int(after > 0) - int(before > 0)
        if w:
            exp[key] = w
    assert got == exp


def test_oracle_q3_tick_deltas_integrate_to_batch_result():
    """Incremental correctness as an algebraic property: q3 has no windows,
    so the Z-set SUM of the per-tick output deltas must equal the output of
    running the whole stream as one tick (the incremental circuit computes
    d(out) = lift(join)(d(in)) integrated; dbsp paper Thm. 4.4 shape)."""
    from collections import Counter
    from dbsp_amd import gen
    evs = gen.generate(50_000, seed=71)
    q_inc = oracle.Query(3)
    integral = Counter()
    for lo in range(0, len(evs), 5_000):
        for r in q_inc.step(evs[lo:lo + 5_000], cap=1 << 22):
            integral[(int(r["k"]), int(r["v"]))] += int(r["w"])
    q_inc.close()
    integral = {kv: w for kv, w in integral.items() if w != 0}
    q_one = oracle.Query(3)
    single = {(int(r["k"]), int(r["v"])): int(r["w"])
              for r in q_one.step(evs, cap=1 << 22)}
    q_one.close()
    assert integral == single


def test_oracle_window_reference_vectors():
    """The window operator replayed against the reference's own in-tree
    tests (time_series/window.rs:249-455: sliding, tumbling, shrinking) —
    per-tick bounds, input deltas, and expected output deltas transcribed
    to tests/golden/window_ops.json."""
    g = load_golden("window_ops.json")
    for case in g["cases"]:
        trace = np.empty(0, dtype=ROW_DT)
        have_prev = False
        s0 = e0 = 0
        for tick, (b, inp, exp) in enumerate(
                zip(case["bounds"], case["inputs"], case["expected"])):
            s1, e1 = b
            batch = oracle.consolidate(rows_of([tuple(r) for r in inp]))
            got = oracle.window(trace, batch, have_prev, s0, e0, s1, e1)
            assert zset(got) == zset(rows_of([tuple(r) for r in exp])), (
                f"{case['name']}: tick {tick}")
            trace = oracle.merge(trace, batch)
            have_prev, s0, e0 = True, s1, e1


def test_oracle_distinct_reference_vectors():
    """Incremental distinct replayed against the reference's own
    distinct_indexed_test (operator/distinct.rs:825-891): the integral of the
    emitted deltas must match the reference's per-tick distinct integrals."""
    g = load_golden("distinct_indexed.json")
    trace = np.empty(0, dtype=ROW_DT)
    integral = np.empty(0, dtype=ROW_DT)
    for tick, t in enumerate(g["ticks"]):
        delta = oracle.consolidate(rows_of([tuple(r) for r in t["delta"]]))
        out = oracle.distinct_inc(delta, trace)
        integral = oracle.merge(integral, out)
        assert zset(integral) == zset(rows_of(
            [tuple(r) for r in t["integral"]])), f"tick {tick}"
        trace = oracle.merge(trace, delta)


def test_oracle_sharded_q3_unions_to_global():
    """bench.py's all-cores cpu_baseline runs N key-sharded oracle workers
    (persons by id, auctions by seller — the shard points of queries/q3.rs);
    their outputs must union to the single-circuit result (the reference's
    multi-worker equality claim, join.rs:1019-1033 / shard.rs:35-60)."""
    import sys
    from pathlib import Path
    sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
    from bench import _shard_filter
    from dbsp_amd import gen

    evs = gen.generate(200_000, seed=7)
    tick = 20_000
    world = 4
    single = oracle.Query(3)
    shards = [oracle.Query(3) for _ in range(world)]
    for lo in range(0, len(evs), tick):
        chunk = evs[lo:lo + tick]
        expect = zset(single.step(chunk, cap=1 << 22))
        got = {}
        for r in range(world):
            for row in shards[r].step(_shard_filter(chunk, 3, world, r),
                                      cap=1 << 22):
                key = (int(row["k"]), int(row["v"]))
                got[key] = got.get(key, 0) + int(row["w"])
        got = {k: w for k, w in got.items() if w != 0}
        assert got == expect, f"tick at {lo}"
    single.close()
    for q in shards:
        q.close()


def _q4_golden_events(tick):
    import numpy as np
    from dbsp_amd import EVENT_DT
    evs = np.zeros(len(tick["events"]), dtype=EVENT_DT)
    for i, e in enumerate(tick["events"]):
        if e["kind"] == 1:
            evs[i] = (1, e["id"], 1, e["category"], e["date_time"],
                      e["expires"], e["w"])
        else:
            evs[i] = (2, e["auction"], 1, e["price"], e["date_time"], 0,
                      e["w"])
    return evs


def test_oracle_q4_reference_vectors():
    """q4 replayed against the reference's own in-tree test
    (queries/q4.rs:94-239): per-tick (category, avg) output deltas."""
    g = load_golden("q4_category_avg.json")
    q = oracle.Query(4)
    for t, tick in enumerate(g["ticks"]):
        out = q.step(_q4_golden_events(tick), cap=1 << 20)
        got = zset(out)
        exp = {(int(k), int(v)): int(w) for k, v, w in tick["expected"]}
        assert got == exp, f"tick {t}: {got} != {exp}"
    q.close()


def _q6_golden_events(tick):
    import numpy as np
    from dbsp_amd import EVENT_DT
    evs = np.zeros(len(tick["events"]), dtype=EVENT_DT)
    for i, e in enumerate(tick["events"]):
        if e["kind"] == 1:
            evs[i] = (1, e["id"], e["seller"], 1, e["date_time"],
                      e["expires"], e["w"])
        else:
            evs[i] = (2, e["auction"], 1, e["price"], e["date_time"], 0,
                      e["w"])
    return evs


def test_oracle_q6_reference_vectors():
    """q6 replayed against the reference's own in-tree tests (queries/q6.rs:
    single auction, multiple auctions, >10-auction last-10 eviction)."""
    g = load_golden("q6_seller_avg.json")
    for case in g["cases"]:
        q = oracle.Query(6)
        for t, tick in enumerate(case["ticks"]):
            out = q.step(_q6_golden_events(tick), cap=1 << 20)
            got = zset(out)
            exp = {(int(k), int(v)): int(w) for k, v, w in tick["expected"]}
            assert got == exp, f'{case["name"]} tick {t}: {got} != {exp}'
        q.close()
