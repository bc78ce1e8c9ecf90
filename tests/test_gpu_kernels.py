"""GPU kernel parity vs the CPU oracle (bit-exact: i64 weights, u64 keys).

Each test drives a C-ABI entry point of libdbsp_hip.so (include/dbsp_hip.h)
with device buffers and compares against oracle/liboracle_dbsp.so on the same
inputs — randomized plus the adversarial shapes the reference's own tests use
(empty, all-equal, disjoint, zero-sum: trace/consolidation/tests,
trace/layers/test.rs)."""
import numpy as np
import pytest

from dbsp_amd import ROW_DT
from dbsp_amd import oracle
from helpers import rows_of, zset

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ctx():
    from dbsp_amd.engine import Ctx
    c = Ctx(0)
    yield c
    c.close()


def _random_rows(rng, n, key_range=1000, val_range=50, w_range=3):
    out = np.empty(n, dtype=ROW_DT)
    out["k"] = rng.integers(0, key_range, n)
    out["v"] = rng.integers(0, val_range, n)
    out["w"] = rng.integers(-w_range, w_range + 1, n)
    return out


def _sorted_eq(a, b):
    return np.array_equal(np.sort(a, order=["k", "v", "w"]),
                          np.sort(b, order=["k", "v", "w"]))


def test_sort_consolidate_parity(ctx):
    rng = np.random.default_rng(1)
    cases = [
        _random_rows(rng, 0),
        _random_rows(rng, 1),
        _random_rows(rng, 40_000),
        _random_rows(rng, 100_000, key_range=2**40, val_range=2**30),
        _random_rows(rng, 2048),          # exactly one sort tile
        _random_rows(rng, 2049),
        _random_rows(rng, 65_536),        # dense-range path (small key box)
        _random_rows(rng, 65_537),
        _random_rows(rng, 20_000, key_range=5, val_range=200),  # q5 tick shape
        _random_rows(rng, 40_000, key_range=2**40, val_range=2**30),  # chunked
    ]
    # medium-path all-equal keys, zero-sum weights (single segment, nseg=1)
    z40 = np.zeros(40_000, dtype=ROW_DT)
    z40["k"] = 3
    z40["v"] = 4
    z40["w"] = np.where(np.arange(40_000) % 2 == 0, 1, -1)
    cases.append(z40)
    # all-equal keys, zero-sum weights
    z = np.zeros(1000, dtype=ROW_DT)
    z["k"] = 7
    z["v"] = 9
    z["w"] = np.where(np.arange(1000) % 2 == 0, 1, -1)
    cases.append(z)
    # bitonic fast-path band (n <= 2048): pow2 boundaries, duplicate-heavy
    # (consolidation across thread ownership), cancel-heavy (zero-weight
    # drop), and wide 64-bit keys (the bitonic compares full (k, v))
    for n in (2, 3, 1023, 1024, 1025, 2047):
        cases.append(_random_rows(rng, n, key_range=2**63, val_range=2**63))
    dup = _random_rows(rng, 1500, key_range=40, val_range=3)
    cases.append(dup)
    canc = _random_rows(rng, 2000, key_range=50, val_range=2, w_range=1)
    cases.append(canc)
    for r in cases:
        got = ctx.sort_consolidate(r)
        exp = oracle.consolidate(r)
        assert np.array_equal(got, exp), f"n={len(r)}"


def test_merge_parity(ctx):
    rng = np.random.default_rng(2)
    for na, nb in [(0, 0), (0, 100), (100, 0), (1, 1), (5000, 3),
                   (40_000, 40_000), (300_000, 1000)]:
        a = oracle.consolidate(_random_rows(rng, na, key_range=5000))
        b = oracle.consolidate(_random_rows(rng, nb, key_range=5000))
        got = ctx.merge(a, b)
        exp = oracle.merge(a, b)
        assert np.array_equal(got, exp), f"na={na} nb={nb}"
    # full cancellation
    a = oracle.consolidate(_random_rows(rng, 10_000))
    neg = a.copy()
    neg["w"] = -neg["w"]
    assert len(ctx.merge(a, neg)) == 0


def test_join_parity(ctx):
    rng = np.random.default_rng(3)
    for proj in [0, 1, 2, 5, 6, 7]:
        d = oracle.consolidate(_random_rows(rng, 5000, key_range=300))
        t = oracle.consolidate(_random_rows(rng, 20_000, key_range=300))
        got = ctx.join(d, t, proj)
        exp = oracle.join_raw(d, t, proj, cap=len(d) * 400)
        assert zset(got) == zset(exp), f"proj={proj}"
    # rounding projections (param)
    for proj in [3, 4]:
        d = oracle.consolidate(_random_rows(rng, 1000, key_range=100,
                                            val_range=2**40))
        t = oracle.consolidate(_random_rows(rng, 1000, key_range=100,
                                            val_range=2**40))
        got = ctx.join(d, t, proj, param=10_000)
        exp = oracle.join_raw(d, t, proj, param=10_000, cap=len(d) * 100)
        assert zset(got) == zset(exp), f"proj={proj}"
    # empty sides
    e = np.empty(0, dtype=ROW_DT)
    assert len(ctx.join(e, t, 0)) == 0
    assert len(ctx.join(d, e, 0)) == 0


def test_agg_linear_parity(ctx):
    rng = np.random.default_rng(4)
    in_trace = oracle.consolidate(_random_rows(rng, 30_000, key_range=500,
                                               val_range=1))
    out_trace = oracle.consolidate(_random_rows(rng, 400, key_range=500,
                                                val_range=10, w_range=1))
    keys = np.unique(in_trace["k"])[:200]
    got = ctx.agg_linear_upsert(keys, in_trace, out_trace)
    exp = oracle.agg_linear_upsert(keys, in_trace, out_trace)
    assert zset(got) == zset(exp)
    # keys absent from both traces
    missing = np.array([10**9, 10**9 + 1], dtype=np.uint64)
    got = ctx.agg_linear_upsert(missing, in_trace, out_trace)
    assert len(got) == 0


def test_agg_max_parity(ctx):
    rng = np.random.default_rng(5)
    in_trace = oracle.consolidate(_random_rows(rng, 5000, key_range=100,
                                               val_range=1000, w_range=2))
    out_trace = oracle.consolidate(_random_rows(rng, 100, key_range=100,
                                                val_range=1000, w_range=1))
    keys = np.unique(in_trace["k"])[:50]
    got = ctx.agg_max_upsert(keys, in_trace, out_trace)
    exp = oracle.agg_max_upsert(keys, in_trace, out_trace)
    assert zset(got) == zset(exp)


def test_window_parity(ctx):
    rng = np.random.default_rng(6)
    trace = oracle.consolidate(_random_rows(rng, 20_000, key_range=100_000))
    batch = oracle.consolidate(_random_rows(rng, 2000, key_range=100_000))
    cases = [
        (False, 0, 0, 10_000, 60_000),
        (True, 10_000, 60_000, 30_000, 90_000),   # slide forward
        (True, 10_000, 60_000, 20_000, 40_000),   # shrink right
        (True, 10_000, 60_000, 10_000, 60_000),   # unchanged
        (True, 0, 0, 0, 0),                       # empty window
    ]
    for hp, s0, e0, s1, e1 in cases:
        got = ctx.window(trace, batch, hp, s0, e0, s1, e1)
        exp = oracle.window(trace, batch, hp, s0, e0, s1, e1)
        assert zset(got) == zset(exp), (hp, s0, e0, s1, e1)


def test_shard_partition_parity(ctx):
    rng = np.random.default_rng(7)
    rows = oracle.consolidate(_random_rows(rng, 50_000, key_range=2**50))
    for nshards in [2, 4, 8]:
        got, offs = ctx.shard_partition(rows, nshards)
        assert offs[-1] == len(rows)
        seed = 0x7F95EF85BE33C337
        for s in range(nshards):
            part = got[offs[s]:offs[s + 1]]
            # every row in shard s hashes to s (hash.rs:9-13 + shard.rs:183)
            for k in np.unique(part["k"]):
                assert oracle.xxh3_u64(int(k), seed) % nshards == s
        # partition preserves the multiset
        assert zset(got) == zset(rows)


def test_device_xxh3_matches_oracle(ctx):
    from dbsp_amd.engine import xxh3_u64
    rng = np.random.default_rng(8)
    for k in [0, 1, 2**63] + [int(x) for x in rng.integers(0, 2**63, 20)]:
        assert xxh3_u64(k) == oracle.xxh3_u64(k)


def test_distinct_inc_parity(ctx):
    """Incremental distinct (operator/distinct.rs root scope): delta vs a
    multi-batch integral whose per-pair totals only emerge across batches."""
    rng = np.random.default_rng(41)
    b1 = oracle.consolidate(_random_rows(rng, 3000, key_range=100, val_range=4))
    b2 = oracle.consolidate(_random_rows(rng, 3000, key_range=100, val_range=4))
    delta = oracle.consolidate(_random_rows(rng, 2000, key_range=100, val_range=4))
    merged = oracle.merge(b1, b2)
    got = ctx.distinct_inc(delta, [b1, b2])
    exp = oracle.distinct_inc(delta, merged)
    assert np.array_equal(got, exp)
    # adversarial: weights cancelling across batches (before == 0), and
    # transitions in both directions
    neg = b1.copy()
    neg["w"] = -neg["w"]
    got = ctx.distinct_inc(delta, [b1, neg])
    exp = oracle.distinct_inc(delta, np.empty(0, dtype=ROW_DT))
    assert np.array_equal(got, exp)
    # empty trace and empty delta
    assert len(ctx.distinct_inc(np.empty(0, dtype=ROW_DT), [b1])) == 0


def test_window_reference_vectors(ctx):
    """dbsp_window replayed against the reference's own in-tree window tests
    (time_series/window.rs:249-455 sliding/tumbling/shrinking, transcribed to
    tests/golden/window_ops.json): per-tick bounds + input deltas must yield
    the reference's exact output deltas, including left-shrinks and
    jump-forward windows."""
    from helpers import load_golden, rows_of
    import numpy as np
    g = load_golden("window_ops.json")
    for case in g["cases"]:
        trace = np.empty(0, dtype=ROW_DT)
        have_prev = False
        s0 = e0 = 0
        for tick, (b, inp, exp) in enumerate(
                zip(case["bounds"], case["inputs"], case["expected"])):
            s1, e1 = b
            batch = oracle.consolidate(rows_of([tuple(r) for r in inp]))
            got = ctx.window(trace, batch, have_prev, s0, e0, s1, e1)
            assert zset(got) == zset(rows_of([tuple(r) for r in exp])), (
                f"{case['name']}: tick {tick}")
            trace = oracle.merge(trace, batch)
            have_prev, s0, e0 = True, s1, e1


def test_distinct_reference_vectors(ctx):
    """dbsp_distinct_inc replayed against the reference's
    distinct_indexed_test vectors (operator/distinct.rs:825-891)."""
    from helpers import load_golden, rows_of
    g = load_golden("distinct_indexed.json")
    trace = np.empty(0, dtype=ROW_DT)
    integral = np.empty(0, dtype=ROW_DT)
    for tick, t in enumerate(g["ticks"]):
        delta = oracle.consolidate(rows_of([tuple(r) for r in t["delta"]]))
        out = ctx.distinct_inc(delta, [trace] if len(trace) else [])
        integral = oracle.merge(integral, out)
        assert zset(integral) == zset(rows_of(
            [tuple(r) for r in t["integral"]])), f"tick {tick}"
        trace = oracle.merge(trace, delta)


def test_rolling_agg_parity(ctx):
    """Radix-tree rolling aggregate (radix_tree/mod.rs:1-75,
    rolling_aggregate.rs:235-280): per-row [ts-width, ts] partition-local
    weight sums over the flat radix-16 prefix-aggregate tree, bit-exact vs
    the naive oracle scan.  Covers width 0 (self only), narrow, wide and
    whole-history windows, single-partition and many-partition batches, and
    sizes across the tree's level boundaries (16^k edges)."""
    rng = np.random.default_rng(77)
    for n, nparts, trange in [(1, 1, 10), (255, 4, 100), (256, 4, 100),
                              (257, 4, 100), (4096, 1, 10_000),
                              (65_537, 64, 100_000), (200_000, 500, 50_000)]:
        rows = np.empty(n, dtype=ROW_DT)
        rows["k"] = rng.integers(0, nparts, n)
        rows["v"] = rng.integers(0, trange, n)
        rows["w"] = rng.integers(-5, 6, n)
        rows = oracle.consolidate(rows)
        if len(rows) == 0:
            continue
        for width in [0, 3, trange // 7 + 1, 1 << 62]:
            got = ctx.rolling_agg(rows, width)
            exp = oracle.rolling_agg(rows, width)
            assert np.array_equal(got, exp), (n, nparts, width)
