"""Property-based oracle invariants (hypothesis), mirroring the reference's
proptest corpus over its TestBatch model (reference trace/test_batch.rs,
trace/consolidation/tests/, trace/layers/test.rs).  These pin the oracle's
ALGEBRA — idempotence, commutativity, associativity, partition invariance —
independently of the hand-picked vectors, so a regression in the restatement
cannot hide behind matching a fixed fixture.

CPU-only (the oracle is the system under test here; the GPU path is compared
against the oracle by the `-m gpu` suites)."""
import numpy as np
from hypothesis import given, settings, strategies as st

from dbsp_amd import ROW_DT
from dbsp_amd import oracle

SETTINGS = dict(max_examples=60, deadline=None)


# small key/value ranges force duplicate (k, v) pairs and weight cancellation
rows_strategy = st.lists(
    st.tuples(st.integers(0, 24), st.integers(0, 6), st.integers(-3, 3)),
    min_size=0, max_size=120,
)


def _rows(triples):
    out = np.empty(len(triples), dtype=ROW_DT)
    for i, (k, v, w) in enumerate(triples):
        out[i] = (k, v, w)
    return out


def _model_consolidate(triples):
    """Independent python model: sum weights by (k, v), drop zeros, sort."""
    acc = {}
    for k, v, w in triples:
        acc[(k, v)] = acc.get((k, v), 0) + w
    items = sorted((k, v, w) for (k, v), w in acc.items() if w != 0)
    return _rows(items)


@given(rows_strategy)
@settings(**SETTINGS)
def test_consolidate_matches_model(triples):
    got = oracle.consolidate(_rows(triples))
    exp = _model_consolidate(triples)
    assert np.array_equal(got, exp)


@given(rows_strategy)
@settings(**SETTINGS)
def test_consolidate_idempotent(triples):
    once = oracle.consolidate(_rows(triples))
    twice = oracle.consolidate(once.copy())
    assert np.array_equal(once, twice)


@given(rows_strategy, rows_strategy)
@settings(**SETTINGS)
def test_merge_is_consolidated_concat(a, b):
    ca, cb = oracle.consolidate(_rows(a)), oracle.consolidate(_rows(b))
    got = oracle.merge(ca, cb)
    exp = _model_consolidate(a + b)
    assert np.array_equal(got, exp)


@given(rows_strategy, rows_strategy)
@settings(**SETTINGS)
def test_merge_commutes(a, b):
    ca, cb = oracle.consolidate(_rows(a)), oracle.consolidate(_rows(b))
    assert np.array_equal(oracle.merge(ca, cb), oracle.merge(cb, ca))


@given(rows_strategy, rows_strategy, rows_strategy)
@settings(**SETTINGS)
def test_merge_associates(a, b, c):
    ca, cb, cc = (oracle.consolidate(_rows(x)) for x in (a, b, c))
    left = oracle.merge(oracle.merge(ca, cb), cc)
    right = oracle.merge(ca, oracle.merge(cb, cc))
    assert np.array_equal(left, right)


@given(rows_strategy, st.integers(2, 5))
@settings(**SETTINGS)
def test_shard_partition_unions_to_global(triples, nshards):
    """xxh3 key partition invariance (reference shard.rs:35-60): the union of
    per-shard consolidations equals the global consolidation — the property
    the reference's multi-worker equality tests pin (join.rs:1019-1033)."""
    all_rows = oracle.consolidate(_rows(triples))
    parts = [[] for _ in range(nshards)]
    for k, v, w in triples:
        parts[oracle.xxh3_u64(k) % nshards].append((k, v, w))
    merged = _rows([])
    for p in parts:
        merged = oracle.merge(merged, oracle.consolidate(_rows(p)))
    assert np.array_equal(merged, all_rows)


@given(rows_strategy, rows_strategy)
@settings(**SETTINGS)
def test_distinct_incremental_matches_model(delta, trace):
    """DistinctIncremental (reference operator/distinct.rs:273,404-462):
    output delta = distinct(trace + delta) - distinct(trace), where
    distinct(z) keeps weight 1 for every (k, v) with positive weight."""
    ct = oracle.consolidate(_rows(trace))
    cd = oracle.consolidate(_rows(delta))
    got = oracle.distinct_inc(cd, ct)

    def distinct(triples):
        acc = {}
        for k, v, w in triples:
            acc[(k, v)] = acc.get((k, v), 0) + w
        return {kv: 1 for kv, w in acc.items() if w > 0}

    before = distinct(trace)
    after = distinct(trace + delta)
    diff = {}
    for kv in set(before) | set(after):
        d = after.get(kv, 0) - before.get(kv, 0)
        if d:
            diff[kv] = d
    exp = _rows(sorted((k, v, w) for (k, v), w in diff.items()))
    assert np.array_equal(got, exp)


@given(rows_strategy, st.integers(0, 20), st.integers(0, 28))
@settings(**SETTINGS)
def test_window_first_tick_selects_batch(triples, lo, span):
    """First tick (no previous bounds): only the BATCH region fires
    (window.rs:209-216) — the output is the batch rows with key in
    [lo, hi)."""
    hi = lo + span
    cb = oracle.consolidate(_rows(triples))
    got = oracle.window(_rows([]), cb, False, lo, hi, lo, hi)
    exp = cb[(cb["k"] >= lo) & (cb["k"] < hi)]
    assert np.array_equal(oracle.consolidate(got.copy()), np.asarray(exp))


@given(rows_strategy, st.integers(0, 12))
@settings(**SETTINGS)
def test_rolling_agg_matches_model(triples, width):
    """Radix-tree rolling aggregate semantics (reference
    rolling_aggregate.rs:235-280, range_of = [ts - width, ts] saturating at
    0): per row, the sum of weights of same-partition rows whose timestamp
    falls in the window."""
    # unique (partition, ts) pairs with weight from the last occurrence
    acc = {}
    for k, v, w in triples:
        acc[(k, v)] = w
    rows = _rows(sorted((k, v, w) for (k, v), w in acc.items()))
    got = oracle.rolling_agg(rows, width)
    for k, ts, s in got:
        t0 = max(0, int(ts) - width)
        exp = sum(int(w) for kk, vv, w in rows
                  if kk == k and t0 <= vv <= ts)
        assert int(s) == exp


# f64-weight variants: weights are f64 BIT PATTERNS in the w column;
# dyadic values (multiples of 0.25) make every reduction order bit-exact,
# so the i64-style equalities must hold verbatim
f64_rows_strategy = st.lists(
    st.tuples(st.integers(0, 24), st.integers(0, 6), st.integers(-8, 8)),
    min_size=0, max_size=100,
)


def _rows_f64(triples):
    out = np.empty(len(triples), dtype=ROW_DT)
    for i, (k, v, q) in enumerate(triples):
        out[i] = (k, v, np.float64(q * 0.25).view(np.int64))
    return out


def _model_consolidate_f64(triples):
    acc = {}
    for k, v, q in triples:
        acc[(k, v)] = acc.get((k, v), 0) + q
    items = sorted((k, v, q) for (k, v), q in acc.items() if q != 0)
    return _rows_f64(items)


@given(f64_rows_strategy)
@settings(**SETTINGS)
def test_consolidate_f64_matches_model(triples):
    got = oracle.consolidate_f64(_rows_f64(triples))
    exp = _model_consolidate_f64(triples)
    assert np.array_equal(got, exp)


@given(f64_rows_strategy, f64_rows_strategy)
@settings(**SETTINGS)
def test_merge_f64_is_consolidated_concat(a, b):
    ca = oracle.consolidate_f64(_rows_f64(a))
    cb = oracle.consolidate_f64(_rows_f64(b))
    got = oracle.merge_f64(ca, cb)
    exp = _model_consolidate_f64(a + b)
    assert np.array_equal(got, exp)


@given(rows_strategy, rows_strategy, st.integers(0, 20), st.integers(0, 24),
       st.integers(0, 10), st.integers(0, 10))
@settings(**SETTINGS)
def test_window_integral_invariant(b1, b2, s1, span1, ds, de):
    """Two ticks of the 3-region retract/insert scan (window.rs:144-220)
    integrate to a plain range select: after the bounds slide MONOTONICALLY
    (the operator's contract — watermark-driven windows only move forward)
    from [s1, e1) to [s2, e2), sum(outputs) == select(sum(inputs),
    [s2, e2))."""
    e1 = s1 + span1
    s2 = s1 + ds
    e2 = max(e1 + de, s2)
    cb1 = oracle.consolidate(_rows(b1))
    out1 = oracle.window(_rows([]), cb1, False, s1, e1, s1, e1)
    cb2 = oracle.consolidate(_rows(b2))
    out2 = oracle.window(cb1, cb2, True, s1, e1, s2, e2)
    got = oracle.merge(oracle.consolidate(out1.copy()),
                       oracle.consolidate(out2.copy()))
    total = _model_consolidate(b1 + b2)
    exp = total[(total["k"] >= s2) & (total["k"] < e2)]
    assert np.array_equal(got, np.asarray(exp))
