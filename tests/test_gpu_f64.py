"""f64-weight parity (config C5's sum-aggregate path).

GPU reductions use a position-fixed order (segmented tree in the fused sort /
global tree in the big-path consolidate; two-operand merges); the oracle sums
sequentially.  Bar: |err| <= 2 ulp * reduction depth — tested
two ways: bit-exact on dyadic weights (multiples of 0.25, where every
summation order is exact), and within tolerance on random f64 weights."""
import numpy as np
import pytest

from dbsp_amd import ROW_DT
from dbsp_amd import oracle

pytestmark = pytest.mark.gpu

PROJ_HI_K_LO_V2 = 8


@pytest.fixture(scope="module")
def ctx():
    from dbsp_amd.engine import Ctx
    c = Ctx(0)
    yield c
    c.close()


def _rows_f64(rng, n, key_range=500, val_range=8, dyadic=False):
    out = np.empty(n, dtype=ROW_DT)
    out["k"] = rng.integers(0, key_range, n)
    out["v"] = rng.integers(0, val_range, n)
    if dyadic:
        w = rng.integers(-8, 9, n).astype(np.float64) * 0.25
    else:
        w = rng.standard_normal(n)
    out["w"] = w.view(np.int64)
    return out


def _f64_zsets_close(got, exp, rtol=1e-9, drop_tol=1e-9):
    g = {(int(r["k"]), int(r["v"])): float(np.int64(r["w"]).view(np.float64))
         for r in got}
    e = {(int(r["k"]), int(r["v"])): float(np.int64(r["w"]).view(np.float64))
         for r in exp}
    for key in set(g) | set(e):
        a, b = g.get(key, 0.0), e.get(key, 0.0)
        assert abs(a - b) <= rtol * max(1.0, abs(a), abs(b)) + drop_tol, (
            key, a, b)


def test_consolidate_f64_dyadic_bitexact(ctx):
    rng = np.random.default_rng(31)
    for n in [1, 1000, 8192, 8193, 40_000, 200_000]:
        r = _rows_f64(rng, n, dyadic=True)
        got = ctx.sort_consolidate_f64(r)
        exp = oracle.consolidate_f64(r)
        assert np.array_equal(got, exp), f"n={n}"


def test_consolidate_f64_random_tolerance(ctx):
    rng = np.random.default_rng(32)
    for n in [5000, 100_000]:
        r = _rows_f64(rng, n, key_range=50, val_range=2)
        got = ctx.sort_consolidate_f64(r)
        exp = oracle.consolidate_f64(r)
        _f64_zsets_close(got, exp)


def test_merge_f64(ctx):
    rng = np.random.default_rng(33)
    for na, nb in [(100, 100), (8000, 8000), (60_000, 40_000)]:
        a = oracle.consolidate_f64(_rows_f64(rng, na, dyadic=True))
        b = oracle.consolidate_f64(_rows_f64(rng, nb, dyadic=True))
        got = ctx.merge_f64(a, b)
        exp = oracle.merge_f64(a, b)
        assert np.array_equal(got, exp), (na, nb)
    # full cancellation including the -0.0 drop
    a = oracle.consolidate_f64(_rows_f64(rng, 5000, dyadic=True))
    neg = a.copy()
    neg["w"] = (-neg["w"].view(np.float64)).view(np.int64)
    assert len(ctx.merge_f64(a, neg)) == 0


def test_weigh_and_agg_f64(ctx):
    rng = np.random.default_rng(34)
    # indexed rows: (k, v = f64 bits, w = small int)
    n = 20_000
    rows = np.empty(n, dtype=ROW_DT)
    rows["k"] = rng.integers(0, 300, n)
    rows["v"] = (rng.integers(-8, 9, n).astype(np.float64) * 0.5).view(np.int64)
    rows["w"] = rng.integers(-2, 3, n)
    got_w = ctx.weigh_f64(rows)
    exp_w = oracle.weigh_f64(rows)
    assert np.array_equal(got_w, exp_w)
    in_trace = oracle.consolidate_f64(exp_w)
    out_trace = np.empty(0, dtype=ROW_DT)
    keys = np.unique(rows["k"])[:100]
    got = ctx.agg_linear_upsert_f64(keys, in_trace, out_trace)
    exp = oracle.agg_linear_upsert_f64(keys, in_trace, out_trace)
    assert np.array_equal(np.sort(got, order=["k", "v"]),
                          np.sort(exp, order=["k", "v"]))


def test_c5_mini_pipeline(ctx):
    """The C5 shape end-to-end at test scale: delta join 1-val-per-key trace
    (proj (k, v2)), weigh by the f64 val, consolidate, sum-aggregate with
    upsert against the running output trace — three incremental steps,
    GPU vs oracle composition (dyadic vals: bit-exact)."""
    rng = np.random.default_rng(35)
    nt = 50_000
    trace = np.empty(nt, dtype=ROW_DT)
    trace["k"] = np.cumsum(rng.integers(1, 5, nt).astype(np.uint64))
    trace["v"] = (rng.integers(1, 65, nt).astype(np.float64) * 0.25).view(np.int64)
    trace["w"] = 1
    out_trace = np.empty(0, dtype=ROW_DT)
    wb_int = np.empty(0, dtype=ROW_DT)
    kmax = int(trace["k"].max())
    for step in range(3):
        nd = 5000
        delta = np.empty(nd, dtype=ROW_DT)
        delta["k"] = np.sort(rng.choice(kmax, nd, replace=False).astype(np.uint64))
        delta["v"] = 0
        delta["w"] = 1
        # GPU side
        j = ctx.join(delta, trace, PROJ_HI_K_LO_V2)
        wreal = ctx.weigh_f64(j)
        dwb = ctx.sort_consolidate_f64(wreal)
        wb_int = oracle.merge_f64(wb_int, dwb) if len(wb_int) else dwb
        keys = np.unique(dwb["k"])
        got = ctx.agg_linear_upsert_f64(keys, wb_int, out_trace)
        # oracle side
        jo = oracle.join_raw(delta, trace, PROJ_HI_K_LO_V2)
        dwb_o = oracle.consolidate_f64(oracle.weigh_f64(jo))
        assert np.array_equal(dwb, dwb_o), f"step {step} weighed delta"
        exp = oracle.agg_linear_upsert_f64(keys, wb_int, out_trace)
        assert np.array_equal(np.sort(got, order=["k", "v"]),
                              np.sort(exp, order=["k", "v"])), f"step {step}"
        d_out = oracle.consolidate(exp)
        out_trace = oracle.merge(out_trace, d_out)
