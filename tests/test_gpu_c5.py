"""Config C5 engine parity (BASELINE configs[4]): the query-100 engine's
incremental 'trace x delta join + f64 sum aggregate' loop vs a CPU replica
built from the oracle primitives, at a small scale the oracle finishes in
seconds.

The engine generates operands on device (counter-based splitmix64,
kernels.hip c5_gen_rows); the replica regenerates the identical rows in
numpy.  Per tick the replica follows the same operator sequence the engine's
c5_step documents (join carrying the trace val -> weigh -> consolidate ->
fold into the weighted integral -> aggregate affected keys + upsert vs the
output trace -> TraceAppend), each oracle function citing the reference
lines it restates.  Bar: i64 structure bit-exact; f64 sums within
2 ulp * reduction depth (checked via relative tolerance on the emitted
aggregate values)."""
import numpy as np
import pytest

from dbsp_amd import ROW_DT
from dbsp_amd import oracle

pytestmark = pytest.mark.gpu

M64 = np.uint64(0xFFFFFFFFFFFFFFFF)
PHI = np.uint64(0x9E3779B97F4A7C15)


def smix(x):
    """splitmix64 finalizer — mirrors c5_mix in kernels.hip."""
    z = (x + PHI) & M64
    z = ((z ^ (z >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)) & M64
    z = ((z ^ (z >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)) & M64
    return z ^ (z >> np.uint64(31))


def c5_gen(n, stride, jitter, seed, val_mode):
    i = np.arange(n, dtype=np.uint64)
    h = smix((np.uint64(seed) + i * PHI) & M64)
    out = np.empty(n, dtype=ROW_DT)
    out["k"] = (np.uint64(stride) * i + (h % np.uint64(jitter) if jitter else 0)) & M64
    if val_mode == 0:
        u = (smix(h) >> np.uint64(11)).astype(np.float64) * 2.0**-53
        out["v"] = u.view(np.uint64)
    else:
        out["v"] = 0
    out["w"] = 1
    return out


def c5_oracle_tick(trace, wint, out_trace, delta):
    """One C5 tick on the CPU oracle; returns (new wint, new out_trace,
    output delta)."""
    # caps sized for this workload (1 val/key: <= 1 match per delta row;
    # join_raw's default cap is delta*trace — petabytes at C5 scale)
    joined = oracle.join_raw(delta, trace, 8, cap=4 * len(delta) + 1024)
    if len(joined) == 0:
        return wint, out_trace, np.empty(0, dtype=ROW_DT)
    weighed = oracle.weigh_f64(joined)
    dwb = oracle.consolidate_f64(weighed)
    wint = oracle.merge_f64(wint, dwb) if len(wint) else dwb
    keys = np.unique(dwb["k"])
    upd = oracle.agg_linear_upsert_f64(keys, wint, out_trace,
                                       cap=4 * len(keys) + 1024)
    upd = oracle.consolidate(upd)
    if len(upd):
        out_trace = oracle.merge(out_trace, upd)
    return wint, out_trace, upd


def test_c5_engine_parity():
    from dbsp_amd.engine import Ctx, Engine
    n_trace, n_delta, steps, seed = 200_000, 20_000, 6, 41
    ctx = Ctx(0)
    eng = Engine(ctx, query=100)
    eng.c5_init(n_trace, n_delta, seed=seed)

    trace = c5_gen(n_trace, 5, 4, seed, 0)
    stride = max(3, (5 * n_trace) // n_delta)
    wint = np.empty(0, dtype=ROW_DT)
    out_trace = np.empty(0, dtype=ROW_DT)
    for t in range(steps):
        eng.step_staged(t * n_delta, (t + 1) * n_delta)
        got = eng.output()
        dseed = (seed + 0x9E3779B97F4A7C15 * (t + 1)) % (1 << 64)
        delta = c5_gen(n_delta, stride, stride - 1, dseed, 1)
        wint, out_trace, exp = c5_oracle_tick(trace, wint, out_trace, delta)
        # structure (keys, vals-as-bits partition, weights): compare as f64
        # z-sets with tolerance on the aggregate bit patterns
        g = {(int(r["k"]), int(r["w"])): np.int64(r["v"]).view(np.float64)
             for r in got}
        e = {(int(r["k"]), int(r["w"])): np.int64(r["v"]).view(np.float64)
             for r in exp}
        assert set(g) == set(e), f"tick {t}: key/weight structure differs"
        for key in g:
            a, b = float(g[key]), float(e[key])
            assert abs(a - b) <= 1e-9 * max(1.0, abs(a), abs(b)), (t, key, a, b)
        trace = oracle.merge(trace, delta)
    eng.close()
    ctx.close()


def test_c5_generator_matches_device():
    """The numpy replica and the device generator produce identical rows
    (spot check via the engine's first tick join against a replica trace —
    covered transitively by test_c5_engine_parity; here check the host-side
    generator invariants directly)."""
    r = c5_gen(100_000, 5, 4, 41, 0)
    k = r["k"].astype(np.uint64)
    assert (k[1:] > k[:-1]).all()  # strictly increasing = sorted unique
    v = r["v"].view(np.float64)
    assert ((v >= 0) & (v < 1)).all()
    d = c5_gen(50_000, 50, 49, 7, 1)
    dk = d["k"].astype(np.uint64)
    assert (dk[1:] > dk[:-1]).all()
    assert (d["v"] == 0).all()
