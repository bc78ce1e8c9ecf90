#!/usr/bin/env python3
"""Kernel-scale roofline bench: drives the C-ABI primitives at the C5-class
sizes (1B-row trace, 10M-row delta) where the merge-path and
join kernels are HBM-bound, and reports achieved algorithmic GB/s against the
8 TB/s HBM3E peak.  torch is used only to materialize device operands
(sorted-unique keys via cumsum of positive gaps — no host staging).

Usage (on a GPU box):  python tools/kbench.py [--rows 500000000]
Prints one JSON line per primitive.
"""
import argparse
import ctypes
import json
import os
import sys
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parents[1]
sys.path.insert(0, str(ROOT / "database-stream-processor_amd" / "python"))

import torch  # noqa: E402

from dbsp_amd.engine import Ctx, BatchStruct, _L  # noqa: E402

HBM_PEAK_GBS = 8000.0


def sorted_unique_batch(n, seed, max_gap=8):
    g = torch.Generator(device="cuda").manual_seed(seed)
    gaps = torch.randint(1, max_gap, (n,), generator=g, dtype=torch.int64,
                         device="cuda")
    k = torch.cumsum(gaps, 0)
    v = torch.zeros(n, dtype=torch.int64, device="cuda")
    w = torch.where(
        torch.rand(n, generator=g, device="cuda") < 0.5,
        torch.tensor(1, dtype=torch.int64, device="cuda"),
        torch.tensor(-1, dtype=torch.int64, device="cuda"))
    torch.cuda.synchronize()
    return k, v, w


def as_batch(k, v, w):
    b = BatchStruct()
    b.k = ctypes.c_void_p(k.data_ptr())
    b.v = ctypes.c_void_p(v.data_ptr())
    b.w = ctypes.c_void_p(w.data_ptr())
    b.len = len(k)
    return b


def bench_merge(ctx, L, n_per_side, reps=3):
    reps = int(os.environ.get("KB_REPS", reps))
    ka, va, wa = sorted_unique_batch(n_per_side, 1)
    kb, vb, wb = sorted_unique_batch(n_per_side, 2)
    a, b = as_batch(ka, va, wa), as_batch(kb, vb, wb)
    times = []
    n_out = 0
    for _ in range(reps):
        out = BatchStruct()
        ctx.sync()
        t0 = time.perf_counter()
        assert L.dbsp_merge(ctx._h, ctypes.byref(a), ctypes.byref(b),
                            ctypes.byref(out)) == 0
        ctx.sync()
        times.append(time.perf_counter() - t0)
        n_out = out.len
        ctx.free_batch(out)
    dt = min(times)
    algo_gb = 24.0 * (2 * n_per_side + n_out) / 1e9
    return {
        "primitive": "dbsp_merge (merge-path trace merge)",
        "rep_ms": [round(t * 1e3, 1) for t in times],
        "rows_in": 2 * n_per_side, "rows_out": n_out,
        "ms": round(dt * 1e3, 2), "algo_GB": round(algo_gb, 2),
        "achieved_GBs": round(algo_gb / dt, 1),
        "frac_of_peak": round(algo_gb / dt / HBM_PEAK_GBS, 4),
    }


def bench_join(ctx, L, n_trace, n_delta, reps=3):
    kt, vt, wt = sorted_unique_batch(n_trace, 3)
    kd, vd, wd = sorted_unique_batch(n_delta, 4,
                                     max_gap=max(2, (6 * n_trace) // n_delta))
    t, d = as_batch(kt, vt, wt), as_batch(kd, vd, wd)
    times = []
    n_out = 0
    for _ in range(reps):
        out = BatchStruct()
        ctx.sync()
        t0 = time.perf_counter()
        assert L.dbsp_join(ctx._h, ctypes.byref(d), ctypes.byref(t), 0, 0,
                           ctypes.byref(out)) == 0
        ctx.sync()
        times.append(time.perf_counter() - t0)
        n_out = out.len
        ctx.free_batch(out)
    dt = min(times)
    # probe model: ~2 effective cache lines per binary-search
    # probe chain per delta row + emitted rows
    algo_gb = (n_delta * 2 * 128 + n_out * 40.0) / 1e9
    return {
        "primitive": "dbsp_join (delta x 1B-row trace probe)",
        "trace_rows": n_trace, "delta_rows": n_delta, "rows_out": n_out,
        "ms": round(dt * 1e3, 2), "algo_GB": round(algo_gb, 2),
        "achieved_GBs": round(algo_gb / dt, 1),
        "frac_of_peak": round(algo_gb / dt / HBM_PEAK_GBS, 4),
    }


def bench_sort(ctx, L, n, reps=3):
    g = torch.Generator(device="cuda").manual_seed(9)
    k = torch.randint(0, 1 << 40, (n,), generator=g, dtype=torch.int64, device="cuda")
    v = torch.randint(0, 1 << 20, (n,), generator=g, dtype=torch.int64, device="cuda")
    w = torch.ones(n, dtype=torch.int64, device="cuda")
    torch.cuda.synchronize()
    b = as_batch(k, v, w)
    times = []
    n_out = 0
    for _ in range(reps):
        out = BatchStruct()
        ctx.sync()
        t0 = time.perf_counter()
        assert L.dbsp_sort_consolidate(ctx._h, b.k, b.v, b.w, n,
                                       ctypes.byref(out)) == 0
        ctx.sync()
        times.append(time.perf_counter() - t0)
        n_out = out.len
        ctx.free_batch(out)
    dt = min(times)
    # LSD radix: passes * (read+write 24B/row); 40+20 significant bits = 8 byte passes
    algo_gb = 8 * 48.0 * n / 1e9
    return {
        "primitive": "dbsp_sort_consolidate (radix, 60-bit keys)",
        "rows": n, "rows_out": n_out, "ms": round(dt * 1e3, 2),
        "algo_GB": round(algo_gb, 2), "achieved_GBs": round(algo_gb / dt, 1),
        "frac_of_peak": round(algo_gb / dt / HBM_PEAK_GBS, 4),
    }


def bench_incremental(ctx, L, n_trace, n_delta, steps=12):
    """C5-class incremental loop (SURVEY.md §8 config 5, i64-weight variant;
    the f64 sum aggregate is a named next step): per step, a 10M-row sorted
    delta joins against the 1B-row trace spine (join is linear: one probe per
    spine batch) and is then merged into the spine under the power-of-two
    leveling — the steady-state trace-maintenance loop at HBM scale."""
    kt, vt, wt = sorted_unique_batch(n_trace, 7)
    wt.fill_(1)  # trace weights +1 so delta inserts never fully cancel
    spine = [(as_batch(kt, vt, wt), None)]  # (batch, dbsp-owned BatchStruct?)
    keep_alive = [(kt, vt, wt)]
    gap = max(2, (9 * n_trace) // (2 * n_delta))
    step_ms = []
    joined_total = 0
    for it in range(steps):
        kd, vd, wd = sorted_unique_batch(n_delta, 100 + it, max_gap=gap)
        wd.fill_(1)
        keep_alive.append((kd, vd, wd))
        d = as_batch(kd, vd, wd)
        ctx.sync()
        t0 = time.perf_counter()
        # join delta against every spine batch (linear in the trace)
        outs = []
        for b, _ in spine:
            o = BatchStruct()
            assert L.dbsp_join(ctx._h, ctypes.byref(d), ctypes.byref(b), 0, 0,
                               ctypes.byref(o)) == 0
            outs.append(o)
        joined = sum(o.len for o in outs)
        for o in outs:
            ctx.free_batch(o)
        # insert the delta into the spine (power-of-two leveling)
        spine.append((d, None))
        while len(spine) >= 2 and spine[-1][0].len * 2 >= spine[-2][0].len:
            (b_top, own_top) = spine.pop()
            (b_below, own_below) = spine.pop()
            m = BatchStruct()
            assert L.dbsp_merge(ctx._h, ctypes.byref(b_below),
                                ctypes.byref(b_top), ctypes.byref(m)) == 0
            for bb, own in ((b_top, own_top), (b_below, own_below)):
                if own is not None:
                    ctx.free_batch(bb)
            spine.append((m, True))
        ctx.sync()
        step_ms.append((time.perf_counter() - t0) * 1e3)
        joined_total += joined
    trace_rows = sum(b.len for b, _ in spine)
    for b, own in spine:
        if own is not None:
            ctx.free_batch(b)
    avg = sum(step_ms[1:]) / max(1, len(step_ms) - 1)
    return {
        "primitive": "incremental join + spine maintenance (C5-class, i64)",
        "trace_rows_initial": n_trace, "delta_rows": n_delta, "steps": steps,
        "trace_rows_final": trace_rows, "join_matches_total": joined_total,
        "ms_per_step": [round(x, 2) for x in step_ms],
        "avg_ms_per_step": round(avg, 2),
        "delta_rows_per_s": round(n_delta / (avg / 1e3), 0),
    }


def bench_c5(ctx, L, n_trace, n_delta, steps=10):
    """Config C5 end-to-end: 1B-row indexed trace
    (1 val/key, f64 vals) x 10M-row delta incremental join, f64 vals weighed
    (f(k,v)=v) into f64 weights, consolidated deterministically, and
    sum-aggregated per key with upsert against the running output trace.
    i64 structural weights bit-exact; f64 sums within 2 ulp * depth."""
    g = torch.Generator(device="cuda").manual_seed(41)
    gaps = torch.randint(1, 8, (n_trace,), generator=g, dtype=torch.int64,
                         device="cuda")
    kt = torch.cumsum(gaps, 0)
    vt = torch.rand(n_trace, generator=g, dtype=torch.float64,
                    device="cuda").view(torch.int64)
    wt = torch.ones(n_trace, dtype=torch.int64, device="cuda")
    torch.cuda.synchronize()
    spine = [(as_batch(kt, vt, wt), None)]
    kmax = int(kt[-1].item())
    wb = None        # f64-weighted integral (auction-sum input trace)
    out_tr = None    # aggregate output trace (k -> f64 bits, i64 weights)
    step_ms = []
    gap = max(2, (9 * n_trace) // (2 * n_delta))
    for it in range(steps):
        gd = torch.Generator(device="cuda").manual_seed(500 + it)
        kd = torch.cumsum(torch.randint(1, gap, (n_delta,), generator=gd,
                                        dtype=torch.int64, device="cuda"), 0)
        vd = torch.zeros(n_delta, dtype=torch.int64, device="cuda")
        wd = torch.ones(n_delta, dtype=torch.int64, device="cuda")
        torch.cuda.synchronize()
        d = as_batch(kd, vd, wd)
        ctx.sync()
        t0 = time.perf_counter()
        # join vs every spine batch (proj (k, v2): carry the f64 val)
        outs = []
        for b, _ in spine:
            o = BatchStruct()
            assert L.dbsp_join(ctx._h, ctypes.byref(d), ctypes.byref(b), 8, 0,
                               ctypes.byref(o)) == 0
            outs.append(o)
        # weigh + consolidate each join output, fold into the f64 integral
        for o in outs:
            if o.len > 0:
                wre = BatchStruct()
                assert L.dbsp_weigh_f64(ctx._h, ctypes.byref(o),
                                        ctypes.byref(wre)) == 0
                dwb = BatchStruct()
                assert L.dbsp_sort_consolidate_f64(ctx._h, wre.k, wre.v, wre.w,
                                                   wre.len,
                                                   ctypes.byref(dwb)) == 0
                ctx.free_batch(wre)
                if wb is None:
                    wb = dwb
                else:
                    m = BatchStruct()
                    assert L.dbsp_merge_f64(ctx._h, ctypes.byref(wb),
                                            ctypes.byref(dwb),
                                            ctypes.byref(m)) == 0
                    ctx.free_batch(wb)
                    ctx.free_batch(dwb)
                    wb = m
            ctx.free_batch(o)
        # aggregate affected keys + upsert
        if wb is not None and wb.len > 0:
            keys = ctypes.c_void_p()
            nk = ctypes.c_int64()
            assert L.dbsp_unique_keys(ctx._h, ctypes.byref(wb),
                                      ctypes.byref(keys),
                                      ctypes.byref(nk)) == 0
            empty = BatchStruct()
            upd = BatchStruct()
            assert L.dbsp_agg_linear_upsert_f64(
                ctx._h, keys, nk.value, ctypes.byref(wb),
                ctypes.byref(out_tr if out_tr else empty),
                ctypes.byref(upd)) == 0
            L.dbsp_dev_free(ctx._h, keys)
            cons = BatchStruct()
            assert L.dbsp_sort_consolidate(ctx._h, upd.k, upd.v, upd.w,
                                           upd.len, ctypes.byref(cons)) == 0
            ctx.free_batch(upd)
            if out_tr is None:
                out_tr = cons
            else:
                m = BatchStruct()
                assert L.dbsp_merge(ctx._h, ctypes.byref(out_tr),
                                    ctypes.byref(cons), ctypes.byref(m)) == 0
                ctx.free_batch(out_tr)
                ctx.free_batch(cons)
                out_tr = m
        # insert the delta into the join trace spine
        spine.append((d, None))
        while len(spine) >= 2 and spine[-1][0].len * 2 >= spine[-2][0].len:
            (b_top, own_top) = spine.pop()
            (b_below, own_below) = spine.pop()
            m = BatchStruct()
            assert L.dbsp_merge(ctx._h, ctypes.byref(b_below),
                                ctypes.byref(b_top), ctypes.byref(m)) == 0
            for bb, own in ((b_top, own_top), (b_below, own_below)):
                if own is not None:
                    ctx.free_batch(bb)
            spine.append((m, True))
        ctx.sync()
        step_ms.append((time.perf_counter() - t0) * 1e3)
    res = {
        "primitive": "C5 end-to-end: 1B trace x 10M delta incremental join "
                     "+ f64 weigh/consolidate/sum-aggregate (i64 join, f64 agg)",
        "trace_rows": n_trace, "delta_rows": n_delta, "steps": steps,
        "agg_out_rows": out_tr.len if out_tr else 0,
        "wb_rows": wb.len if wb else 0,
        "ms_per_step": [round(x, 2) for x in step_ms],
        "avg_ms_per_step": round(sum(step_ms[1:]) / max(1, len(step_ms) - 1), 2),
    }
    for b, own in spine:
        if own is not None:
            ctx.free_batch(b)
    if wb is not None:
        ctx.free_batch(wb)
    if out_tr is not None:
        ctx.free_batch(out_tr)
    return res


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=500_000_000,
                    help="rows per merge side (trace = 2x this)")
    ap.add_argument("--delta", type=int, default=10_000_000)
    ap.add_argument("--sort-rows", type=int, default=50_000_000)
    ap.add_argument("--only", default=None,
                    choices=["merge", "join", "sort", "incremental", "c5"])
    args = ap.parse_args()
    ctx = Ctx(0)
    L = _L()
    legs = {
        "merge": lambda: bench_merge(ctx, L, args.rows),
        "join": lambda: bench_join(ctx, L, 2 * args.rows, args.delta),
        "sort": lambda: bench_sort(ctx, L, args.sort_rows),
        "incremental": lambda: bench_incremental(ctx, L, 2 * args.rows,
                                                 args.delta),
        "c5": lambda: bench_c5(ctx, L, 2 * args.rows, args.delta),
    }
    for name, leg in legs.items():
        if args.only is None or args.only == name:
            print(json.dumps(leg()))
    ctx.close()


if __name__ == "__main__":
    main()
