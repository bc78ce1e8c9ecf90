"""GPU engine bindings (the PRODUCT path).

ctypes layer over libdbsp_hip.so — the C-ABI of include/dbsp_hip.h.  Requires
a real GPU: dbsp_ctx_create fails with DBSP_ERR_NOGPU otherwise (no CPU
fallback, by design).  torch is used only as plumbing for tests that need
device arrays; the engine itself manages its own HBM.
"""
import ctypes

import numpy as np

from . import EVENT_DT, ROW_DT, load_hip_lib

_lib = None


class BatchStruct(ctypes.Structure):
    _fields_ = [("k", ctypes.c_void_p), ("v", ctypes.c_void_p),
                ("w", ctypes.c_void_p), ("len", ctypes.c_int64)]


def _L():
    global _lib
    if _lib is None:
        _lib = load_hip_lib()
        L = _lib
        i64, u64, vp, i32 = (ctypes.c_int64, ctypes.c_uint64, ctypes.c_void_p,
                             ctypes.c_int32)
        L.dbsp_ctx_create.restype = i32
        L.dbsp_ctx_create.argtypes = [ctypes.POINTER(vp), ctypes.c_int]
        L.dbsp_ctx_destroy.argtypes = [vp]
        L.dbsp_ctx_sync.restype = i32
        L.dbsp_ctx_sync.argtypes = [vp]
        L.dbsp_dev_alloc.restype = i32
        L.dbsp_dev_alloc.argtypes = [vp, ctypes.c_size_t, ctypes.POINTER(vp)]
        L.dbsp_dev_free.restype = i32
        L.dbsp_dev_free.argtypes = [vp, vp]
        L.dbsp_h2d.restype = i32
        L.dbsp_h2d.argtypes = [vp, vp, vp, ctypes.c_size_t]
        L.dbsp_d2h.restype = i32
        L.dbsp_d2h.argtypes = [vp, vp, vp, ctypes.c_size_t]
        L.dbsp_sort_consolidate.restype = i32
        L.dbsp_sort_consolidate.argtypes = [vp, vp, vp, vp, i64,
                                            ctypes.POINTER(BatchStruct)]
        L.dbsp_merge.restype = i32
        L.dbsp_merge.argtypes = [vp, ctypes.POINTER(BatchStruct),
                                 ctypes.POINTER(BatchStruct),
                                 ctypes.POINTER(BatchStruct)]
        L.dbsp_join.restype = i32
        L.dbsp_join.argtypes = [vp, ctypes.POINTER(BatchStruct),
                                ctypes.POINTER(BatchStruct), ctypes.c_int, u64,
                                ctypes.POINTER(BatchStruct)]
        L.dbsp_agg_linear_upsert.restype = i32
        L.dbsp_agg_linear_upsert.argtypes = [vp, vp, i64,
                                             ctypes.POINTER(BatchStruct),
                                             ctypes.POINTER(BatchStruct),
                                             ctypes.POINTER(BatchStruct)]
        L.dbsp_agg_max_upsert.restype = i32
        L.dbsp_agg_max_upsert.argtypes = L.dbsp_agg_linear_upsert.argtypes
        L.dbsp_window.restype = i32
        L.dbsp_window.argtypes = [vp, ctypes.POINTER(BatchStruct),
                                  ctypes.POINTER(BatchStruct), ctypes.c_int,
                                  u64, u64, u64, u64,
                                  ctypes.POINTER(BatchStruct)]
        L.dbsp_shard_partition.restype = i32
        L.dbsp_shard_partition.argtypes = [vp, ctypes.POINTER(BatchStruct),
                                           ctypes.c_int,
                                           ctypes.POINTER(BatchStruct), vp]
        L.dbsp_xxh3_u64.restype = u64
        L.dbsp_xxh3_u64.argtypes = [u64, u64]
        L.dbsp_sort_consolidate_f64.restype = i32
        L.dbsp_sort_consolidate_f64.argtypes = [vp, vp, vp, vp, i64,
                                                ctypes.POINTER(BatchStruct)]
        L.dbsp_merge_f64.restype = i32
        L.dbsp_merge_f64.argtypes = [vp, ctypes.POINTER(BatchStruct),
                                     ctypes.POINTER(BatchStruct),
                                     ctypes.POINTER(BatchStruct)]
        L.dbsp_weigh_f64.restype = i32
        L.dbsp_weigh_f64.argtypes = [vp, ctypes.POINTER(BatchStruct),
                                     ctypes.POINTER(BatchStruct)]
        L.dbsp_agg_linear_upsert_f64.restype = i32
        L.dbsp_agg_linear_upsert_f64.argtypes = [vp, vp, i64,
                                                 ctypes.POINTER(BatchStruct),
                                                 ctypes.POINTER(BatchStruct),
                                                 ctypes.POINTER(BatchStruct)]
        L.dbsp_distinct_inc.restype = i32
        L.dbsp_distinct_inc.argtypes = [vp, ctypes.POINTER(BatchStruct),
                                        ctypes.POINTER(BatchStruct),
                                        ctypes.c_int,
                                        ctypes.POINTER(BatchStruct)]
        L.dbsp_unique_keys.restype = i32
        L.dbsp_unique_keys.argtypes = [vp, ctypes.POINTER(BatchStruct),
                                       ctypes.POINTER(vp),
                                       ctypes.POINTER(i64)]
        L.dbsp_comm_unique_id.restype = i32
        L.dbsp_comm_unique_id.argtypes = [vp]
        L.dbsp_comm_init.restype = i32
        L.dbsp_comm_init.argtypes = [vp, ctypes.c_int, ctypes.c_int, vp]
        L.dbsp_engine_create.restype = i32
        L.dbsp_engine_create.argtypes = [ctypes.POINTER(vp), vp, ctypes.c_int,
                                         ctypes.c_int, ctypes.c_int]
        L.dbsp_engine_destroy.argtypes = [vp]
        L.dbsp_engine_step.restype = i32
        L.dbsp_engine_step.argtypes = [vp, vp, i64]
        L.dbsp_engine_stage_events.restype = i32
        L.dbsp_engine_stage_events.argtypes = [vp, vp, i64]
        L.dbsp_engine_step_staged.restype = i32
        L.dbsp_engine_step_staged.argtypes = [vp, i64, i64]
        L.dbsp_engine_run_staged.restype = i32
        L.dbsp_engine_run_staged.argtypes = [vp, i64, i64, i64]
        L.dbsp_engine_output.restype = i32
        L.dbsp_engine_output.argtypes = [vp, vp, i64, ctypes.POINTER(i64)]
        L.dbsp_engine_output_events.restype = i32
        L.dbsp_engine_output_events.argtypes = [vp, vp, i64, ctypes.POINTER(i64)]
        L.dbsp_rolling_agg.restype = i32
        L.dbsp_rolling_agg.argtypes = [vp, ctypes.POINTER(BatchStruct), u64,
                                       ctypes.POINTER(BatchStruct)]
        L.dbsp_engine_c5_init.restype = i32
        L.dbsp_engine_c5_init.argtypes = [vp, i64, i64, u64]
        L.dbsp_engine_kernel_stats.restype = i32
        L.dbsp_engine_kernel_stats.argtypes = [vp, ctypes.c_int,
                                               ctypes.POINTER(ctypes.c_double),
                                               ctypes.POINTER(ctypes.c_double),
                                               ctypes.POINTER(i64)]
    return _lib


def _check(status, what):
    if status != 0:
        raise RuntimeError(f"dbsp {what} failed with status {status}")


def _p(a):
    return a.ctypes.data_as(ctypes.c_void_p)


class Ctx:
    """Per-device context (stream + arena).  Raises without a GPU."""

    def __init__(self, device=0):
        self._lib = _L()
        h = ctypes.c_void_p()
        _check(self._lib.dbsp_ctx_create(ctypes.byref(h), device), "ctx_create")
        self._h = h

    def close(self):
        if getattr(self, "_h", None):
            self._lib.dbsp_ctx_destroy(self._h)
            self._h = None

    def __del__(self):
        self.close()

    def sync(self):
        _check(self._lib.dbsp_ctx_sync(self._h), "sync")

    # ---- host<->device row helpers (test plumbing) ----
    def upload_rows(self, rows: np.ndarray) -> BatchStruct:
        rows = np.ascontiguousarray(rows, dtype=ROW_DT)
        n = len(rows)
        b = BatchStruct()
        k = np.ascontiguousarray(rows["k"])
        v = np.ascontiguousarray(rows["v"])
        w = np.ascontiguousarray(rows["w"])
        for name, host in (("k", k), ("v", v), ("w", w)):
            d = ctypes.c_void_p()
            _check(self._lib.dbsp_dev_alloc(self._h, max(n, 1) * 8,
                                            ctypes.byref(d)), "alloc")
            if n:
                _check(self._lib.dbsp_h2d(self._h, d, _p(host), n * 8), "h2d")
            setattr(b, name, d)
        b.len = n
        self.sync()
        return b

    def download_rows(self, b: BatchStruct) -> np.ndarray:
        n = b.len
        out = np.empty(n, dtype=ROW_DT)
        if n:
            for name in ("k", "v", "w"):
                host = np.empty(n, dtype=np.uint64)
                _check(self._lib.dbsp_d2h(self._h, _p(host),
                                          getattr(b, name), n * 8), "d2h")
                out[name] = host.view(ROW_DT[name])
        return out

    def free_batch(self, b: BatchStruct):
        for name in ("k", "v", "w"):
            ptr = getattr(b, name)
            if ptr:
                self._lib.dbsp_dev_free(self._h, ptr)
                setattr(b, name, None)

    # ---- kernel primitives (GPU parity tests) ----
    def sort_consolidate(self, rows: np.ndarray) -> np.ndarray:
        raw = self.upload_rows(rows)
        out = BatchStruct()
        _check(self._lib.dbsp_sort_consolidate(self._h, raw.k, raw.v, raw.w,
                                               raw.len, ctypes.byref(out)),
               "sort_consolidate")
        self.sync()
        res = self.download_rows(out)
        self.free_batch(raw)
        self.free_batch(out)
        return res

    def merge(self, a_rows, b_rows) -> np.ndarray:
        a = self.upload_rows(a_rows)
        b = self.upload_rows(b_rows)
        out = BatchStruct()
        _check(self._lib.dbsp_merge(self._h, ctypes.byref(a), ctypes.byref(b),
                                    ctypes.byref(out)), "merge")
        res = self.download_rows(out)
        for x in (a, b, out):
            self.free_batch(x)
        return res

    def join(self, delta_rows, trace_rows, proj, param=0) -> np.ndarray:
        d = self.upload_rows(delta_rows)
        t = self.upload_rows(trace_rows)
        out = BatchStruct()
        _check(self._lib.dbsp_join(self._h, ctypes.byref(d), ctypes.byref(t),
                                   proj, param, ctypes.byref(out)), "join")
        self.sync()
        res = self.download_rows(out)
        for x in (d, t, out):
            self.free_batch(x)
        return res

    def agg_linear_upsert(self, keys, in_rows, out_rows) -> np.ndarray:
        keys = np.ascontiguousarray(keys, dtype=np.uint64)
        dk = ctypes.c_void_p()
        _check(self._lib.dbsp_dev_alloc(self._h, max(len(keys), 1) * 8,
                                        ctypes.byref(dk)), "alloc")
        if len(keys):
            _check(self._lib.dbsp_h2d(self._h, dk, _p(keys), len(keys) * 8), "h2d")
        it = self.upload_rows(in_rows)
        ot = self.upload_rows(out_rows)
        out = BatchStruct()
        _check(self._lib.dbsp_agg_linear_upsert(self._h, dk, len(keys),
                                                ctypes.byref(it),
                                                ctypes.byref(ot),
                                                ctypes.byref(out)), "agg")
        self.sync()
        res = self.download_rows(out)
        self._lib.dbsp_dev_free(self._h, dk)
        for x in (it, ot, out):
            self.free_batch(x)
        return res

    def agg_max_upsert(self, keys, in_rows, out_rows) -> np.ndarray:
        keys = np.ascontiguousarray(keys, dtype=np.uint64)
        dk = ctypes.c_void_p()
        _check(self._lib.dbsp_dev_alloc(self._h, max(len(keys), 1) * 8,
                                        ctypes.byref(dk)), "alloc")
        if len(keys):
            _check(self._lib.dbsp_h2d(self._h, dk, _p(keys), len(keys) * 8), "h2d")
        it = self.upload_rows(in_rows)
        ot = self.upload_rows(out_rows)
        out = BatchStruct()
        _check(self._lib.dbsp_agg_max_upsert(self._h, dk, len(keys),
                                             ctypes.byref(it), ctypes.byref(ot),
                                             ctypes.byref(out)), "aggmax")
        self.sync()
        res = self.download_rows(out)
        self._lib.dbsp_dev_free(self._h, dk)
        for x in (it, ot, out):
            self.free_batch(x)
        return res

    def window(self, trace_rows, batch_rows, have_prev, s0, e0, s1, e1):
        t = self.upload_rows(trace_rows)
        b = self.upload_rows(batch_rows)
        out = BatchStruct()
        _check(self._lib.dbsp_window(self._h, ctypes.byref(t), ctypes.byref(b),
                                     1 if have_prev else 0, s0, e0, s1, e1,
                                     ctypes.byref(out)), "window")
        self.sync()
        res = self.download_rows(out)
        for x in (t, b, out):
            self.free_batch(x)
        return res

    def sort_consolidate_f64(self, rows: np.ndarray) -> np.ndarray:
        raw = self.upload_rows(rows)
        out = BatchStruct()
        _check(self._lib.dbsp_sort_consolidate_f64(self._h, raw.k, raw.v, raw.w,
                                                   raw.len, ctypes.byref(out)),
               "sort_consolidate_f64")
        self.sync()
        res = self.download_rows(out)
        self.free_batch(raw)
        self.free_batch(out)
        return res

    def merge_f64(self, a_rows, b_rows) -> np.ndarray:
        a = self.upload_rows(a_rows)
        b = self.upload_rows(b_rows)
        out = BatchStruct()
        _check(self._lib.dbsp_merge_f64(self._h, ctypes.byref(a),
                                        ctypes.byref(b), ctypes.byref(out)),
               "merge_f64")
        res = self.download_rows(out)
        for x in (a, b, out):
            self.free_batch(x)
        return res

    def weigh_f64(self, rows) -> np.ndarray:
        inp = self.upload_rows(rows)
        out = BatchStruct()
        _check(self._lib.dbsp_weigh_f64(self._h, ctypes.byref(inp),
                                        ctypes.byref(out)), "weigh_f64")
        self.sync()
        res = self.download_rows(out)
        for x in (inp, out):
            self.free_batch(x)
        return res

    def agg_linear_upsert_f64(self, keys, in_rows, out_rows) -> np.ndarray:
        keys = np.ascontiguousarray(keys, dtype=np.uint64)
        dk = ctypes.c_void_p()
        _check(self._lib.dbsp_dev_alloc(self._h, max(len(keys), 1) * 8,
                                        ctypes.byref(dk)), "alloc")
        if len(keys):
            _check(self._lib.dbsp_h2d(self._h, dk, _p(keys), len(keys) * 8), "h2d")
        it = self.upload_rows(in_rows)
        ot = self.upload_rows(out_rows)
        out = BatchStruct()
        _check(self._lib.dbsp_agg_linear_upsert_f64(self._h, dk, len(keys),
                                                    ctypes.byref(it),
                                                    ctypes.byref(ot),
                                                    ctypes.byref(out)), "aggf64")
        self.sync()
        res = self.download_rows(out)
        self._lib.dbsp_dev_free(self._h, dk)
        for x in (it, ot, out):
            self.free_batch(x)
        return res

    def distinct_inc(self, delta_rows, trace_batch_rows_list):
        d = self.upload_rows(delta_rows)
        batches = [self.upload_rows(b) for b in trace_batch_rows_list]
        arr = (BatchStruct * max(len(batches), 1))(*batches)
        out = BatchStruct()
        _check(self._lib.dbsp_distinct_inc(self._h, ctypes.byref(d), arr,
                                           len(batches), ctypes.byref(out)),
               "distinct")
        self.sync()
        res = self.download_rows(out)
        self.free_batch(d)
        for b in batches:
            self.free_batch(b)
        self.free_batch(out)
        return res

    def rolling_agg(self, rows, width):
        inp = self.upload_rows(rows)
        out = BatchStruct()
        _check(self._lib.dbsp_rolling_agg(self._h, ctypes.byref(inp), width,
                                          ctypes.byref(out)), "rolling")
        self.sync()
        res = self.download_rows(out)
        for x in (inp, out):
            self.free_batch(x)
        return res

    def shard_partition(self, rows, nshards):
        inp = self.upload_rows(rows)
        out = BatchStruct()
        offs = np.zeros(nshards + 1, dtype=np.int64)
        _check(self._lib.dbsp_shard_partition(self._h, ctypes.byref(inp),
                                              nshards, ctypes.byref(out),
                                              _p(offs)), "shard")
        self.sync()
        res = self.download_rows(out)
        for x in (inp, out):
            self.free_batch(x)
        return res, offs


def xxh3_u64(key, seed=0x7F95EF85BE33C337):
    return _L().dbsp_xxh3_u64(key, seed)


class Engine:
    """One Nexmark query circuit on one GPU (mirror of Runtime::init_circuit +
    DBSPHandle::step)."""

    def __init__(self, ctx: Ctx, query: int, rank=0, world=1):
        self._ctx = ctx
        self._lib = ctx._lib
        h = ctypes.c_void_p()
        _check(self._lib.dbsp_engine_create(ctypes.byref(h), ctx._h, query,
                                            rank, world), "engine_create")
        self._h = h
        self.query = query

    def close(self):
        if getattr(self, "_h", None):
            self._lib.dbsp_engine_destroy(self._h)
            self._h = None

    def __del__(self):
        self.close()

    def stage(self, events: np.ndarray):
        events = np.ascontiguousarray(events, dtype=EVENT_DT)
        _check(self._lib.dbsp_engine_stage_events(self._h, _p(events),
                                                  len(events)), "stage")
        self._staged = events  # keep alive

    def c5_init(self, n_trace, n_delta, seed=41):
        """Config C5 (query 100): device-generate the n_trace-row indexed
        trace and fix the per-tick delta size; step ranges then index delta
        rows (run_staged(0, steps*n_delta, n_delta) = `steps` ticks)."""
        _check(self._lib.dbsp_engine_c5_init(self._h, n_trace, n_delta, seed),
               "c5_init")

    def step_staged(self, lo, hi):
        _check(self._lib.dbsp_engine_step_staged(self._h, lo, hi), "step")

    def run_staged(self, lo, hi, tick):
        """Tick loop inside the C side (the benchmark hot path)."""
        _check(self._lib.dbsp_engine_run_staged(self._h, lo, hi, tick), "run")

    def step(self, events: np.ndarray):
        events = np.ascontiguousarray(events, dtype=EVENT_DT)
        _check(self._lib.dbsp_engine_step(self._h, _p(events), len(events)),
               "step")

    def output(self, cap=1 << 22) -> np.ndarray:
        out = np.empty(cap, dtype=ROW_DT)
        n = ctypes.c_int64()
        _check(self._lib.dbsp_engine_output(self._h, _p(out), cap,
                                            ctypes.byref(n)), "output")
        return out[:n.value].copy()

    def output_events(self, cap=1 << 22) -> np.ndarray:
        out = np.empty(cap, dtype=EVENT_DT)
        n = ctypes.c_int64()
        _check(self._lib.dbsp_engine_output_events(self._h, _p(out), cap,
                                                   ctypes.byref(n)), "output")
        return out[:n.value].copy()

    def kernel_stats(self, kind):
        ms = ctypes.c_double()
        by = ctypes.c_double()
        ln = ctypes.c_int64()
        _check(self._lib.dbsp_engine_kernel_stats(self._h, kind,
                                                  ctypes.byref(ms),
                                                  ctypes.byref(by),
                                                  ctypes.byref(ln)), "stats")
        return ms.value, by.value, ln.value
