"""CPU oracle bindings — TEST INFRASTRUCTURE + bench cpu_baseline ONLY.

Wraps oracle/liboracle_dbsp.so (see its header comment for scope rules and the
reference citations).  The product path (dbsp_amd.engine) never imports this
module.
"""
import ctypes

import numpy as np

from . import EVENT_DT, ROW_DT, load_oracle_lib

_lib = None


def _o():
    global _lib
    if _lib is None:
        _lib = load_oracle_lib()
        L = _lib
        i64, u64, vp = ctypes.c_int64, ctypes.c_uint64, ctypes.c_void_p
        L.oracle_consolidate.restype = i64
        L.oracle_consolidate.argtypes = [vp, i64]
        L.oracle_merge.restype = i64
        L.oracle_merge.argtypes = [vp, i64, vp, i64, vp]
        L.oracle_join.restype = i64
        L.oracle_join.argtypes = [vp, i64, vp, i64, ctypes.c_int, u64, vp, i64]
        L.oracle_agg_linear_upsert.restype = i64
        L.oracle_agg_linear_upsert.argtypes = [vp, i64, vp, i64, vp, i64, vp, i64]
        L.oracle_agg_max_upsert.restype = i64
        L.oracle_agg_max_upsert.argtypes = [vp, i64, vp, i64, vp, i64, vp, i64]
        L.oracle_window.restype = i64
        L.oracle_window.argtypes = [vp, i64, vp, i64, ctypes.c_int, u64, u64, u64, u64, vp, i64]
        L.oracle_xxh3_u64.restype = u64
        L.oracle_xxh3_u64.argtypes = [u64, u64]
        L.oracle_rolling_agg.restype = i64
        L.oracle_rolling_agg.argtypes = [vp, i64, u64, vp]
        L.oracle_query_new.restype = vp
        L.oracle_query_new.argtypes = [ctypes.c_int]
        L.oracle_query_free.argtypes = [vp]
        L.oracle_query_step.restype = i64
        L.oracle_query_step.argtypes = [vp, vp, i64, vp, i64]
        L.oracle_q0_step.restype = i64
        L.oracle_q0_step.argtypes = [vp, i64, vp, i64]
        L.oracle_consolidate_f64.restype = i64
        L.oracle_consolidate_f64.argtypes = [vp, i64]
        L.oracle_merge_f64.restype = i64
        L.oracle_merge_f64.argtypes = [vp, i64, vp, i64, vp]
        L.oracle_weigh_f64.restype = i64
        L.oracle_weigh_f64.argtypes = [vp, i64, vp]
        L.oracle_agg_linear_upsert_f64.restype = i64
        L.oracle_agg_linear_upsert_f64.argtypes = [vp, i64, vp, i64, vp, i64, vp, i64]
        L.oracle_distinct_inc.restype = i64
        L.oracle_distinct_inc.argtypes = [vp, i64, vp, i64, vp]
    return _lib


def _p(a):
    return a.ctypes.data_as(ctypes.c_void_p)


def consolidate(r: np.ndarray) -> np.ndarray:
    r = np.array(r, dtype=ROW_DT)
    n = _o().oracle_consolidate(_p(r), len(r))
    return r[:n].copy()


def merge(a: np.ndarray, b: np.ndarray) -> np.ndarray:
    out = np.empty(len(a) + len(b), dtype=ROW_DT)
    n = _o().oracle_merge(_p(a), len(a), _p(b), len(b), _p(out))
    return out[:n].copy()


def join_raw(delta, trace, proj, param=0, cap=None):
    cap = cap or max(16, len(delta) * max(1, len(trace)) * 4)
    out = np.empty(cap, dtype=ROW_DT)
    n = _o().oracle_join(_p(delta), len(delta), _p(trace), len(trace), proj, param,
                         _p(out), cap)
    assert n >= 0, "oracle_join overflow"
    return out[:n].copy()


def agg_linear_upsert(keys, in_trace, out_trace, cap=1 << 16):
    keys = np.asarray(keys, dtype=np.uint64)
    out = np.empty(cap, dtype=ROW_DT)
    n = _o().oracle_agg_linear_upsert(_p(keys), len(keys), _p(in_trace),
                                      len(in_trace), _p(out_trace), len(out_trace),
                                      _p(out), cap)
    assert n >= 0
    return out[:n].copy()


def agg_max_upsert(keys, in_trace, out_trace, cap=1 << 16):
    keys = np.asarray(keys, dtype=np.uint64)
    out = np.empty(cap, dtype=ROW_DT)
    n = _o().oracle_agg_max_upsert(_p(keys), len(keys), _p(in_trace), len(in_trace),
                                   _p(out_trace), len(out_trace), _p(out), cap)
    assert n >= 0
    return out[:n].copy()


def window(trace, batch, have_prev, s0, e0, s1, e1, cap=None):
    cap = cap or max(16, (len(trace) + len(batch)) * 2)
    out = np.empty(cap, dtype=ROW_DT)
    n = _o().oracle_window(_p(trace), len(trace), _p(batch), len(batch),
                           1 if have_prev else 0, s0, e0, s1, e1, _p(out), cap)
    assert n >= 0
    return out[:n].copy()


def xxh3_u64(key, seed=0x7F95EF85BE33C337):
    return _o().oracle_xxh3_u64(key, seed)


def rolling_agg(rows, width):
    rows = np.ascontiguousarray(rows, dtype=ROW_DT)
    out = np.empty(len(rows), dtype=ROW_DT)
    n = _o().oracle_rolling_agg(_p(rows), len(rows), width, _p(out))
    return out[:n].copy()


class Query:
    """Incremental oracle for Nexmark q3/q5/q8 (q0 via q0_step)."""

    def __init__(self, which: int):
        self._lib = _o()
        self._h = self._lib.oracle_query_new(which)
        assert self._h

    def step(self, events: np.ndarray, cap=1 << 20) -> np.ndarray:
        events = np.asarray(events, dtype=EVENT_DT)
        out = np.empty(cap, dtype=ROW_DT)
        n = self._lib.oracle_query_step(self._h, _p(events), len(events), _p(out), cap)
        assert n >= 0, f"oracle_query_step error {n}"
        return out[:n].copy()

    def close(self):
        if self._h:
            self._lib.oracle_query_free(self._h)
            self._h = None

    def __del__(self):
        self.close()


def q0_step(events: np.ndarray, cap=None) -> np.ndarray:
    events = np.asarray(events, dtype=EVENT_DT)
    cap = cap or len(events) + 16
    out = np.empty(cap, dtype=EVENT_DT)
    n = _o().oracle_q0_step(_p(events), len(events), _p(out), cap)
    assert n >= 0
    return out[:n].copy()


# ---- f64-weight variants (weights are f64 bit patterns in the w column) ----

def consolidate_f64(r: np.ndarray) -> np.ndarray:
    r = np.array(r, dtype=ROW_DT)
    n = _o().oracle_consolidate_f64(_p(r), len(r))
    return r[:n].copy()


def merge_f64(a: np.ndarray, b: np.ndarray) -> np.ndarray:
    out = np.empty(len(a) + len(b), dtype=ROW_DT)
    n = _o().oracle_merge_f64(_p(a), len(a), _p(b), len(b), _p(out))
    return out[:n].copy()


def weigh_f64(r: np.ndarray) -> np.ndarray:
    out = np.empty(len(r), dtype=ROW_DT)
    n = _o().oracle_weigh_f64(_p(np.ascontiguousarray(r, dtype=ROW_DT)), len(r), _p(out))
    return out[:n].copy()


def agg_linear_upsert_f64(keys, in_trace, out_trace, cap=1 << 20):
    keys = np.asarray(keys, dtype=np.uint64)
    out = np.empty(cap, dtype=ROW_DT)
    n = _o().oracle_agg_linear_upsert_f64(_p(keys), len(keys), _p(in_trace),
                                          len(in_trace), _p(out_trace),
                                          len(out_trace), _p(out), cap)
    assert n >= 0
    return out[:n].copy()


def distinct_inc(delta, trace):
    out = np.empty(max(len(delta), 1), dtype=ROW_DT)
    n = _o().oracle_distinct_inc(_p(np.ascontiguousarray(delta, dtype=ROW_DT)),
                                 len(delta),
                                 _p(np.ascontiguousarray(trace, dtype=ROW_DT)),
                                 len(trace), _p(out))
    return out[:n].copy()
