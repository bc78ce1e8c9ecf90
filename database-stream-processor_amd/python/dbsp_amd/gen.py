"""Deterministic Nexmark event stream (ctypes wrapper over libdbsp_gen.so).

See csrc/nexmark_gen.hpp for what is mirrored exactly from the reference
generator and what is re-seeded (generator-RNG parity is
unpinned; oracle and GPU paths consume identical streams from this generator).
"""
import ctypes

import numpy as np

from . import EVENT_DT, load_gen_lib

_lib = None


def _gen_lib():
    global _lib
    if _lib is None:
        _lib = load_gen_lib()
        _lib.dbsp_gen_events.restype = ctypes.c_int64
        _lib.dbsp_gen_events.argtypes = [
            ctypes.c_uint64, ctypes.c_uint64, ctypes.c_double,
            ctypes.c_int64, ctypes.c_void_p,
        ]
        _lib.dbsp_gen_new.restype = ctypes.c_void_p
        _lib.dbsp_gen_new.argtypes = [ctypes.c_uint64, ctypes.c_uint64, ctypes.c_double]
        _lib.dbsp_gen_free.argtypes = [ctypes.c_void_p]
        _lib.dbsp_gen_next.restype = ctypes.c_int64
        _lib.dbsp_gen_next.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64]
    return _lib


DEFAULT_BASE_TIME_MS = 10_000_000
DEFAULT_RATE = 10_000_000.0  # events/s (reference config.rs:50 default)


def generate(n, seed=1, base_time_ms=DEFAULT_BASE_TIME_MS, rate=DEFAULT_RATE):
    """Generate n events as a numpy structured array (EVENT_DT)."""
    lib = _gen_lib()
    out = np.empty(n, dtype=EVENT_DT)
    lib.dbsp_gen_events(seed, base_time_ms, rate, n,
                        out.ctypes.data_as(ctypes.c_void_p))
    return out


class Stream:
    """Chunked generation for long streams."""

    def __init__(self, seed=1, base_time_ms=DEFAULT_BASE_TIME_MS, rate=DEFAULT_RATE):
        self._lib = _gen_lib()
        self._h = self._lib.dbsp_gen_new(seed, base_time_ms, rate)

    def next(self, n):
        out = np.empty(n, dtype=EVENT_DT)
        self._lib.dbsp_gen_next(self._h, out.ctypes.data_as(ctypes.c_void_p), n)
        return out

    def close(self):
        if self._h:
            self._lib.dbsp_gen_free(self._h)
            self._h = None

    def __del__(self):
        self.close()
