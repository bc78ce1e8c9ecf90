"""dbsp_amd — Python driver for the MI355X-native DBSP hot path.

Thin ctypes layer over the C-ABI declared in include/dbsp_hip.h (the FFI line
beneath operator eval; see that header and INTEGRATION.md).  Submodules:

  dbsp_amd.gen     deterministic Nexmark event generator (host, CPU)
  dbsp_amd.engine  GPU engine (product path; requires the HIP extension + a GPU,
                   fails loudly otherwise — never falls back to CPU)
  dbsp_amd.oracle  CPU oracle bindings (TEST INFRASTRUCTURE + bench cpu_baseline
                   only; the product path never imports this)
"""
import ctypes
import os
from pathlib import Path

import numpy as np

PKG_DIR = Path(__file__).resolve().parents[2]   # database-stream-processor_amd/
REPO_DIR = PKG_DIR.parent

# numpy mirrors of the C structs in include/dbsp_hip.h (all-8-byte fields: no padding)
EVENT_DT = np.dtype([
    ("kind", "<u8"), ("f0", "<u8"), ("f1", "<u8"), ("f2", "<u8"),
    ("f3", "<u8"), ("f4", "<u8"), ("w", "<i8"),
])
ROW_DT = np.dtype([("k", "<u8"), ("v", "<u8"), ("w", "<i8")])

KIND_PERSON, KIND_AUCTION, KIND_BID = 0, 1, 2


def _load(path: Path) -> ctypes.CDLL:
    if not path.exists():
        raise FileNotFoundError(
            f"{path} not built. Run `make -C {REPO_DIR}` (or python -c "
            f"'import __graft_entry__; __graft_entry__.build()')")
    return ctypes.CDLL(str(path), mode=ctypes.RTLD_GLOBAL)


def load_gen_lib() -> ctypes.CDLL:
    return _load(PKG_DIR / "libdbsp_gen.so")


def load_hip_lib() -> ctypes.CDLL:
    """The product HIP extension.  Raises if absent — no silent fallback."""
    return _load(PKG_DIR / "libdbsp_hip.so")


def load_oracle_lib() -> ctypes.CDLL:
    # test infrastructure only — see oracle/zset_oracle.cpp header
    return _load(REPO_DIR / "oracle" / "liboracle_dbsp.so")


def rows(k, v, w) -> np.ndarray:
    out = np.empty(len(k), dtype=ROW_DT)
    out["k"], out["v"], out["w"] = k, v, w
    return out


def rows_from_list(triples) -> np.ndarray:
    out = np.empty(len(triples), dtype=ROW_DT)
    for i, (k, v, w) in enumerate(triples):
        out[i] = (k, v, w)
    return out


def sort_rows(r: np.ndarray) -> np.ndarray:
    """Canonical (k, v) ordering for comparisons in tests."""
    return np.sort(r, order=["k", "v"])
