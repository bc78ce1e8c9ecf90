"""RCCL exchange path on one GPU (world=1 self-exchange).

The driver's round-end scale bench runs bench.py at N=2/4/8 ranks; only one
GPU is available interactively, so this exercises the exact code path —
xxh3 partition kernel -> RCCL grouped send/recv of the count row and the
three columns -> re-consolidate (engine.cpp shard_exchange/alltoallv_cols) —
as a world=1 self-exchange, both at the C-ABI level and through the full q3
engine with DBSP_FORCE_SHARD=1, parity-checked against the oracle."""
import ctypes
import os

import numpy as np
import pytest

from dbsp_amd import ROW_DT, gen, oracle
from helpers import zset

pytestmark = pytest.mark.gpu


def _make_ctx_with_comm():
    from dbsp_amd.engine import Ctx, _L
    ctx = Ctx(0)
    L = _L()
    L.dbsp_comm_unique_id.restype = ctypes.c_int32
    nccl_id = np.zeros(128, dtype=np.uint8)
    assert L.dbsp_comm_unique_id(nccl_id.ctypes.data_as(ctypes.c_void_p)) == 0
    assert L.dbsp_comm_init(ctx._h, 0, 1,
                            nccl_id.ctypes.data_as(ctypes.c_void_p)) == 0
    return ctx, L


def test_alltoallv_self_roundtrip():
    from dbsp_amd.engine import BatchStruct
    ctx, L = _make_ctx_with_comm()
    L.dbsp_comm_alltoallv.restype = ctypes.c_int32
    rng = np.random.default_rng(3)
    rows = np.empty(10_000, dtype=ROW_DT)
    rows["k"] = rng.integers(0, 1 << 40, len(rows))
    rows["v"] = rng.integers(0, 1 << 20, len(rows))
    rows["w"] = rng.integers(-3, 4, len(rows))
    send = ctx.upload_rows(rows)
    send_counts = np.array([len(rows)], dtype=np.int64)
    recv_counts = np.zeros(1, dtype=np.int64)
    recv = BatchStruct()
    st = L.dbsp_comm_alltoallv(
        ctx._h, ctypes.byref(send),
        send_counts.ctypes.data_as(ctypes.c_void_p), ctypes.byref(recv),
        len(rows), recv_counts.ctypes.data_as(ctypes.c_void_p))
    assert st == 0
    ctx.sync()
    assert recv_counts[0] == len(rows)
    got = ctx.download_rows(recv)
    assert np.array_equal(got, rows)
    ctx.free_batch(send)
    ctx.free_batch(recv)
    ctx.close()


@pytest.mark.parametrize("query", [3, 5, 8])
def test_engine_with_forced_shard_exchange(query):
    """Full q3/q5/q8 through the partition + RCCL alltoallv + re-consolidate
    path, including the watermark allreduce and the aggregate/join reshard
    points (world=1 self-exchange), tick-by-tick parity vs the oracle."""
    os.environ["DBSP_FORCE_SHARD"] = "1"
    try:
        from dbsp_amd.engine import Engine
        ctx, _ = _make_ctx_with_comm()
        eng = Engine(ctx, query=query)
        q = oracle.Query(query)
        evs = gen.generate(50_000, seed=23)
        eng.stage(evs)
        for lo in range(0, len(evs), 10_000):
            hi = lo + 10_000
            eng.step_staged(lo, hi)
            got = eng.output()
            exp = q.step(evs[lo:hi])
            assert zset(got) == zset(exp), f"q{query} tick [{lo},{hi})"
        eng.close()
        ctx.close()
    finally:
        os.environ.pop("DBSP_FORCE_SHARD", None)
