"""Multi-worker sharding semantics on CPU (gloo, world_size=2).

The reference tests multi-worker runs by re-running a scenario at several
worker counts in one process and asserting gathered outputs are identical
(reference operator/join.rs:1019-1033,1339-1385).  The MI355X engine's
analog — xxh3 % world key partitioning + all-to-all exchange + per-rank
incremental evaluation (shard.rs:88-199 -> RCCL alltoallv) — is validated
here without a GPU: two processes partition the per-tick flat-mapped deltas
by the reference hash, exchange them over gloo send/recv (mirroring the
RCCL column exchange in engine.cpp::alltoallv_cols), run the per-rank
incremental q3 pipeline with oracle primitives, and the union of rank
outputs must equal the single-worker oracle.
"""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

import conftest  # noqa: F401  (sys.path setup)
from dbsp_amd import ROW_DT, EVENT_DT

SEED = 0x7F95EF85BE33C337
WORLD = 2


def _partition(rows, world, xxh3):
    shards = [[] for _ in range(world)]
    for r in rows:
        shards[xxh3(int(r["k"])) % world].append(r)
    return [np.array(s, dtype=ROW_DT) if s else np.empty(0, dtype=ROW_DT)
            for s in shards]


def _exchange(shards, rank, world):
    """gloo all-to-all of row columns via send/recv pairs (the gloo mirror of
    engine.cpp alltoallv_cols: counts first, then the three columns)."""
    recv = []
    for peer in range(world):
        if peer == rank:
            recv.append(shards[rank])
            continue
        send_n = torch.tensor([len(shards[peer])], dtype=torch.int64)
        recv_n = torch.zeros(1, dtype=torch.int64)
        if rank < peer:
            dist.send(send_n, peer)
            dist.recv(recv_n, peer)
        else:
            dist.recv(recv_n, peer)
            dist.send(send_n, peer)
        buf_send = torch.from_numpy(
            np.ascontiguousarray(shards[peer].view(np.uint8)))
        buf_recv = torch.zeros(int(recv_n.item()) * ROW_DT.itemsize,
                               dtype=torch.uint8)
        if rank < peer:
            if len(buf_send):
                dist.send(buf_send, peer)
            if len(buf_recv):
                dist.recv(buf_recv, peer)
        else:
            if len(buf_recv):
                dist.recv(buf_recv, peer)
            if len(buf_send):
                dist.send(buf_send, peer)
        recv.append(buf_recv.numpy().view(ROW_DT))
    return np.concatenate(recv) if recv else np.empty(0, dtype=ROW_DT)


def _worker(rank, world, events_by_tick, out_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = os.environ.get("DBSP_TEST_PORT", "29511")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from dbsp_amd import oracle

    a_int = np.empty(0, dtype=ROW_DT)
    p_int = np.empty(0, dtype=ROW_DT)
    all_out = []
    for evs in events_by_tick:
        # each rank ingests a disjoint slice of the tick's events (weak scaling)
        mine = evs[rank::world]
        dA, dP = [], []
        for e in mine:
            if e["kind"] == 1 and e["f2"] == 10:
                dA.append((e["f1"], e["f0"], e["w"]))
            if e["kind"] == 0 and e["f3"] in (1, 2, 3):
                dP.append((e["f0"],
                           (int(e["f1"]) << 8) | (int(e["f2"]) << 4) | int(e["f3"]),
                           e["w"]))
        dA = np.array(dA, dtype=ROW_DT) if dA else np.empty(0, dtype=ROW_DT)
        dP = np.array(dP, dtype=ROW_DT) if dP else np.empty(0, dtype=ROW_DT)
        # shard + exchange (co-locate join keys), then consolidate
        dA = oracle.consolidate(_exchange(_partition(dA, world, oracle.xxh3_u64),
                                          rank, world))
        dP = oracle.consolidate(_exchange(_partition(dP, world, oracle.xxh3_u64),
                                          rank, world))
        out1 = oracle.join_raw(dA, p_int, 0)          # dA x P_prev
        out2 = oracle.join_raw(dP, a_int, 1)          # A_prev x dP
        out3 = oracle.join_raw(dA, dP, 0)             # dA x dP
        a_int = oracle.merge(a_int, dA)
        p_int = oracle.merge(p_int, dP)
        out = oracle.consolidate(np.concatenate([out1, out2, out3]))
        all_out.append(out)
    dist.barrier()
    out_q.put((rank, [o.tobytes() for o in all_out]))
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_sharded_q3_equals_single_worker():
    from dbsp_amd import gen, oracle
    events = gen.generate(20_000, seed=5)
    ticks = [events[i:i + 4000] for i in range(0, 20_000, 4000)]

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, WORLD, ticks, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, outs = q.get(timeout=150)
        results[rank] = [np.frombuffer(o, dtype=ROW_DT) for o in outs]
    for p in procs:
        p.join(timeout=30)

    # single-worker oracle on the same streams
    ref = oracle.Query(3)
    for t, evs in enumerate(ticks):
        expected = ref.step(evs)
        union = np.concatenate([results[r][t] for r in range(WORLD)])
        union = oracle.consolidate(union)
        exp = oracle.consolidate(expected)
        assert np.array_equal(np.sort(union, order=["k", "v"]),
                              np.sort(exp, order=["k", "v"])), f"tick {t}"
