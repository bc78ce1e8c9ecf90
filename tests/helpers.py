"""Shared test helpers: event construction, string interning, Z-set compare."""
import json
from pathlib import Path

import numpy as np

from dbsp_amd import EVENT_DT, ROW_DT, KIND_PERSON, KIND_AUCTION, KIND_BID

GOLDEN = Path(__file__).resolve().parent / "golden"

# Dictionary ids (mirror the generator dictionaries; see nexmark_gen.hpp):
# US_STATES (reference generator/people.rs:18-25) and US_CITIES (:27-38).
STATE_IDS = {"AZ": 0, "CA": 1, "ID": 2, "OR": 3, "WA": 4, "WY": 5}
CITY_IDS = {"Phoenix": 0, "Los Angeles": 1, "San Francisco": 2, "Boise": 3,
            "Portland": 4, "Bend": 5, "Redmond": 6, "Seattle": 7, "Kent": 8,
            "Cheyenne": 9}


class Intern:
    """Test-local string -> id dictionary (Z-set equality is intern invariant)."""

    def __init__(self, start=1):
        self.map = {}
        self.next = start

    def __call__(self, s):
        if s not in self.map:
            self.map[s] = self.next
            self.next += 1
        return self.map[s]


def state_id(s, intern=None):
    if s in STATE_IDS:
        return STATE_IDS[s]
    # out-of-dictionary states (e.g. "NL" in the q3 golden test) map into the
    # unused 4-bit id range 6..15 — must only be distinct from CA/ID/OR
    return 6 + ((intern(s) if intern else hash(s)) % 10)


def person_event(pid, name_id, city_id, sid, dt=0, w=1):
    e = np.zeros((), dtype=EVENT_DT)
    e["kind"], e["f0"], e["f1"], e["f2"], e["f3"], e["f4"], e["w"] = (
        KIND_PERSON, pid, name_id, city_id, sid, dt, w)
    return e


def auction_event(aid, seller, category, dt=0, expires=2000, w=1):
    e = np.zeros((), dtype=EVENT_DT)
    e["kind"], e["f0"], e["f1"], e["f2"], e["f3"], e["f4"], e["w"] = (
        KIND_AUCTION, aid, seller, category, dt, expires, w)
    return e


def bid_event(auction, dt, bidder=1, price=99, w=1):
    e = np.zeros((), dtype=EVENT_DT)
    e["kind"], e["f0"], e["f1"], e["f2"], e["f3"], e["f4"], e["w"] = (
        KIND_BID, auction, bidder, price, dt, 0, w)
    return e


def events(*evs):
    out = np.empty(len(evs), dtype=EVENT_DT)
    for i, e in enumerate(evs):
        out[i] = e
    return out


def pack_person(name_id, city_id, sid):
    # compact person tuple id: name(10b) | city(4b) | state(4b) — short keys
    # mean fewer radix digit passes (helpers mirror oracle/kernels exactly)
    return (name_id << 8) | ((city_id & 0xF) << 4) | (sid & 0xF)


def zset(rows_arr):
    """rows (structured ROW_DT array) -> {(k, v): w} with zero weights dropped."""
    d = {}
    for r in np.asarray(rows_arr, dtype=ROW_DT):
        key = (int(r["k"]), int(r["v"]))
        d[key] = d.get(key, 0) + int(r["w"])
    return {k: w for k, w in d.items() if w != 0}


def rows_of(triples):
    out = np.empty(len(triples), dtype=ROW_DT)
    mask = (1 << 64) - 1
    for i, (k, v, w) in enumerate(triples):
        out[i] = (int(k) & mask, int(v) & mask, w)
    return out


def load_golden(name):
    with open(GOLDEN / name) as f:
        return json.load(f)


def np_consolidate(rows_arr):
    """Independent numpy restatement of consolidate for model tests."""
    r = np.asarray(rows_arr, dtype=ROW_DT)
    if len(r) == 0:
        return r
    order = np.lexsort((r["v"], r["k"]))
    r = r[order]
    keys = np.stack([r["k"], r["v"]], axis=1)
    head = np.ones(len(r), dtype=bool)
    head[1:] = (keys[1:] != keys[:-1]).any(axis=1)
    seg = np.cumsum(head) - 1
    sums = np.zeros(seg[-1] + 1, dtype=np.int64)
    np.add.at(sums, seg, r["w"].astype(np.int64))
    uk = r["k"][head]
    uv = r["v"][head]
    nz = sums != 0
    out = np.empty(nz.sum(), dtype=ROW_DT)
    out["k"], out["v"], out["w"] = uk[nz], uv[nz], sums[nz]
    return out
