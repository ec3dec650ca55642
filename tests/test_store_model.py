"""Property-based differential test of the oracle MEMTABLE semantics.

The oracle is the parity anchor for every GPU test, so its store semantics
get their own independent restatement here: a ~40-line Python event-log
model of memtable visibility (point history chains, range tombstones as a
seq floor, merge folding) checked against oracle/wb_oracle.c `orc_apply`/
`orc_get` on hypothesis-generated batch streams. Semantics restated from
rocksdb 5.7.fb memtable read rules (un-vendored dep):

  - follower assigns seqs: base = latest_seq+1 per batch (orc_apply,
    rocksdb_assumption_test.cpp:136-187)
  - Get(k): newest->oldest point history, stopping at the max seq of any
    range tombstone covering k (begin <= k < end bytewise); MERGE records
    collect operands; VALUE is the base; DELETION/SINGLE_DELETION stop
    with no base
  - fold: CONCAT joins base+operands oldest->newest with ','; U64ADD sums
    little-endian u64 of the first min(len,8) value bytes
"""
import pytest

from hypothesis import given, settings, strategies as st

import oracle_ffi
from pywb import PyBatch

KEYS = [bytes([b]) * n for b in (0x10, 0x40, 0x41, 0x80, 0xFF) for n in (1, 3)]

op = st.one_of(
    st.tuples(st.just("put"), st.sampled_from(KEYS), st.binary(max_size=24)),
    st.tuples(st.just("delete"), st.sampled_from(KEYS), st.just(b"")),
    st.tuples(st.just("single_delete"), st.sampled_from(KEYS), st.just(b"")),
    st.tuples(st.just("merge"), st.sampled_from(KEYS), st.binary(max_size=12)),
    st.tuples(st.just("delete_range"), st.sampled_from(KEYS),
              st.sampled_from(KEYS)),
)


class ModelStore:
    """Event-log restatement: no hash table, no arena — just the rules."""

    def __init__(self, merge_op):
        self.merge_op = merge_op
        self.points = []  # (seq, kind, key, val) newest last
        self.tombs = []   # (seq, begin, end)
        self.latest = 0

    @staticmethod
    def _cfkey(cf, k):
        import struct as _s
        return _s.pack("<I", cf) + k

    def apply(self, ops):
        """ops: (kind, a, b) with plain kinds, or cf kinds where a =
        (cf_id, key) / (cf_id, begin, end) — cf-namespaced like the
        engine/oracle: stored key = [cf LE4 | key], range tombstones
        prefix BOTH bounds."""
        seq = self.latest + 1
        for kind, a, b in ops:
            if kind == "log_data":  # WAL-only marker: no seq, no memtable
                continue
            if kind == "delete_range":
                self.tombs.append((seq, a, b))
            elif kind == "cf_delete_range":
                cf, bk, ek = a
                self.tombs.append((seq, self._cfkey(cf, bk),
                                   self._cfkey(cf, ek)))
            elif kind.startswith("cf_"):
                cf, k = a
                self.points.append((seq, kind[3:], self._cfkey(cf, k), b))
            else:
                self.points.append((seq, kind, a, b))
            seq += 1
        self.latest = seq - 1

    def get(self, k):
        floor = max((s for s, b, e in self.tombs if b <= k < e), default=0)
        operands, base = [], None
        for s, kind, key, val in reversed(self.points):
            if key != k or s <= floor:
                continue
            if kind == "merge":
                operands.append(val)
                continue
            if kind == "put":
                base = val
            break  # put/delete/single_delete all stop the walk
        if base is None and not operands:
            return None
        if not operands:
            return base  # a plain Put reads back verbatim (rocksdb: the
            # merge operator only runs when operands are newer)
        if self.merge_op == oracle_ffi.Store.MERGE_U64ADD:
            acc = sum(int.from_bytes(v[:8], "little")
                      for v in ([base] if base is not None else []) + operands)
            return (acc % 2**64).to_bytes(8, "little")
        parts = ([base] if base is not None else []) + operands[::-1]
        return b",".join(parts)


@pytest.fixture(scope="module")
def lib():
    return oracle_ffi.load()


@settings(max_examples=150, deadline=None, derandomize=True)
@given(batches=st.lists(st.lists(op, min_size=1, max_size=6),
                        min_size=1, max_size=8),
       merge_op=st.sampled_from([0, 1]))
def test_oracle_store_matches_model(batches, merge_op):
    lib = oracle_ffi.load()
    ost = oracle_ffi.Store(lib, 1, merge_op=merge_op)
    model = ModelStore(merge_op)
    for ops in batches:
        b = PyBatch()
        for kind, a, v in ops:
            if kind in ("put", "merge"):
                getattr(b, kind)(a, v)
            elif kind == "delete_range":
                b.delete_range(a, v)
            else:
                getattr(b, kind)(a)
        assert ost.apply(0, b.data())
        model.apply(ops)
    assert ost.latest_seq(0) == model.latest
    for k in KEYS + [b"\x00", b"absent"]:
        assert ost.get(0, k) == model.get(k), k
