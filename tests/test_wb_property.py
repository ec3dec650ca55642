"""Property-based differential test of the WriteBatch rep layout.

Three independent restatements of the rocksdb 5.7.fb format (db/
write_batch.cc, un-vendored) are cross-checked on
hypothesis-generated op sequences:
  - product builder (rocksplicator_amd.Batch, csrc/builder.cpp)
  - pure-Python builder (tests/pywb.PyBatch)
  - oracle decoder (oracle/wb_oracle.c orc_decode)
Checked: byte-identical encodings, decode round-trip of every slice, and
the seq-consumption rule (LogData/Noop/2PC markers consume no seq;
rocksdb_assumption_test.cpp:136-187 is the reference's statement of it).
"""
import pytest

from hypothesis import given, settings, strategies as st

import oracle_ffi
from pywb import PyBatch

slices = st.binary(min_size=0, max_size=300)
keys = st.binary(min_size=1, max_size=80)

op = st.one_of(
    st.tuples(st.just("put"), keys, slices),
    st.tuples(st.just("delete"), keys, st.just(b"")),
    st.tuples(st.just("single_delete"), keys, st.just(b"")),
    st.tuples(st.just("merge"), keys, slices),
    st.tuples(st.just("delete_range"), keys, keys),
    st.tuples(st.just("log_data"), slices, st.just(b"")),
)

cf_ids = st.integers(min_value=1, max_value=2**32 - 1)

CONSUMES = {"put", "delete", "single_delete", "merge", "delete_range",
            "cf_put", "cf_delete", "cf_single_delete", "cf_merge",
            "cf_delete_range"}


@pytest.fixture(scope="module")
def lib():
    return oracle_ffi.load()


@settings(max_examples=250, deadline=None, derandomize=True)
@given(ops=st.lists(op, min_size=0, max_size=12),
       seq=st.integers(min_value=0, max_value=2**64 - 1))
def test_builders_agree_and_oracle_decodes(ops, seq):
    import rocksplicator_amd as ra
    lib = oracle_ffi.load()

    pb, gb = PyBatch(seq=seq), ra.Batch().set_seq(seq)
    for kind, a, b in ops:
        if kind in ("put", "merge"):
            getattr(pb, kind)(a, b)
            getattr(gb, kind)(a, b)
        elif kind in ("delete", "single_delete"):
            getattr(pb, kind)(a)
            getattr(gb, kind)(a)
        elif kind == "delete_range":
            pb.delete_range(a, b)
            gb.delete_range(a, b)
        else:
            pb.log_data(a)
            gb.log_data(a)
    rep_py, rep_c = pb.data(), gb.data()
    assert rep_py == rep_c  # two builders, identical bytes

    got_seq, cnt, recs = oracle_ffi.decode(lib, rep_c)
    assert got_seq == seq
    assert cnt == sum(1 for k, _, _ in ops if k in CONSUMES)
    assert len(recs) == len(ops)

    next_seq = seq
    for (kind, a, b), r in zip(ops, recs):
        consumes = kind in CONSUMES
        assert bool(r.consumes_seq) == consumes
        if consumes:
            # record i gets base+i over consumers (uint64 wrap at 2^64)
            assert r.seq == next_seq % 2**64
            next_seq += 1
        key = rep_c[r.key_off:r.key_off + r.key_len]
        val = rep_c[r.val_off:r.val_off + r.val_len]
        if kind == "log_data":
            assert val == a and r.key_len == 0  # blob rides in the val slot
        elif kind == "delete_range":
            assert (key, val) == (a, b)  # begin/end keys
        elif kind in ("put", "merge"):
            assert (key, val) == (a, b)
        else:
            assert key == a and r.val_len == 0


@settings(max_examples=120, deadline=None, derandomize=True)
@given(ops=st.lists(st.tuples(cf_ids, keys, slices), min_size=1, max_size=6),
       seq=st.integers(min_value=0, max_value=2**48))
def test_cf_builders_agree_all_variants(ops, seq):
    """All five CF record kinds through BOTH builders (product gra_wb_cf_*
    vs pywb), byte-identical, and decoded back by the oracle with the
    cf ids and slices intact."""
    import rocksplicator_amd as ra
    lib = oracle_ffi.load()
    pb, gb = PyBatch(seq=seq), ra.Batch().set_seq(seq)
    for cf, k, v in ops:
        for b in (pb, gb):
            b.cf_put(cf, k, v)
            b.cf_delete(cf, k)
            b.cf_single_delete(cf, k)
            b.cf_merge(cf, k, v)
            b.cf_delete_range(cf, k, k + b"\xff")
    rep_py, rep_c = pb.data(), gb.data()
    assert rep_py == rep_c
    got_seq, cnt, recs = oracle_ffi.decode(lib, rep_c)
    assert got_seq == seq and cnt == 5 * len(ops)
    for i, (cf, k, v) in enumerate(ops):
        for j in range(5):
            r = recs[5 * i + j]
            assert r.cf_id == cf
            assert rep_c[r.key_off:r.key_off + r.key_len] == k


@settings(max_examples=120, deadline=None, derandomize=True)
@given(ops=st.lists(
    st.tuples(st.integers(min_value=0, max_value=2**32 - 1), keys, slices),
    min_size=1, max_size=8))
def test_cf_variants_decode(ops):
    """CF-prefixed tags (varint cf_id first) through pywb -> oracle; the
    apply path treats unknown CFs as corruption, but the decoder must still
    parse the layout (rep framing is CF-agnostic)."""
    lib = oracle_ffi.load()
    pb = PyBatch(seq=7)
    for cf, k, v in ops:
        pb.cf_put(cf, k, v)
        pb.cf_delete(cf, k)
        pb.cf_merge(cf, k, v)
    rep = pb.data()
    got_seq, cnt, recs = oracle_ffi.decode(lib, rep)
    assert got_seq == 7 and cnt == 3 * len(ops) and len(recs) == 3 * len(ops)
    for i, (cf, k, v) in enumerate(ops):
        for j, want_val in ((0, v), (1, b""), (2, v)):
            r = recs[3 * i + j]
            assert r.cf_id == cf
            assert rep[r.key_off:r.key_off + r.key_len] == k
            assert rep[r.val_off:r.val_off + r.val_len] == want_val


@settings(max_examples=200, deadline=None, derandomize=True)
@given(data=st.data())
def test_corrupt_reps_rejected_cleanly(data):
    """Mutation fuzz: random byte flips/truncations of a valid rep must
    never crash the oracle decoder/applier, and a rejected batch must not
    advance the store's seq (all-or-nothing apply, two-pass validation)."""
    lib = oracle_ffi.load()
    ops = data.draw(st.lists(op, min_size=1, max_size=6))
    b = PyBatch(seq=0)
    for kind, a, v in ops:
        if kind in ("put", "merge"):
            getattr(b, kind)(a, v)
        elif kind == "delete_range":
            b.delete_range(a, v)
        elif kind == "log_data":
            b.log_data(a)
        else:
            getattr(b, kind)(a)
    rep = bytearray(b.data())
    mode = data.draw(st.sampled_from(["flip", "truncate", "extend"]))
    if mode == "flip":
        i = data.draw(st.integers(0, len(rep) - 1))
        rep[i] ^= data.draw(st.integers(1, 255))
    elif mode == "truncate":
        rep = rep[:data.draw(st.integers(0, len(rep) - 1))]
    else:
        rep += bytes(data.draw(st.integers(1, 8)))
    rep = bytes(rep)

    try:
        seq, cnt, recs = oracle_ffi.decode(lib, rep)
        decoded_ok = True
    except ValueError:
        decoded_ok = False

    st_ = oracle_ffi.Store(lib, 1)
    before = st_.latest_seq(0)
    applied = st_.apply(0, rep)
    after = st_.latest_seq(0)
    if applied:
        assert decoded_ok  # apply is gated on the same two-pass decode
        consumed = sum(1 for r in recs if r.consumes_seq)
        assert after == before + consumed
    else:
        assert after == before  # rejected batch applies nothing


@settings(max_examples=200, deadline=None, derandomize=True)
@given(blob=st.binary(max_size=600))
def test_snappy_decoder_rejects_garbage(blob):
    """Arbitrary bytes into both snappy decoders: bounded, no crash, and
    the two independent restatements agree on accept/reject AND on the
    decoded bytes when they accept."""
    import ctypes as CT
    import rocksplicator_amd as ra
    lib = oracle_ffi.load()
    plib = ra.load()
    lib.orc_snappy_decompress.argtypes = [
        CT.c_char_p, CT.c_size_t, CT.c_char_p, CT.c_size_t,
        CT.POINTER(CT.c_size_t)]
    cap = 700
    obuf = CT.create_string_buffer(cap + 16)
    olen = CT.c_size_t()
    orc = lib.orc_snappy_decompress(blob, len(blob), obuf, cap, CT.byref(olen))
    pbuf = CT.create_string_buffer(cap + 16)
    plen = plib.gra_snappy_decompress(blob, len(blob), pbuf, cap)
    assert (orc == 0) == (plen != 0xFFFFFFFF), (orc, plen)
    if orc == 0:
        assert olen.value == plen
        assert obuf.raw[:olen.value] == pbuf.raw[:plen]
