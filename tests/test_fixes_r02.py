"""Round-2 regression tests for the advisor findings:

1. (high) Snappy literal-length 32-bit overflow: a 4-extra-byte literal
   length near 2^32 made ip+len / op+len wrap in uint32 so both bounds
   checks passed and the copy loop wrote ~4 GB out of bounds. Product and
   oracle codecs must BOTH reject the stream (they diverged before the fix).
2. (low) CF DeleteRange key namespacing: a cf!=0 range tombstone must cover
   keys of that cf only — begin AND end keys are cf-prefixed consistently
   in engine and oracle (before the fix the oracle prefixed neither, the
   engine prefixed begin only).

CPU-only: product codec via the libgra.so host export, semantics via the
oracle. GPU-side parity for the same streams lives in test_gpu_parity.py.
"""
import ctypes as C
import os
import struct
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import rocksplicator_amd as ra  # noqa: E402

import oracle_ffi  # noqa: E402
from pywb import PyBatch, varint32  # noqa: E402


@pytest.fixture(scope="module")
def glib():
    return ra.load()


@pytest.fixture(scope="module")
def olib():
    return oracle_ffi.load()


def _pylit_overflow_stream():
    """varint(ulen=64), 1-byte literal 'A' (so op>0, ip>0), then a literal
    tag with 4 extra length bytes encoding len-1 = 0xFFFFFFFE
    -> len = 0xFFFFFFFF. In uint32: ip+len and op+len both wrap below
    slen/ulen, passing the old checks."""
    body = bytes([64])                      # varint32 ulen = 64
    body += bytes([0 << 2]) + b"A"          # literal len 1
    body += bytes([63 << 2])                # literal, 4 extra length bytes
    body += b"\xfe\xff\xff\xff"             # len-1 = 0xFFFFFFFE
    body += b"B" * 8                        # a few bytes of "data"
    return body


def test_snappy_literal_overflow_rejected_product(glib):
    s = _pylit_overflow_stream()
    dst = C.create_string_buffer(64 + 32)
    r = glib.gra_snappy_decompress(s, len(s), dst, 64)
    assert r == 0xFFFFFFFF, "product codec must reject the overflow stream"


def test_snappy_literal_overflow_rejected_oracle(olib):
    s = _pylit_overflow_stream()
    dst = C.create_string_buffer(64 + 32)
    dlen = C.c_size_t()
    rc = olib.orc_snappy_decompress(s, len(s), dst, 64, C.byref(dlen))
    assert rc != 0, "oracle codec must reject the overflow stream"


def test_snappy_roundtrip_still_works(glib, olib):
    payload = (b"abcdefgh" * 100) + os.urandom(64) + b"\x00" * 200
    comp = C.create_string_buffer(len(payload) * 2 + 64)
    clen = glib.gra_snappy_compress(payload, len(payload), comp, len(payload) * 2 + 64)
    assert clen > 0
    out = C.create_string_buffer(len(payload) + 32)
    r = glib.gra_snappy_decompress(comp.raw[:clen], clen, out, len(payload))
    assert r == len(payload) and out.raw[:r] == payload
    # cross-codec: oracle decompresses the product's stream
    out2 = C.create_string_buffer(len(payload) + 32)
    dlen = C.c_size_t()
    rc = olib.orc_snappy_decompress(comp.raw[:clen], clen, out2,
                                    len(payload), C.byref(dlen))
    assert rc == 0 and out2.raw[:dlen.value] == payload


def _cfkey(cf, k):
    return struct.pack("<I", cf) + k


def test_cf_range_delete_scoped_to_cf_oracle(olib):
    st = oracle_ffi.Store(olib, 1)
    rep = (PyBatch()
           .put(b"bb", b"v0")                 # cf 0
           .cf_put(1, b"bb", b"v1")           # cf 1
           .cf_put(2, b"bb", b"v2")           # cf 2
           .cf_delete_range(1, b"aa", b"cc")  # covers cf-1 [aa, cc) ONLY
           .data())
    assert st.apply(0, rep)
    assert st.get(0, b"bb") == b"v0", "cf-0 key must survive a cf-1 range delete"
    assert st.get(0, _cfkey(1, b"bb")) is None, "cf-1 key inside range is deleted"
    assert st.get(0, _cfkey(2, b"bb")) == b"v2", "cf-2 key must survive"


def test_cf_range_delete_bounds_oracle(olib):
    st = oracle_ffi.Store(olib, 1)
    rep = (PyBatch()
           .cf_put(1, b"aa", b"x").cf_put(1, b"cc", b"y").cf_put(1, b"b", b"z")
           .cf_delete_range(1, b"aa", b"cc")
           .data())
    assert st.apply(0, rep)
    assert st.get(0, _cfkey(1, b"aa")) is None   # begin inclusive
    assert st.get(0, _cfkey(1, b"b")) is None    # interior
    assert st.get(0, _cfkey(1, b"cc")) == b"y"   # end exclusive


def test_cf_range_delete_checksum_stable(olib):
    """Checksum folds the prefixed representation on both sides; two stores
    applying the same stream must agree (guards the arena layout change)."""
    rep = (PyBatch().cf_put(3, b"k1", b"v").cf_delete_range(3, b"k0", b"k2")
           .put(b"p", b"q").data())
    sums = []
    for _ in range(2):
        st = oracle_ffi.Store(olib, 1)
        assert st.apply(0, rep)
        sums.append(olib.orc_shard_checksum(st.h, 0))
    assert sums[0] == sums[1] != 0
