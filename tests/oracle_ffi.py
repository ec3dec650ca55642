"""ctypes binding for oracle/libwb_oracle.so — TEST INFRASTRUCTURE ONLY.

Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may import
this module (oracle isolation rule): the oracle is the parity checker, never
the measured or shipped path.
"""
import ctypes as C
import os

_REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_LIB = os.path.join(_REPO, "oracle", "libwb_oracle.so")


class OrcRecord(C.Structure):
    _fields_ = [
        ("type", C.c_uint8),
        ("consumes_seq", C.c_uint8),
        ("cf_id", C.c_uint32),
        ("seq", C.c_uint64),
        ("key_off", C.c_uint32),
        ("key_len", C.c_uint32),
        ("val_off", C.c_uint32),
        ("val_len", C.c_uint32),
    ]


class OrcUpdateDesc(C.Structure):
    _fields_ = [
        ("shard", C.c_uint32),
        ("len", C.c_uint32),
        ("off", C.c_uint64),
        ("ts", C.c_int64),
    ]


def load():
    lib = C.CDLL(_LIB)
    lib.orc_wb_create.restype = C.c_void_p
    lib.orc_wb_destroy.argtypes = [C.c_void_p]
    lib.orc_wb_clear.argtypes = [C.c_void_p]
    for f in ("orc_wb_put", "orc_wb_merge"):
        getattr(lib, f).argtypes = [C.c_void_p, C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
    for f in ("orc_wb_delete", "orc_wb_single_delete"):
        getattr(lib, f).argtypes = [C.c_void_p, C.c_char_p, C.c_size_t]
    lib.orc_wb_delete_range.argtypes = [C.c_void_p, C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
    lib.orc_wb_put_log_data.argtypes = [C.c_void_p, C.c_char_p, C.c_size_t]
    lib.orc_wb_set_seq.argtypes = [C.c_void_p, C.c_uint64]
    lib.orc_wb_count.argtypes = [C.c_void_p]
    lib.orc_wb_count.restype = C.c_uint32
    lib.orc_wb_data.argtypes = [C.c_void_p, C.POINTER(C.c_size_t)]
    lib.orc_wb_data.restype = C.POINTER(C.c_uint8)
    lib.orc_decode.argtypes = [
        C.c_char_p, C.c_size_t, C.POINTER(OrcRecord), C.c_uint32,
        C.POINTER(C.c_uint32), C.POINTER(C.c_uint64), C.POINTER(C.c_uint32),
    ]
    lib.orc_store_create.restype = C.c_void_p
    lib.orc_store_create.argtypes = [C.c_uint32, C.c_int]
    lib.orc_store_destroy.argtypes = [C.c_void_p]
    lib.orc_apply.argtypes = [C.c_void_p, C.c_uint32, C.c_char_p, C.c_size_t, C.c_int64]
    lib.orc_latest_seq.argtypes = [C.c_void_p, C.c_uint32]
    lib.orc_latest_seq.restype = C.c_uint64
    lib.orc_get.argtypes = [
        C.c_void_p, C.c_uint32, C.c_char_p, C.c_size_t,
        C.c_char_p, C.c_size_t, C.POINTER(C.c_size_t),
    ]
    lib.orc_shard_checksum.argtypes = [C.c_void_p, C.c_uint32]
    lib.orc_shard_checksum.restype = C.c_uint64
    lib.orc_store_bytes.argtypes = [C.c_void_p]
    lib.orc_store_bytes.restype = C.c_uint64
    lib.orc_cpu_apply_bench.restype = C.c_double
    lib.orc_cpu_apply_bench.argtypes = [
        C.c_void_p, C.c_void_p, C.POINTER(OrcUpdateDesc), C.c_uint64, C.c_int,
    ]
    lib.orc_cpu_apply_bench_wal.restype = C.c_double
    lib.orc_cpu_apply_bench_wal.argtypes = [
        C.c_void_p, C.c_void_p, C.POINTER(OrcUpdateDesc), C.c_uint64, C.c_int,
    ]
    lib.orc_cpu_snappy_apply_bench.restype = C.c_double
    lib.orc_cpu_snappy_apply_bench.argtypes = [
        C.c_void_p, C.c_void_p, C.POINTER(OrcUpdateDesc), C.c_uint64, C.c_int,
    ]
    return lib


class Batch:
    """Pythonic wrapper over the oracle WriteBatch rep builder."""

    def __init__(self, lib):
        self.lib = lib
        self.h = lib.orc_wb_create()

    def __del__(self):
        if getattr(self, "h", None):
            self.lib.orc_wb_destroy(self.h)
            self.h = None

    def put(self, k, v):
        self.lib.orc_wb_put(self.h, k, len(k), v, len(v))
        return self

    def delete(self, k):
        self.lib.orc_wb_delete(self.h, k, len(k))
        return self

    def single_delete(self, k):
        self.lib.orc_wb_single_delete(self.h, k, len(k))
        return self

    def merge(self, k, v):
        self.lib.orc_wb_merge(self.h, k, len(k), v, len(v))
        return self

    def delete_range(self, bk, ek):
        self.lib.orc_wb_delete_range(self.h, bk, len(bk), ek, len(ek))
        return self

    def log_data(self, blob):
        self.lib.orc_wb_put_log_data(self.h, blob, len(blob))
        return self

    def set_seq(self, seq):
        self.lib.orc_wb_set_seq(self.h, seq)
        return self

    @property
    def count(self):
        return self.lib.orc_wb_count(self.h)

    def data(self):
        n = C.c_size_t()
        p = self.lib.orc_wb_data(self.h, C.byref(n))
        return bytes(C.cast(p, C.POINTER(C.c_uint8 * n.value)).contents)


class Store:
    MERGE_CONCAT = 0
    MERGE_U64ADD = 1

    def __init__(self, lib, nshards, merge_op=0):
        self.lib = lib
        self.h = lib.orc_store_create(nshards, merge_op)

    def __del__(self):
        if getattr(self, "h", None):
            self.lib.orc_store_destroy(self.h)
            self.h = None

    def apply(self, shard, rep, ts=0):
        return bool(self.lib.orc_apply(self.h, shard, rep, len(rep), ts))

    def latest_seq(self, shard):
        return self.lib.orc_latest_seq(self.h, shard)

    def get(self, shard, key, cap=1 << 20):
        buf = C.create_string_buffer(cap)
        vlen = C.c_size_t()
        rc = self.lib.orc_get(self.h, shard, key, len(key), buf, cap, C.byref(vlen))
        if rc == 1:
            return None
        assert rc == 0, f"orc_get rc={rc}"
        return buf.raw[: vlen.value]


def decode(lib, rep):
    """Decode rep -> (base_seq, count, [records]); raises on corruption."""
    nrec = C.c_uint32()
    seq = C.c_uint64()
    cnt = C.c_uint32()
    rc = lib.orc_decode(rep, len(rep), None, 0, C.byref(nrec), C.byref(seq), C.byref(cnt))
    if rc != 0:
        raise ValueError(f"corrupt rep rc={rc}")
    arr = (OrcRecord * nrec.value)()
    rc = lib.orc_decode(rep, len(rep), arr, nrec.value, C.byref(nrec), C.byref(seq), C.byref(cnt))
    assert rc == 0
    return seq.value, cnt.value, list(arr)
