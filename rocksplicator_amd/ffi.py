"""ctypes binding for the MI355X-native replication apply path (libgra.so).

The C-ABI is declared in include/rocksplicator_gpu.h; each entry point cites
the reference interface it replaces. The follower apply path (Engine +
handle_replicate_response) is GPU-only by design and raises loudly when no
HIP device is present.
"""
import ctypes as C
import os

_REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_LIB_PATH = os.path.join(_REPO, "rocksplicator_amd", "libgra.so")

GRA_OK = 0
GRA_NOT_FOUND = 1
GRA_BUF_TOO_SMALL = 2
GRA_ERR = -1
GRA_CORRUPT = -2
GRA_NO_GPU = -3
GRA_FULL = -4

MERGE_CONCAT = 0
MERGE_U64ADD = 1


class GraEngineOpts(C.Structure):
    _fields_ = [
        ("nshards", C.c_uint32),
        ("device", C.c_int),
        ("merge_op", C.c_int),
        ("store_ring", C.c_int),
        ("store_bytes", C.c_uint64),
        ("staging_bytes", C.c_uint64),
        ("max_wb_records", C.c_uint32),
        ("retain_log", C.c_int),
        ("log_bytes", C.c_uint64),
        ("drain_host", C.c_int),
    ]


class GraKeyRef(C.Structure):
    _fields_ = [("off", C.c_uint32), ("len", C.c_uint32)]


class GraGetResult(C.Structure):
    _fields_ = [("status", C.c_uint32), ("vlen", C.c_uint32)]


class GraDbCounters(C.Structure):
    _fields_ = [(n, C.c_uint64) for n in (
        "updates_applied", "in_bytes", "apply_failures", "updates_served",
        "out_bytes", "latency_ms_sum", "latency_samples", "latest_seq")]


class GraServedUpdate(C.Structure):
    _fields_ = [
        ("seq", C.c_uint64),
        ("ts", C.c_int64),
        ("off", C.c_uint32),
        ("len", C.c_uint32),
    ]


class GraUpdateDesc(C.Structure):
    _fields_ = [
        ("shard", C.c_uint32),
        ("len", C.c_uint32),
        ("off", C.c_uint64),
        ("ts", C.c_int64),
    ]


class GraStats(C.Structure):
    _fields_ = [
        ("h2d_ms", C.c_double),
        ("snappy_ms", C.c_double),
        ("decode_ms", C.c_double),
        ("scan_ms", C.c_double),
        ("emit_ms", C.c_double),
        ("copy_ms", C.c_double),
        ("runfix_ms", C.c_double),
        ("total_ms", C.c_double),
        ("ticks", C.c_uint64),
        ("updates", C.c_uint64),
        ("records", C.c_uint64),
        ("blob_bytes", C.c_uint64),
        ("payload_bytes", C.c_uint64),
    ]


class GraGenOpts(C.Structure):
    _fields_ = [
        ("nshards", C.c_uint32),
        ("key_len", C.c_uint32),
        ("val_len", C.c_uint32),
        ("kind", C.c_uint32),
        ("key_space", C.c_uint64),
        ("zipf_s", C.c_double),
        ("seed", C.c_uint64),
        ("compressible", C.c_uint32),
        ("_pad", C.c_uint32),
    ]


_lib = None


def load():
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_LIB_PATH):
        raise RuntimeError(
            f"{_LIB_PATH} not built — run `make` (hipcc --offload-arch=gfx950)")
    lib = C.CDLL(_LIB_PATH)
    lib.gra_last_error.restype = C.c_char_p
    lib.gra_engine_opts_init.argtypes = [C.POINTER(GraEngineOpts)]
    lib.gra_engine_create.argtypes = [C.POINTER(GraEngineOpts), C.POINTER(C.c_void_p)]
    lib.gra_engine_destroy.argtypes = [C.c_void_p]
    lib.gra_open.restype = C.c_void_p
    lib.gra_open.argtypes = [C.c_void_p, C.c_uint32]
    lib.gra_close.argtypes = [C.c_void_p]
    lib.gra_handle_replicate_response.argtypes = [C.c_void_p, C.c_char_p, C.c_size_t, C.c_int64]
    lib.gra_latest_seq.argtypes = [C.c_void_p]
    lib.gra_latest_seq.restype = C.c_uint64
    lib.gra_write_leader.argtypes = [C.c_void_p, C.c_char_p, C.c_size_t, C.POINTER(C.c_uint64)]
    lib.gra_get.argtypes = [C.c_void_p, C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t, C.POINTER(C.c_size_t)]
    lib.gra_flush.argtypes = [C.c_void_p]
    lib.gra_get_updates.argtypes = [C.c_void_p, C.c_uint64, C.c_uint32,
                                    C.POINTER(GraServedUpdate),
                                    C.POINTER(C.c_uint32), C.c_char_p,
                                    C.c_size_t, C.c_int]
    lib.gra_wait_ack.argtypes = [C.c_void_p, C.c_uint64, C.c_int, C.c_int]
    lib.gra_db_counters.argtypes = [C.c_void_p, C.POINTER(GraDbCounters)]
    lib.gra_multiget.argtypes = [C.c_void_p, C.c_uint32, C.POINTER(GraKeyRef),
                                 C.c_char_p, C.c_size_t, C.c_char_p,
                                 C.c_uint32, C.POINTER(GraGetResult)]
    lib.gra_shard_checksum.argtypes = [C.c_void_p, C.POINTER(C.c_uint64)]
    lib.gra_pin_alloc.argtypes = [C.c_void_p, C.c_size_t, C.POINTER(C.POINTER(C.c_uint8))]
    lib.gra_pin_free.argtypes = [C.c_void_p, C.POINTER(C.c_uint8)]
    lib.gra_upload.argtypes = [C.c_void_p, C.POINTER(C.c_uint8), C.c_size_t, C.POINTER(GraUpdateDesc), C.c_uint64, C.POINTER(C.c_void_p)]
    lib.gra_replay_destroy.argtypes = [C.c_void_p]
    lib.gra_replay_tick.argtypes = [C.c_void_p, C.c_uint64, C.c_uint64]
    lib.gra_replay_prepare.argtypes = [C.c_void_p, C.c_uint64, C.c_uint64]
    lib.gra_drain_prewarm.argtypes = [C.c_void_p, C.c_size_t, C.c_uint32]
    lib.gra_replay_tick_h2d.argtypes = [C.c_void_p, C.c_uint64, C.c_uint64]
    lib.gra_replay_sync.argtypes = [C.c_void_p]
    lib.gra_stats.argtypes = [C.c_void_p, C.POINTER(GraStats)]
    lib.gra_stats_reset.argtypes = [C.c_void_p]
    # builder
    lib.gra_wb_create.restype = C.c_void_p
    lib.gra_wb_destroy.argtypes = [C.c_void_p]
    lib.gra_wb_clear.argtypes = [C.c_void_p]
    for f in ("gra_wb_put", "gra_wb_merge"):
        getattr(lib, f).argtypes = [C.c_void_p, C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
    for f in ("gra_wb_delete", "gra_wb_single_delete"):
        getattr(lib, f).argtypes = [C.c_void_p, C.c_char_p, C.c_size_t]
    lib.gra_wb_delete_range.argtypes = [C.c_void_p, C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
    lib.gra_wb_cf_put.argtypes = [C.c_void_p, C.c_uint32, C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
    lib.gra_wb_cf_delete.argtypes = [C.c_void_p, C.c_uint32, C.c_char_p, C.c_size_t]
    lib.gra_wb_cf_single_delete.argtypes = [C.c_void_p, C.c_uint32, C.c_char_p, C.c_size_t]
    lib.gra_wb_cf_merge.argtypes = [C.c_void_p, C.c_uint32, C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
    lib.gra_wb_cf_delete_range.argtypes = [C.c_void_p, C.c_uint32, C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
    lib.gra_wb_put_log_data.argtypes = [C.c_void_p, C.c_char_p, C.c_size_t]
    lib.gra_wb_set_seq.argtypes = [C.c_void_p, C.c_uint64]
    lib.gra_wb_count.argtypes = [C.c_void_p]
    lib.gra_wb_count.restype = C.c_uint32
    lib.gra_wb_data.argtypes = [C.c_void_p, C.POINTER(C.c_size_t)]
    lib.gra_wb_data.restype = C.POINTER(C.c_uint8)
    lib.gra_gen_stream.argtypes = [C.POINTER(GraGenOpts), C.c_uint64, C.POINTER(C.c_uint8), C.c_size_t, C.POINTER(C.c_size_t), C.POINTER(GraUpdateDesc), C.c_int64]
    lib.gra_upload_snappy.argtypes = [C.c_void_p, C.POINTER(C.c_uint8), C.c_size_t, C.POINTER(GraUpdateDesc), C.c_uint64, C.POINTER(C.c_uint32), C.POINTER(C.c_uint32), C.POINTER(C.c_void_p)]
    lib.gra_snappy_compress.restype = C.c_uint32
    lib.gra_snappy_compress.argtypes = [C.c_char_p, C.c_uint32, C.c_char_p, C.c_uint32]
    lib.gra_snappy_decompress.restype = C.c_uint32
    lib.gra_snappy_decompress.argtypes = [C.c_char_p, C.c_uint32, C.c_char_p, C.c_uint32]
    lib.gra_snappy_compress_stream.argtypes = [
        C.POINTER(C.c_uint8), C.POINTER(GraUpdateDesc), C.c_uint64,
        C.POINTER(C.c_uint8), C.c_size_t, C.POINTER(C.c_size_t),
        C.POINTER(GraUpdateDesc), C.POINTER(C.c_uint32), C.c_int]
    _lib = lib
    return lib


def last_error(lib):
    return lib.gra_last_error().decode()


class Batch:
    """WriteBatch rep builder (product side)."""

    def __init__(self, lib=None):
        self.lib = lib or load()
        self.h = self.lib.gra_wb_create()

    def __del__(self):
        if getattr(self, "h", None):
            self.lib.gra_wb_destroy(self.h)
            self.h = None

    def put(self, k, v):
        self.lib.gra_wb_put(self.h, k, len(k), v, len(v))
        return self

    def delete(self, k):
        self.lib.gra_wb_delete(self.h, k, len(k))
        return self

    def single_delete(self, k):
        self.lib.gra_wb_single_delete(self.h, k, len(k))
        return self

    def merge(self, k, v):
        self.lib.gra_wb_merge(self.h, k, len(k), v, len(v))
        return self

    def delete_range(self, bk, ek):
        self.lib.gra_wb_delete_range(self.h, bk, len(bk), ek, len(ek))
        return self

    def log_data(self, blob):
        self.lib.gra_wb_put_log_data(self.h, blob, len(blob))
        return self

    def cf_put(self, cf, k, v):
        self.lib.gra_wb_cf_put(self.h, cf, k, len(k), v, len(v))
        return self

    def cf_delete(self, cf, k):
        self.lib.gra_wb_cf_delete(self.h, cf, k, len(k))
        return self

    def cf_single_delete(self, cf, k):
        self.lib.gra_wb_cf_single_delete(self.h, cf, k, len(k))
        return self

    def cf_merge(self, cf, k, v):
        self.lib.gra_wb_cf_merge(self.h, cf, k, len(k), v, len(v))
        return self

    def cf_delete_range(self, cf, bk, ek):
        self.lib.gra_wb_cf_delete_range(self.h, cf, bk, len(bk), ek, len(ek))
        return self

    def set_seq(self, seq):
        self.lib.gra_wb_set_seq(self.h, seq)
        return self

    @property
    def count(self):
        return self.lib.gra_wb_count(self.h)

    def data(self):
        n = C.c_size_t()
        p = self.lib.gra_wb_data(self.h, C.byref(n))
        return bytes(C.cast(p, C.POINTER(C.c_uint8 * n.value)).contents)


class Engine:
    """Per-GPU apply engine. Raises RuntimeError(no GPU) off-GPU — no CPU
    fallback exists for the follower apply path."""

    def __init__(self, nshards, device=-1, merge_op=0, store_ring=0,
                 store_bytes=0, staging_bytes=0, retain_log=0, log_bytes=0,
                 drain_host=0):
        self.lib = load()
        opts = GraEngineOpts()
        self.lib.gra_engine_opts_init(C.byref(opts))
        opts.nshards = nshards
        opts.device = device
        opts.merge_op = merge_op
        opts.store_ring = store_ring
        opts.retain_log = retain_log
        opts.drain_host = drain_host
        if log_bytes:
            opts.log_bytes = log_bytes
        if store_bytes:
            opts.store_bytes = store_bytes
        if staging_bytes:
            opts.staging_bytes = staging_bytes
        h = C.c_void_p()
        rc = self.lib.gra_engine_create(C.byref(opts), C.byref(h))
        if rc != GRA_OK:
            raise RuntimeError(f"gra_engine_create rc={rc}: {last_error(self.lib)}")
        self.h = h

    def close(self):
        if getattr(self, "h", None):
            self.lib.gra_engine_destroy(self.h)
            self.h = None

    def __del__(self):
        self.close()

    def open(self, shard):
        db = self.lib.gra_open(self.h, shard)
        if not db:
            raise RuntimeError(last_error(self.lib))
        return Db(self, db)

    def flush(self):
        rc = self.lib.gra_flush(self.h)
        if rc != GRA_OK:
            raise RuntimeError(f"gra_flush rc={rc}: {last_error(self.lib)}")

    def stats(self):
        s = GraStats()
        self.lib.gra_stats(self.h, C.byref(s))
        return s

    def stats_reset(self):
        self.lib.gra_stats_reset(self.h)

    def upload(self, arena, arena_bytes, descs, n):
        out = C.c_void_p()
        rc = self.lib.gra_upload(self.h, arena, arena_bytes, descs, n, C.byref(out))
        if rc != GRA_OK:
            raise RuntimeError(f"gra_upload rc={rc}: {last_error(self.lib)}")
        return Replay(self, out)

    def upload_snappy(self, comp_arena, comp_bytes, descs, n, ulens, counts=None):
        out = C.c_void_p()
        ul = (C.c_uint32 * n)(*ulens)
        cn = (C.c_uint32 * n)(*(counts if counts is not None else [1] * n))
        rc = self.lib.gra_upload_snappy(self.h, comp_arena, comp_bytes, descs,
                                        n, ul, cn, C.byref(out))
        if rc != GRA_OK:
            raise RuntimeError(f"gra_upload_snappy rc={rc}: {last_error(self.lib)}")
        return Replay(self, out)

    def pin_alloc(self, nbytes):
        p = C.POINTER(C.c_uint8)()
        rc = self.lib.gra_pin_alloc(self.h, nbytes, C.byref(p))
        if rc != GRA_OK:
            raise RuntimeError(f"gra_pin_alloc rc={rc}: {last_error(self.lib)}")
        return p


class Db:
    def __init__(self, engine, h):
        self.engine = engine
        self.lib = engine.lib
        self.h = h

    def close(self):
        if self.h:
            self.lib.gra_close(self.h)
            self.h = None

    def handle_replicate_response(self, rep, ts=0):
        return bool(self.lib.gra_handle_replicate_response(self.h, rep, len(rep), ts))

    def latest_seq(self):
        return self.lib.gra_latest_seq(self.h)

    def write_leader(self, rep):
        seq = C.c_uint64()
        rc = self.lib.gra_write_leader(self.h, rep, len(rep), C.byref(seq))
        if rc == GRA_CORRUPT:
            raise ValueError("corrupt WriteBatch rep")
        if rc != GRA_OK:
            raise RuntimeError(f"gra_write_leader rc={rc}")
        return seq.value

    def get(self, key, cap=1 << 20):
        buf = C.create_string_buffer(cap)
        vlen = C.c_size_t()
        rc = self.lib.gra_get(self.h, key, len(key), buf, cap, C.byref(vlen))
        if rc == GRA_NOT_FOUND:
            return None
        if rc != GRA_OK:
            raise RuntimeError(f"gra_get rc={rc}: {last_error(self.lib)}")
        return buf.raw[: vlen.value]

    def multiget(self, keys, val_stride=4096):
        """Batched point reads served from the device store (one block per
        query); merge-folding queries transparently fall back to the host
        path. Returns a list of values (None = miss)."""
        nq = len(keys)
        if nq == 0:
            return []
        keybuf = b"".join(keys)
        refs = (GraKeyRef * nq)()
        off = 0
        for i, k in enumerate(keys):
            refs[i] = GraKeyRef(off, len(k))
            off += len(k)
        valbuf = C.create_string_buffer(nq * val_stride)
        out = (GraGetResult * nq)()
        rc = self.lib.gra_multiget(self.h, nq, refs, keybuf, len(keybuf),
                                   valbuf, val_stride, out)
        if rc != GRA_OK:
            raise RuntimeError(f"gra_multiget rc={rc}: {last_error(self.lib)}")
        raw = valbuf.raw  # one copy; per-slice .raw would copy 2.5MB each
        res = []
        for i in range(nq):
            st = out[i].status
            if st == 1:  # miss
                res.append(None)
            elif st == 0:  # found
                assert out[i].vlen <= val_stride, "value exceeds stride"
                res.append(raw[i * val_stride:i * val_stride + out[i].vlen])
            else:  # needs host (merge fold / host-origin runs)
                res.append(self.get(keys[i]))
        return res

    def checksum(self):
        """Order-independent full-store content checksum (device-computed;
        comparable with the oracle's orc_shard_checksum at any size)."""
        v = C.c_uint64()
        rc = self.lib.gra_shard_checksum(self.h, C.byref(v))
        if rc != GRA_OK:
            raise RuntimeError(f"gra_shard_checksum rc={rc}: {last_error(self.lib)}")
        return v.value

    def counters(self):
        """Per-db stats ≅ the reference's per-db counter fan-out
        (replicator_stats.cpp:33-102). Returns a dict of counter names."""
        c = GraDbCounters()
        self.lib.gra_db_counters(self.h, C.byref(c))
        return {n: getattr(c, n) for n, _ in GraDbCounters._fields_}

    def wait_ack(self, seq, confirmed=True, timeout_ms=2000):
        """MaxNumberBox::wait equivalent: True when the downstream ack
        reached seq before the timeout."""
        return self.lib.gra_wait_ack(self.h, seq, 1 if confirmed else 0,
                                     timeout_ms) == GRA_OK

    def get_updates(self, since_seq, max_updates=50, cap=1 << 22,
                    observer=False):
        """Leader serving: [(seq, ts, rep_bytes), ...] with
        base seq > since_seq — the reference Update triple. An observer's
        request does not post an ACK (replicated_db.cpp:452-456).
        max_updates=0 means no limit (replicator.thrift:36-38), bounded
        here by the byte cap via a generous entry allocation."""
        if max_updates == 0:
            max_updates = max(64, cap // 16)  # every entry needs >=12B rep
        out = (GraServedUpdate * max_updates)()
        buf = C.create_string_buffer(cap)
        n = C.c_uint32()
        rc = self.lib.gra_get_updates(self.h, since_seq, max_updates, out,
                                      C.byref(n), buf, cap,
                                      1 if observer else 0)
        if rc != GRA_OK:
            raise RuntimeError(f"gra_get_updates rc={rc}: {last_error(self.lib)}")
        return [(out[i].seq, out[i].ts, buf.raw[out[i].off:out[i].off + out[i].len])
                for i in range(n.value)]


class Replay:
    def __init__(self, engine, h):
        self.engine = engine
        self.lib = engine.lib
        self.h = h

    def __del__(self):
        if getattr(self, "h", None):
            self.lib.gra_replay_destroy(self.h)
            self.h = None

    def tick(self, first, n):
        rc = self.lib.gra_replay_tick(self.h, first, n)
        if rc != GRA_OK:
            raise RuntimeError(f"gra_replay_tick rc={rc}: {last_error(self.lib)}")

    def prepare(self, first, n):
        """Pre-build the window's device-cached plan (untimed setup)."""
        rc = self.lib.gra_replay_prepare(self.h, first, n)
        if rc != GRA_OK:
            raise RuntimeError(f"gra_replay_prepare rc={rc}: {last_error(self.lib)}")

    def tick_h2d(self, first, n):
        rc = self.lib.gra_replay_tick_h2d(self.h, first, n)
        if rc != GRA_OK:
            raise RuntimeError(f"gra_replay_tick_h2d rc={rc}: {last_error(self.lib)}")

    def sync(self):
        rc = self.lib.gra_replay_sync(self.h)
        if rc != GRA_OK:
            raise RuntimeError(f"gra_replay_sync rc={rc}: {last_error(self.lib)}")


def gen_stream(nshards, n_updates, key_len=16, val_len=128, kind=0,
               key_space=1 << 24, zipf_s=0.99, seed=0, ts=0, arena=None,
               arena_cap=None, compressible=0):
    """Generate a deterministic synthetic replay stream into `arena`
    (a ctypes pointer/buffer) — returns (arena, used_bytes, descs)."""
    lib = load()
    g = GraGenOpts(nshards, key_len, val_len, kind, key_space, zipf_s, seed,
                   compressible, 0)
    if arena is None:
        worst = n_updates * (23 + key_len + val_len + 16) + 64
        arena_cap = worst
        arena = (C.c_uint8 * worst)()
    descs = (GraUpdateDesc * n_updates)()
    used = C.c_size_t()
    rc = lib.gra_gen_stream(C.byref(g), n_updates, C.cast(arena, C.POINTER(C.c_uint8)),
                            arena_cap, C.byref(used), descs, ts)
    if rc != GRA_OK:
        raise RuntimeError(f"gra_gen_stream rc={rc}")
    return arena, used.value, descs
