"""GPU parity tests — the follower apply path vs the CPU oracle on identical
update streams. Bit-exact bar: latest seq per shard and per-key Get equality
(the reference's own verification pattern, rocksdb_replicator_test.cpp:
162-207 and rocksdb_assumption_test.cpp:329-432).

All tests here require a real MI355X (@pytest.mark.gpu) and exercise the
HIP pipeline through the C-ABI — there is no CPU fallback to hide behind.
"""
import ctypes as C
import random

import pytest

import oracle_ffi
import rocksplicator_amd as ra
from pywb import PyBatch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def olib():
    return oracle_ffi.load()


def oracle_apply_stream(olib, nshards, raw, descs, n, merge_op=0):
    st = oracle_ffi.Store(olib, nshards, merge_op=merge_op)
    for i in range(n):
        d = descs[i]
        assert st.apply(d.shard, raw[d.off:d.off + d.len], d.ts)
    return st


def collect_keys(olib, raw, descs, n, per_shard_cap=64):
    """shard -> list of distinct full keys seen (for Get probes)."""
    keys = {}
    for i in range(n):
        d = descs[i]
        blob = raw[d.off:d.off + d.len]
        _, _, recs = oracle_ffi.decode(olib, blob)
        for r in recs:
            if not r.consumes_seq or r.type == 0x0F:
                continue
            k = blob[r.key_off:r.key_off + r.key_len]
            keys.setdefault(d.shard, [])
            if len(keys[d.shard]) < per_shard_cap and k not in keys[d.shard]:
                keys[d.shard].append(k)
    return keys


def check_parity(engine, olib_store, keys, shards):
    for s in shards:
        db = engine.open(s)
        assert db.latest_seq() == olib_store.latest_seq(s), f"shard {s} seq"
        for k in keys.get(s, []):
            assert db.get(k) == olib_store.get(s, k), f"shard {s} key {k.hex()}"
        db.close()


# ---------------- streaming ingest (HandleReplicateResponse) ----------------

def test_stream_small_batches(olib):
    e = ra.Engine(nshards=4)
    dbs = [e.open(s) for s in range(4)]
    ost = oracle_ffi.Store(olib, 4)
    rng = random.Random(1)
    blobs = []
    for i in range(200):
        s = rng.randrange(4)
        b = PyBatch()
        k = f"key{rng.randrange(50)}".encode()
        if rng.random() < 0.7:
            b.put(k, rng.randbytes(rng.randrange(1, 200)))
        else:
            b.delete(k)
        blobs.append((s, b.data()))
    for s, rep in blobs:
        assert dbs[s].handle_replicate_response(rep, ts=123)
        assert ost.apply(s, rep, 123)
    e.flush()
    for s in range(4):
        assert dbs[s].latest_seq() == ost.latest_seq(s)
        for i in range(50):
            k = f"key{i}".encode()
            assert dbs[s].get(k) == ost.get(s, k), (s, k)
    e.close()


def test_stream_multi_record_batches(olib):
    e = ra.Engine(nshards=2)
    db = e.open(0)
    ost = oracle_ffi.Store(olib, 2)
    # the assumption-test batch shape (del, put, put, merge) + logdata trailer
    rep = (PyBatch().delete(b"key1").put(b"key2", b"value2")
           .put(b"key2", b"value2").merge(b"key1", b"value1")
           .log_data(b"12345678").data())
    assert db.handle_replicate_response(rep)
    assert ost.apply(0, rep)
    rep2 = PyBatch().put(b"key1", b"v1").data()
    assert db.handle_replicate_response(rep2)
    assert ost.apply(0, rep2)
    e.flush()
    assert db.latest_seq() == ost.latest_seq(0) == 5
    for k in (b"key1", b"key2", b"nope"):
        assert db.get(k) == ost.get(0, k)
    e.close()


def test_stream_edge_cases(olib):
    e = ra.Engine(nshards=1)
    db = e.open(0)
    ost = oracle_ffi.Store(olib, 1)
    cases = [
        PyBatch().log_data(b"only-log-data").data(),       # count 0
        PyBatch().put(b"", b"empty-key-value").data(),     # empty key
        PyBatch().put(b"ek", b"").data(),                  # empty value
        PyBatch().put(b"big", bytes(range(256)) * 256).data(),  # 64 KB value
        PyBatch().delete_range(b"a", b"c").data(),
        PyBatch().cf_put(3, b"cfk", b"cfv").data(),        # cf-prefixed
        PyBatch().single_delete(b"ek").data(),
    ]
    for rep in cases:
        assert db.handle_replicate_response(rep)
        assert ost.apply(0, rep)
    e.flush()
    assert db.latest_seq() == ost.latest_seq(0)
    for k in (b"", b"ek", b"big", b"a", b"b", b"cfk"):
        assert db.get(k) == ost.get(0, k), k
    e.close()


def test_corrupt_batch_poisons_shard(olib):
    e = ra.Engine(nshards=2)
    db = e.open(0)
    good = PyBatch().put(b"k", b"v").data()
    assert db.handle_replicate_response(good)
    e.flush()
    assert db.latest_seq() == 1
    bad = bytearray(PyBatch().put(b"x", b"y").data())
    bad[8] = 9  # count mismatch -> corruption on GPU validation
    assert db.handle_replicate_response(bytes(bad))  # accepted (async)
    e.flush()
    # poisoned: next call fails once (reference delayed-re-pull cadence),
    # latest_seq reports the durable seq
    assert db.latest_seq() == 1
    assert not db.handle_replicate_response(good)
    # after the failure signal, the shard recovers
    assert db.handle_replicate_response(good)
    e.flush()
    assert db.latest_seq() == 2
    assert db.get(b"k") == b"v"
    e.close()


# ---------------- replay path (blobs resident in HBM) ----------------

@pytest.mark.parametrize("kind,merge_op,nupd,nshards,vlen", [
    (0, 0, 20000, 64, 128),   # config #2 shape (scaled down)
    (1, 0, 20000, 256, 256),  # zipf keys
    (2, 1, 20000, 64, 64),    # mixed put/delete/merge with u64add
])
def test_replay_parity(olib, kind, merge_op, nupd, nshards, vlen):
    arena, used, descs = ra.gen_stream(nshards=nshards, n_updates=nupd,
                                       key_len=16, val_len=vlen, kind=kind,
                                       key_space=1 << 16, seed=99 + kind)
    raw = bytes(arena)[:used]
    ost = oracle_apply_stream(olib, nshards, raw, descs, nupd, merge_op=merge_op)
    e = ra.Engine(nshards=nshards, merge_op=merge_op)
    rep = e.upload(C.cast(arena, C.POINTER(C.c_uint8)), used, descs, nupd)
    # several ticks, windows must be shard-grouped (generator guarantees it
    # only at 50-update boundaries)
    tick = 5000
    for first in range(0, nupd, tick):
        rep.tick(first, min(tick, nupd - first))
    rep.sync()
    keys = collect_keys(olib, raw, descs, min(nupd, 4000))
    check_parity(e, ost, keys, range(0, nshards, max(1, nshards // 16)))
    s = e.stats()
    assert s.updates == nupd
    assert s.records == nupd  # 1 record per update in generator streams
    e.close()


def test_replay_h2d_parity(olib):
    nshards, nupd = 32, 10000
    arena, used, descs = ra.gen_stream(nshards=nshards, n_updates=nupd,
                                       key_len=16, val_len=128, seed=5)
    raw = bytes(arena)[:used]
    ost = oracle_apply_stream(olib, nshards, raw, descs, nupd)
    e = ra.Engine(nshards=nshards)
    # pinned arena for PCIe-inclusive path
    pin = e.pin_alloc(used)
    C.memmove(pin, arena, used)
    rep = e.upload(pin, used, descs, nupd)
    for first in range(0, nupd, 2500):
        rep.tick_h2d(first, 2500)
    rep.sync()
    keys = collect_keys(olib, raw, descs, 4000)
    check_parity(e, ost, keys, range(0, nshards, 4))
    st = e.stats()
    assert st.h2d_ms > 0
    e.close()


def test_leader_write_then_get_on_gpu_box(olib):
    """Host-side leader path (WriteToLeader semantics) — same run format."""
    e = ra.Engine(nshards=1, merge_op=1)
    db = e.open(0)
    ost = oracle_ffi.Store(olib, 1, merge_op=1)
    one = (1).to_bytes(8, "little")
    for i in range(10):
        rep = PyBatch().merge(b"ctr", one).data()
        seq = db.write_leader(rep)
        assert ost.apply(0, rep)
        assert seq == ost.latest_seq(0)
    assert db.get(b"ctr") == ost.get(0, b"ctr")
    assert int.from_bytes(db.get(b"ctr"), "little") == 10
    e.close()


def test_mixed_leader_and_replicated(olib):
    """Leader-written host runs and GPU-applied device runs interleave in one
    shard's run list; Get must merge across both."""
    e = ra.Engine(nshards=1)
    db = e.open(0)
    ost = oracle_ffi.Store(olib, 1)
    db.write_leader(PyBatch().put(b"a", b"host1").data())
    ost.apply(0, PyBatch().put(b"a", b"host1").data())
    assert db.handle_replicate_response(PyBatch().put(b"a", b"gpu1").put(b"b", b"gpu2").data())
    ost.apply(0, PyBatch().put(b"a", b"gpu1").put(b"b", b"gpu2").data())
    e.flush()
    db.write_leader(PyBatch().delete(b"b").data())
    ost.apply(0, PyBatch().delete(b"b").data())
    assert db.latest_seq() == ost.latest_seq(0) == 4
    for k in (b"a", b"b"):
        assert db.get(k) == ost.get(0, k)
    e.close()


def test_full_config_shape_parity(olib):
    """Config #3 at a size the oracle still finishes in seconds (1024 shards,
    16B/1KB Zipf): per-shard seq equality for ALL shards + sampled per-key
    Get equality + the size-independent invariant that total applied records
    equals total seq advance (a6)."""
    nshards, nupd = 1024, 102400
    arena, used, descs = ra.gen_stream(nshards=nshards, n_updates=nupd,
                                       key_len=16, val_len=1024, kind=1,
                                       seed=2024)
    raw = bytes(arena)[:used]
    ost = oracle_apply_stream(olib, nshards, raw, descs, nupd)
    e = ra.Engine(nshards=nshards, store_bytes=4 << 30)
    rep = e.upload(C.cast(arena, C.POINTER(C.c_uint8)), used, descs, nupd)
    for first in range(0, nupd, 51200):
        rep.tick(first, 51200)
    rep.sync()
    st = e.stats()
    assert st.records == nupd
    total_seq = 0
    for s in range(nshards):
        db = e.open(s)
        seq = db.latest_seq()
        assert seq == ost.latest_seq(s), f"shard {s}"
        total_seq += seq
        db.close()
    assert total_seq == nupd  # every update consumed exactly one seq
    keys = collect_keys(olib, raw, descs, 6000, per_shard_cap=8)
    check_parity(e, ost, keys, range(0, nshards, 61))
    e.close()


def test_snappy_replay_parity(olib):
    """Config #5: Snappy-compressed payloads, mixed Put/Delete/Merge
    70/20/10, GPU decompress stage + apply vs oracle decompress + apply."""
    import ctypes as CT
    plib = ra.load()
    nshards, nupd = 64, 20000
    arena, used, descs = ra.gen_stream(nshards=nshards, n_updates=nupd,
                                       key_len=16, val_len=256, kind=2,
                                       key_space=1 << 16, seed=77,
                                       compressible=1)
    raw = bytes(arena)[:used]
    # oracle leg: decompress-then-apply must equal plain apply of originals
    ost = oracle_apply_stream(olib, nshards, raw, descs, nupd, merge_op=1)
    # compress each update (host transport side)
    comp = CT.create_string_buffer(used + used // 4 + 64 * nupd)
    cdescs = (ra.ffi.GraUpdateDesc * nupd)()
    ulens = []
    off = 0
    for i in range(nupd):
        d = descs[i]
        blob = raw[d.off:d.off + d.len]
        cap = len(blob) + len(blob) // 6 + 64
        tmp = CT.create_string_buffer(cap)
        clen = plib.gra_snappy_compress(blob, len(blob), tmp, cap)
        assert clen > 0
        CT.memmove(CT.byref(comp, off), tmp, clen)
        cdescs[i] = ra.ffi.GraUpdateDesc(d.shard, clen, off, d.ts)
        ulens.append(d.len)
        off += clen
    assert off < used  # compressible streams actually compressed
    e = ra.Engine(nshards=nshards, merge_op=1)
    rep = e.upload_snappy(CT.cast(comp, CT.POINTER(CT.c_uint8)), off,
                          cdescs, nupd, ulens)
    for first in range(0, nupd, 5000):
        rep.tick(first, 5000)
    rep.sync()
    st = e.stats()
    assert st.snappy_ms > 0
    keys = collect_keys(olib, raw, descs, 4000)
    check_parity(e, ost, keys, range(0, nshards, 4))
    e.close()


def test_concurrent_streaming_threads(olib):
    """HandleReplicateResponse from many threads (cross-shard concurrency is
    the reference's executor model, rocksdb_replicator.cpp:41-67) — per-shard
    order preserved, parity vs oracle."""
    import threading
    nshards, per_shard = 16, 300
    e = ra.Engine(nshards=nshards)
    ost = oracle_ffi.Store(olib, nshards)
    streams = {}
    for s in range(nshards):
        rng = random.Random(1000 + s)
        blobs = []
        for i in range(per_shard):
            b = PyBatch()
            k = f"s{s}k{rng.randrange(40)}".encode()
            if rng.random() < 0.75:
                b.put(k, rng.randbytes(rng.randrange(1, 256)))
            else:
                b.delete(k)
            blobs.append(b.data())
        streams[s] = blobs
        for rep in blobs:
            assert ost.apply(s, rep)

    def worker(s):
        db = e.open(s)
        for rep in streams[s]:
            assert db.handle_replicate_response(rep, ts=7)
        db.close()

    threads = [threading.Thread(target=worker, args=(s,)) for s in range(nshards)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    e.flush()
    for s in range(nshards):
        db = e.open(s)
        assert db.latest_seq() == ost.latest_seq(s) == per_shard
        c = db.counters()
        assert c["updates_applied"] == per_shard
        assert c["in_bytes"] == sum(len(r) for r in streams[s])
        assert c["apply_failures"] == 0
        for i in range(40):
            k = f"s{s}k{i}".encode()
            assert db.get(k) == ost.get(s, k)
        db.close()
    e.close()


def test_million_update_properties(olib):
    """BASELINE config #3 at full per-tick scale (1M+ updates), checked via
    size-independent properties (tier rule: full sizes via invariants the
    domain offers): every update applies exactly once (sum of per-shard seqs
    == n), record count == n, payload bytes match the generator's shapes."""
    nshards, nupd = 1024, 1228800
    arena, used, descs = ra.gen_stream(nshards=nshards, n_updates=nupd,
                                       key_len=16, val_len=1024, kind=1,
                                       seed=31337)
    e = ra.Engine(nshards=nshards, store_bytes=4 << 30, store_ring=1)
    rep = e.upload(C.cast(arena, C.POINTER(C.c_uint8)), used, descs, nupd)
    tick = 409600
    for first in range(0, nupd, tick):
        rep.tick(first, tick)
    rep.sync()
    st = e.stats()
    assert st.updates == nupd
    assert st.records == nupd
    # payload: every record is 16B key + 1KB value -> 1040 raw, 1056 aligned
    assert st.payload_bytes == nupd * ((16 + 1024 + 15) & ~15)
    assert st.blob_bytes == used
    total_seq = 0
    for s in range(nshards):
        db = e.open(s)
        total_seq += db.latest_seq()
        db.close()
    assert total_seq == nupd
    e.close()


def test_ring_store_wraps_safely(olib):
    """Bench-mode ring store: many ticks overflowing a small arena must wrap
    without corruption of the seq/stat bookkeeping."""
    nshards, nupd = 64, 400000
    arena, used, descs = ra.gen_stream(nshards=nshards, n_updates=nupd,
                                       key_len=16, val_len=128, seed=55)
    e = ra.Engine(nshards=nshards, store_ring=1, store_bytes=16 << 20)  # 16 MB
    rep = e.upload(C.cast(arena, C.POINTER(C.c_uint8)), used, descs, nupd)
    tick = 50000  # ~8 MB payload per tick -> wraps every ~2 ticks
    for first in range(0, nupd, tick):
        rep.tick(first, tick)
    rep.sync()
    st = e.stats()
    assert st.updates == nupd and st.records == nupd
    total = sum(e.open(s).latest_seq() for s in range(nshards))
    assert total == nupd
    e.close()


def test_drain_host_mode_matches_lazy(olib):
    """drain_host=1 eagerly materializes host memtable runs at flush; Get
    results must equal the lazy device-resident mode and the oracle."""
    nshards, nupd = 16, 5000
    arena, used, descs = ra.gen_stream(nshards=nshards, n_updates=nupd,
                                       key_len=16, val_len=64, seed=66)
    raw = bytes(arena)[:used]
    ost = oracle_apply_stream(olib, nshards, raw, descs, nupd)
    e = ra.Engine(nshards=nshards, drain_host=1)
    rep = e.upload(C.cast(arena, C.POINTER(C.c_uint8)), used, descs, nupd)
    rep.tick(0, nupd)
    rep.sync()
    keys = collect_keys(olib, raw, descs, 3000)
    check_parity(e, ost, keys, range(nshards))
    e.close()


def test_fuzz_all_record_types(olib):
    """Seeded fuzz over every WAL-legal record shape the 5.7.fb format
    allows — multi-record batches mixing Put/Delete/SingleDelete/Merge/
    RangeDelete, CF-prefixed variants, LogData, Noop and 2PC markers
    (consume no seq) — streamed through HandleReplicateResponse and
    compared per key against the oracle."""
    rng = random.Random(0xF022)
    nshards = 8
    e = ra.Engine(nshards=nshards, merge_op=0)
    dbs = [e.open(s) for s in range(nshards)]
    ost = oracle_ffi.Store(olib, nshards, merge_op=0)
    keys = [f"key{i:03d}".encode() for i in range(120)]
    for _ in range(1500):
        s = rng.randrange(nshards)
        b = PyBatch()
        for _ in range(rng.randrange(1, 6)):
            k = rng.choice(keys)
            r = rng.random()
            if r < 0.35:
                b.put(k, rng.randbytes(rng.randrange(0, 200)))
            elif r < 0.5:
                b.merge(k, rng.randbytes(rng.randrange(1, 32)))
            elif r < 0.6:
                b.delete(k)
            elif r < 0.65:
                b.single_delete(k)
            elif r < 0.70:
                lo, hi = sorted([rng.choice(keys), rng.choice(keys)])
                if lo != hi:
                    if rng.random() < 0.3:
                        b.cf_delete_range(rng.randrange(1, 4), lo, hi)
                    else:
                        b.delete_range(lo, hi)
            elif r < 0.78:
                b.cf_put(rng.randrange(1, 4), k, rng.randbytes(16))
            elif r < 0.83:
                b.cf_delete(rng.randrange(1, 4), k)
            elif r < 0.88:
                b.cf_merge(rng.randrange(1, 4), k, rng.randbytes(8))
            elif r < 0.93:
                b.log_data(rng.randbytes(rng.randrange(0, 64)))
            elif r < 0.96:
                b.noop()
            elif r < 0.98:
                b.begin_prepare()
            else:
                b.commit_xid(rng.randbytes(8))
        rep = b.data()
        assert dbs[s].handle_replicate_response(rep, ts=1) == ost.apply(s, rep, 1)
    e.flush()
    for s in range(nshards):
        assert dbs[s].latest_seq() == ost.latest_seq(s), s
        for k in keys:
            assert dbs[s].get(k) == ost.get(s, k), (s, k)
        # cf-namespaced probes
        for cf in range(1, 4):
            for k in keys[:30]:
                ck = cf.to_bytes(4, "little") + k
                assert dbs[s].get(ck) == ost.get(s, ck), (s, cf, k)
    e.close()


@pytest.mark.parametrize("klen,vlen", [
    (7, 100),    # tiny unaligned keys; odd values
    (24, 1000),  # value dst unaligned (24 % 16 != 0) -> dword-funnel fallback
    (40, 512),   # keys > 32B -> key copy-task path (not inlined anywhere)
    (3, 17),     # everything misaligned and tiny
])
def test_replay_parity_odd_shapes(olib, klen, vlen):
    nupd, nshards = 15000, 32
    arena, used, descs = ra.gen_stream(nshards=nshards, n_updates=nupd,
                                       key_len=klen, val_len=vlen, kind=2,
                                       key_space=1 << 14, seed=400 + klen)
    raw = bytes(arena)[:used]
    ost = oracle_apply_stream(olib, nshards, raw, descs, nupd)
    e = ra.Engine(nshards=nshards)
    rep = e.upload(C.cast(arena, C.POINTER(C.c_uint8)), used, descs, nupd)
    for first in range(0, nupd, 5000):
        rep.tick(first, 5000)
    rep.sync()
    keys = collect_keys(olib, raw, descs, 4000, per_shard_cap=24)
    check_parity(e, ost, keys, range(0, nshards, 3))
    e.close()


def test_precise_corruption_truncation(olib):
    """A corrupt batch mid-tick must affect ONLY its shard, and that shard
    keeps exactly the records from batches before the corrupt one."""
    e = ra.Engine(nshards=3)
    dbs = [e.open(s) for s in range(3)]
    ost = oracle_ffi.Store(olib, 3)
    # shard 0: clean stream; shard 1: good, good, BAD, good; shard 2: clean
    for i in range(5):
        for s in (0, 2):
            rep = PyBatch().put(f"s{s}k{i}".encode(), b"v").data()
            assert dbs[s].handle_replicate_response(rep)
            assert ost.apply(s, rep)
    good1 = [PyBatch().put(f"g{i}".encode(), f"w{i}".encode()).data()
             for i in range(4)]
    bad = bytearray(PyBatch().put(b"bad", b"bad").data())
    bad[8] = 7  # count mismatch
    assert dbs[1].handle_replicate_response(good1[0])
    assert dbs[1].handle_replicate_response(good1[1])
    assert dbs[1].handle_replicate_response(bytes(bad))
    assert dbs[1].handle_replicate_response(good1[2])  # post-bad: dropped
    e.flush()
    # clean shards unaffected and parity-green
    for s in (0, 2):
        assert dbs[s].latest_seq() == ost.latest_seq(s) == 5
        for i in range(5):
            assert dbs[s].get(f"s{s}k{i}".encode()) == b"v"
    # shard 1: exactly the 2 pre-corruption batches kept, then poisoned
    assert dbs[1].latest_seq() == 2
    assert dbs[1].get(b"g0") == b"w0" and dbs[1].get(b"g1") == b"w1"
    assert dbs[1].get(b"g2") is None  # post-bad batch rolled back
    assert not dbs[1].handle_replicate_response(good1[2])  # fail-once signal
    # re-pull from the durable boundary succeeds
    assert dbs[1].handle_replicate_response(good1[2])
    assert dbs[1].handle_replicate_response(good1[3])
    e.flush()
    assert dbs[1].latest_seq() == 4
    assert dbs[1].get(b"g2") == b"w2" and dbs[1].get(b"g3") == b"w3"
    e.close()


def test_corruption_decision_matrix(olib):
    """Random single-byte corruptions of valid batches: the GPU validator
    must accept/reject exactly like the oracle (truncations, bad varints,
    unknown tags, count mismatches, slice overruns...). Accepted mutants
    must also APPLY identically."""
    rng = random.Random(0xBADF00D)
    e = ra.Engine(nshards=1)
    db = e.open(0)
    ost = oracle_ffi.Store(olib, 1)
    agree_reject = agree_accept = 0
    for trial in range(300):
        b = PyBatch()
        for _ in range(rng.randrange(1, 4)):
            k = f"k{rng.randrange(20)}".encode()
            if rng.random() < 0.7:
                b.put(k, rng.randbytes(rng.randrange(0, 60)))
            else:
                b.delete(k)
        rep = bytearray(b.data())
        mode = rng.random()
        if mode < 0.45:  # single-byte mutation
            rep[rng.randrange(len(rep))] = rng.randrange(256)
        elif mode < 0.7:  # truncation
            del rep[rng.randrange(12, len(rep)):]
        elif mode < 0.8:  # garbage suffix
            rep += rng.randbytes(rng.randrange(1, 8))
        # else: leave valid
        rep = bytes(rep)
        oracle_ok = ost.apply(0, rep)
        # engine: submit + flush, then read the poison signal
        assert db.handle_replicate_response(rep)
        e.flush()
        gpu_ok = db.handle_replicate_response(b"")  # len<12 -> False anyway
        # gpu_ok False here means either poisoned (bad rep) or the empty
        # probe was rejected; disambiguate via seq vs oracle
        assert db.latest_seq() == ost.latest_seq(0), (
            f"trial {trial}: oracle_ok={oracle_ok} rep={rep.hex()}")
        if oracle_ok:
            agree_accept += 1
        else:
            agree_reject += 1
        # clear any poison signal so the next trial starts clean
        db.handle_replicate_response(PyBatch().put(b"sync", b"1").data())
        ost.apply(0, PyBatch().put(b"sync", b"1").data())
        e.flush()
        assert db.latest_seq() == ost.latest_seq(0)
    assert agree_reject > 30 and agree_accept > 30  # matrix actually exercised
    # final content parity
    for i in range(20):
        k = f"k{i}".encode()
        assert db.get(k) == ost.get(0, k)
    e.close()


def test_multiget_device_reads(olib):
    """Batched point reads served by k_multiget from the device store ==
    host-path gra_get == oracle, across puts/overwrites/deletes/range
    tombstones/merges (merge queries route to the host fold)."""
    nshards, nupd = 16, 30000
    arena, used, descs = ra.gen_stream(nshards=nshards, n_updates=nupd,
                                       key_len=16, val_len=200, kind=2,
                                       key_space=1 << 12, seed=808)
    raw = bytes(arena)[:used]
    ost = oracle_apply_stream(olib, nshards, raw, descs, nupd, merge_op=0)
    e = ra.Engine(nshards=nshards, merge_op=0)
    rep = e.upload(C.cast(arena, C.POINTER(C.c_uint8)), used, descs, nupd)
    rep.tick(0, nupd)
    rep.sync()
    keys = collect_keys(olib, raw, descs, 8000, per_shard_cap=64)
    for s in range(0, nshards, 3):
        db = e.open(s)
        ks = keys.get(s, []) + [b"definitely-missing-key-0123"]
        got = db.multiget(ks)
        for k, v in zip(ks, got):
            assert v == ost.get(s, k), (s, k.hex())
        db.close()
    # range tombstone + merge interplay through the device path
    db = e.open(0)
    assert db.handle_replicate_response(PyBatch().put(b"rk1", b"a").put(b"rk2", b"b").data())
    assert db.handle_replicate_response(PyBatch().delete_range(b"rk1", b"rk2").data())
    assert db.handle_replicate_response(PyBatch().merge(b"rk3", b"m1").data())
    e.flush()
    ost.apply(0, PyBatch().put(b"rk1", b"a").put(b"rk2", b"b").data())
    ost.apply(0, PyBatch().delete_range(b"rk1", b"rk2").data())
    ost.apply(0, PyBatch().merge(b"rk3", b"m1").data())
    got = db.multiget([b"rk1", b"rk2", b"rk3"])
    assert got[0] is None and got[1] == b"b" and got[2] == b"m1"
    assert [ost.get(0, k) for k in (b"rk1", b"rk2", b"rk3")] == got
    e.close()


def test_oversize_batch_rejected(olib):
    """A batch with more records than max_wb_records (DoS guard) is rejected
    like corruption: shard poisons, seq rolls back, clean recovery."""
    e = ra.Engine(nshards=1)  # default cap 1024 records
    db = e.open(0)
    big = PyBatch()
    for i in range(1100):
        big.put(f"k{i}".encode(), b"v")
    assert db.handle_replicate_response(big.data())  # accepted (async)
    e.flush()
    assert db.latest_seq() == 0  # rolled back
    assert not db.handle_replicate_response(PyBatch().put(b"a", b"b").data())
    assert db.handle_replicate_response(PyBatch().put(b"a", b"b").data())
    e.flush()
    assert db.latest_seq() == 1 and db.get(b"a") == b"b"
    # oracle has no record cap: this is an engine-level operational guard
    e.close()


def test_tick_h2d_rejects_snappy_and_device_uploads(olib):
    import ctypes as CT
    plib = ra.load()
    arena, used, descs = ra.gen_stream(nshards=2, n_updates=100, key_len=16,
                                       val_len=64, kind=2, seed=1,
                                       compressible=1)
    raw = bytes(arena)[:used]
    comp = CT.create_string_buffer(used + used // 4 + 64 * 100)
    cdescs = (ra.ffi.GraUpdateDesc * 100)()
    ulens = []
    off = 0
    for i in range(100):
        d = descs[i]
        blob = raw[d.off:d.off + d.len]
        cap = len(blob) + len(blob) // 6 + 64
        tmp = CT.create_string_buffer(cap)
        clen = plib.gra_snappy_compress(blob, len(blob), tmp, cap)
        CT.memmove(CT.byref(comp, off), tmp, clen)
        cdescs[i] = ra.ffi.GraUpdateDesc(d.shard, clen, off, 0)
        ulens.append(d.len)
        off += clen
    e = ra.Engine(nshards=2)
    rep = e.upload_snappy(CT.cast(comp, CT.POINTER(CT.c_uint8)), off,
                          cdescs, 100, ulens)
    with pytest.raises(RuntimeError, match="tick_h2d"):
        rep.tick_h2d(0, 50)
    rep.tick(0, 100)  # the supported path still works
    rep.sync()
    e.close()


def test_full_store_checksum_parity(olib):
    """'Checksum of checksums' (any-size parity property): the device store's
    order-independent per-record content hash must equal the oracle's on the
    same stream — full-store bit equality without per-key probing."""
    nshards, nupd = 256, 200000
    arena, used, descs = ra.gen_stream(nshards=nshards, n_updates=nupd,
                                       key_len=16, val_len=200, kind=2,
                                       key_space=1 << 14, seed=424242)
    raw = bytes(arena)[:used]
    ost = oracle_apply_stream(olib, nshards, raw, descs, nupd)
    e = ra.Engine(nshards=nshards, store_bytes=2 << 30)
    rep = e.upload(C.cast(arena, C.POINTER(C.c_uint8)), used, descs, nupd)
    for first in range(0, nupd, 51200):
        rep.tick(first, min(51200, nupd - first))
    rep.sync()
    mismatches = []
    for s in range(nshards):
        db = e.open(s)
        gpu = db.checksum()
        orc = olib.orc_shard_checksum(ost.h, s)
        if gpu != orc:
            mismatches.append((s, hex(gpu), hex(orc)))
        db.close()
    assert not mismatches, mismatches[:5]
    # mixed host(leader)+device runs also checksum consistently
    db = e.open(0)
    rep2 = PyBatch().put(b"hostside", b"entry").data()
    db.write_leader(rep2)
    assert ost.apply(0, rep2)
    assert db.checksum() == olib.orc_shard_checksum(ost.h, 0)
    e.close()


def test_tiny_staging_forces_overflow_ticks(olib):
    """A deliberately tiny staging buffer forces the writer-overflow tick
    path (chunk reservation fails -> abandon + tick + retry) on every few
    updates; parity must hold."""
    e = ra.Engine(nshards=2, staging_bytes=1 << 20)  # 1 MiB staging
    dbs = [e.open(s) for s in range(2)]
    ost = oracle_ffi.Store(olib, 2)
    big = bytes(range(256)) * 128  # 32 KiB values
    for i in range(60):  # ~2 MB of blobs through a 1 MB buffer
        s = i & 1
        rep = PyBatch().put(f"k{i:02d}".encode(), big).data()
        assert dbs[s].handle_replicate_response(rep)
        assert ost.apply(s, rep)
    e.flush()
    for s in range(2):
        assert dbs[s].latest_seq() == ost.latest_seq(s)
        assert dbs[s].checksum() == olib.orc_shard_checksum(ost.h, s)
    st = e.stats()
    assert st.ticks >= 2  # the overflow path actually forced multiple ticks
    e.close()


def test_bad_device_index_fails_gracefully(olib):
    with pytest.raises(RuntimeError):
        ra.Engine(nshards=1, device=99)


def test_store_full_rolls_back_and_poisons(olib):
    """Non-ring store exhaustion (GRA_FULL condition): the overflowing
    tick applies nothing; seqs roll back to the durable boundary and the
    shard poisons until the operator intervenes — no silent seq advance."""
    e = ra.Engine(nshards=1, store_bytes=1 << 20, store_ring=0)  # 1 MiB store
    db = e.open(0)
    small = PyBatch().put(b"first", b"x").data()
    assert db.handle_replicate_response(small)
    e.flush()
    assert db.latest_seq() == 1
    big_val = bytes(512 * 1024)  # two of these exceed the 1 MiB arena
    for i in range(4):
        db.handle_replicate_response(
            PyBatch().put(f"big{i}".encode(), big_val).data())
    e.flush()
    c = db.counters()
    assert c["apply_failures"] >= 1
    assert db.latest_seq() <= 3  # durable boundary, not the optimistic 5
    assert not db.handle_replicate_response(small)  # poisoned: fail once
    # durable data intact
    assert db.get(b"first") == b"x"
    e.close()


@pytest.mark.gpu
def test_failed_upload_leaves_seq_state_untouched(olib):
    """A rejected upload (bad descriptor) must not advance any shard's
    next_seq: the engine validates every descriptor before assigning seqs,
    so a caller can fix the input and retry without a seq gap."""
    import ctypes as C
    eng = ra.Engine(nshards=4)
    try:
        db = eng.open(1)
        # seed one good batch so the shard has nonzero seq state
        assert db.handle_replicate_response(ra.Batch().put(b"k", b"v").data(), 1)
        eng.flush()
        seq0 = db.latest_seq()

        arena, used, descs = ra.ffi.gen_stream(
            nshards=4, n_updates=64, key_len=16, val_len=64, kind=0, seed=42)
        descs[50].off = used + 1000  # out of range -> whole upload rejected
        with pytest.raises(RuntimeError, match="bad desc"):
            eng.upload(C.cast(arena, C.POINTER(C.c_uint8)), used, descs, 64)
        assert db.latest_seq() == seq0

        # fixed input applies with contiguous seqs (no gap from the failure)
        arena2, used2, descs2 = ra.ffi.gen_stream(
            nshards=4, n_updates=64, key_len=16, val_len=64, kind=0, seed=42)
        rep = eng.upload(C.cast(arena2, C.POINTER(C.c_uint8)), used2, descs2, 64)
        rep.tick(0, 64)
        rep.sync()
        n_shard1 = sum(1 for d in descs2 if d.shard == 1)
        assert db.latest_seq() == seq0 + n_shard1

        # snappy comp-bounds validation rejects out-of-range comp descs
        cdescs = (ra.ffi.GraUpdateDesc * 1)()
        cdescs[0].shard, cdescs[0].len, cdescs[0].off = 0, 100, 64
        buf = (C.c_uint8 * 128)()
        with pytest.raises(RuntimeError, match="out of range"):
            eng.upload_snappy(C.cast(buf, C.POINTER(C.c_uint8)), 128,
                              cdescs, 1, [100])
    finally:
        eng.close()


def test_stale_tick_after_corruption_dropped(olib):
    """a clean tick staged BEFORE a corrupt batch's
    failure was detected carries base_seqs that assumed the corrupt batch
    applied. The err==0 ingest path must drop such groups while the shard is
    poisoned — without the fix durable_seq re-advances past the rolled-back
    boundary and the failed batch is silently skipped on re-pull."""
    e = ra.Engine(nshards=1)
    db = e.open(0)
    b1 = PyBatch().put(b"g1", b"v1").data()
    bad = bytearray(PyBatch().put(b"bad", b"bad").data())
    bad[8] = 2  # header count 2, body has 1 record -> GPU validation fails
    b3 = PyBatch().put(b"g3", b"v3").data()
    # hand-build a replay arena: 3 one-batch ticks for shard 0. Upload
    # assigns optimistic seqs (b3 base assumes bad's 2 seqs applied).
    blobs = [b1, bytes(bad), b3]
    used = sum(len(b) for b in blobs)
    arena = (C.c_uint8 * (used + 64))()
    descs = (ra.ffi.GraUpdateDesc * 3)()
    off = 0
    for i, b in enumerate(blobs):
        C.memmove(C.byref(arena, off), b, len(b))
        descs[i].shard, descs[i].len, descs[i].off, descs[i].ts = 0, len(b), off, 0
        off += len(b)
    rep = e.upload(C.cast(arena, C.POINTER(C.c_uint8)), used, descs, 3)
    rep.tick(0, 1)   # applies: seq 1
    rep.tick(1, 1)   # corrupt: poisons, rolls back
    rep.tick(2, 1)   # STALE (base seq 4): must be dropped, not applied
    rep.sync()
    assert db.latest_seq() == 1, "stale tick must not advance durable_seq"
    assert db.get(b"g1") == b"v1"
    assert db.get(b"g3") is None, "stale batch content must not be visible"
    # reference cadence: fail once, then re-pull applies b3 at the correct seq
    assert not db.handle_replicate_response(b3)
    assert db.handle_replicate_response(b3)
    e.flush()
    assert db.latest_seq() == 2
    assert db.get(b"g3") == b"v3"
    # oracle agreement on the recovered stream
    ost = oracle_ffi.Store(olib, 1)
    assert ost.apply(0, b1)
    assert not ost.apply(0, bytes(bad))
    assert ost.apply(0, b3)
    assert db.latest_seq() == ost.latest_seq(0)
    assert db.checksum() == olib_checksum(olib, ost)
    e.close()


def olib_checksum(olib, ost, shard=0):
    return olib.orc_shard_checksum(ost.h, shard)


def test_stale_streaming_ticks_after_corruption(olib):
    """Same hazard through the streaming path: updates staged between the
    corrupt batch and its detection (separate flushes -> separate ticks)
    must be dropped, and the fail-once report must drain them before
    clearing the poison flag."""
    e = ra.Engine(nshards=1)
    db = e.open(0)
    ost = oracle_ffi.Store(olib, 1)
    good0 = PyBatch().put(b"a", b"1").data()
    assert db.handle_replicate_response(good0)
    assert ost.apply(0, good0)
    e.flush()
    bad = bytearray(PyBatch().put(b"x", b"y").data())
    bad[8] = 3
    tail1 = PyBatch().put(b"b", b"2").data()
    tail2 = PyBatch().put(b"c", b"3").data()
    # stage bad + two clean updates BEFORE any flush: same buffer, but the
    # engine may split them across ticks; either way detection happens at
    # flush and both tails are stale
    assert db.handle_replicate_response(bytes(bad))
    assert db.handle_replicate_response(tail1)
    assert db.handle_replicate_response(tail2)
    assert not ost.apply(0, bytes(bad))
    e.flush()
    assert db.latest_seq() == 1
    assert db.get(b"b") is None and db.get(b"c") is None
    # fail-once, then the caller re-pulls the SAME updates from durable+1
    assert not db.handle_replicate_response(tail1)
    for repb in (tail1, tail2):
        assert db.handle_replicate_response(repb)
        assert ost.apply(0, repb)
    e.flush()
    assert db.latest_seq() == ost.latest_seq(0) == 3
    for k in (b"a", b"b", b"c"):
        assert db.get(k) == ost.get(0, k)
    assert db.checksum() == olib_checksum(olib, ost)
    e.close()


def test_cf_range_delete_parity(olib):
    """cf!=0 DeleteRange — begin AND end keys are
    cf-prefixed consistently in the engine store and the oracle, so a cf-1
    range tombstone covers cf-1 keys only. Checks host Get, device multiget
    and the device-computed checksum against the oracle."""
    import struct as _s

    def cfkey(cf, k):
        return _s.pack("<I", cf) + k

    e = ra.Engine(nshards=1)
    db = e.open(0)
    ost = oracle_ffi.Store(olib, 1)
    reps = [
        PyBatch().put(b"bb", b"v0").data(),
        PyBatch().cf_put(1, b"bb", b"v1").cf_put(2, b"bb", b"v2").data(),
        PyBatch().cf_put(1, b"aa", b"va").cf_put(1, b"cc", b"vc").data(),
        PyBatch().cf_delete_range(1, b"aa", b"cc").data(),
        PyBatch().cf_put(1, b"ab", b"post").data(),  # written above the tomb
    ]
    for rep in reps:
        assert db.handle_replicate_response(rep)
        assert ost.apply(0, rep)
    e.flush()
    assert db.latest_seq() == ost.latest_seq(0)
    probes = [b"bb", cfkey(1, b"bb"), cfkey(2, b"bb"), cfkey(1, b"aa"),
              cfkey(1, b"cc"), cfkey(1, b"ab"), b"aa"]
    for k in probes:
        assert db.get(k) == ost.get(0, k), k.hex()
    # semantic spot checks (not just parity)
    assert db.get(b"bb") == b"v0"               # cf0 survives cf1 range del
    assert db.get(cfkey(1, b"bb")) is None      # cf1 in range: deleted
    assert db.get(cfkey(2, b"bb")) == b"v2"     # cf2 survives
    assert db.get(cfkey(1, b"cc")) == b"vc"     # end exclusive
    assert db.get(cfkey(1, b"ab")) == b"post"   # newer than tombstone
    # device multiget agrees with host get
    for k, val in zip(probes, db.multiget(probes)):
        assert val == db.get(k), k.hex()
    # bit-level store parity
    assert db.checksum() == olib_checksum(olib, ost)
    e.close()


def test_drain_ring_mode_parity(olib):
    """Ring store + drain_host: the production drain shape — device store
    recycles, pinned host arenas own the runs (k_drain path). Get, checksum
    and seq must equal the oracle; multiget routes through the host runs."""
    nshards, nupd = 16, 8000
    arena, used, descs = ra.gen_stream(nshards=nshards, n_updates=nupd,
                                       key_len=16, val_len=96, kind=2,
                                       seed=77)
    raw = bytes(arena)[:used]
    ost = oracle_apply_stream(olib, nshards, raw, descs, nupd, merge_op=1)
    e = ra.Engine(nshards=nshards, store_ring=1, drain_host=1, merge_op=1)
    rep = e.upload(C.cast(arena, C.POINTER(C.c_uint8)), used, descs, nupd)
    for first in range(0, nupd, 2000):  # several ticks -> several arenas
        rep.tick(first, 2000)
    rep.sync()
    keys = collect_keys(olib, raw, descs, 4000)
    check_parity(e, ost, keys, range(nshards))
    for s in range(0, nshards, 4):
        db = e.open(s)
        assert db.checksum() == olib_checksum(olib, ost, s), s
        ks = keys.get(s, [])[:16]
        if ks:
            for k, v in zip(ks, db.multiget(ks)):
                assert v == ost.get(s, k), (s, k)
        db.close()
    e.close()


def test_drain_streaming_with_corruption(olib):
    """Drained runs + the corruption/rollback machinery interact correctly:
    a corrupt batch mid-stream must not leave phantom drained content."""
    e = ra.Engine(nshards=1, store_ring=1, drain_host=1)
    db = e.open(0)
    ost = oracle_ffi.Store(olib, 1)
    good = [PyBatch().put(f"k{i}".encode(), f"v{i}".encode()).data()
            for i in range(6)]
    bad = bytearray(PyBatch().put(b"x", b"y").data())
    bad[8] = 5
    for rep in good[:3]:
        assert db.handle_replicate_response(rep)
        assert ost.apply(0, rep)
    e.flush()
    assert db.handle_replicate_response(bytes(bad))
    assert not ost.apply(0, bytes(bad))
    e.flush()
    assert not db.handle_replicate_response(good[3])  # fail-once
    for rep in good[3:]:
        assert db.handle_replicate_response(rep)
        assert ost.apply(0, rep)
    e.flush()
    assert db.latest_seq() == ost.latest_seq(0) == 6
    for i in range(6):
        assert db.get(f"k{i}".encode()) == f"v{i}".encode()
    assert db.get(b"x") is None
    assert db.checksum() == olib_checksum(olib, ost)
    e.close()


def test_multiget_mixed_host_device_runs(olib):
    """Leader-written host runs and GPU-applied device runs interleave in one
    shard; gra_multiget now merges both halves by seq instead of falling
    back wholesale. Every outcome must equal gra_get (which equals the
    oracle)."""
    e = ra.Engine(nshards=1)
    db = e.open(0)
    ost = oracle_ffi.Store(olib, 1)

    def both(rep, via):
        if via == "host":
            db.write_leader(rep)
        else:
            assert db.handle_replicate_response(rep)
            e.flush()
        assert ost.apply(0, rep)

    both(PyBatch().put(b"a", b"dev1").put(b"b", b"dev1").data(), "gpu")
    both(PyBatch().put(b"a", b"host2").delete(b"b").data(), "host")
    both(PyBatch().put(b"c", b"dev3").put(b"b", b"dev3").data(), "gpu")
    both(PyBatch().delete(b"c").merge(b"m", b"x").data(), "host")
    both(PyBatch().merge(b"m", b"y").delete_range(b"a", b"ab").data(), "gpu")
    probes = [b"a", b"b", b"c", b"m", b"zz"]
    assert db.latest_seq() == ost.latest_seq(0)
    for k in probes:
        assert db.get(k) == ost.get(0, k), k
    for k, v in zip(probes, db.multiget(probes)):
        assert v == ost.get(0, k), k
    # newer device run over an older host terminator and vice versa
    both(PyBatch().put(b"b", b"dev-final").data(), "gpu")
    both(PyBatch().put(b"a", b"host-final").data(), "host")
    for k, v in zip([b"a", b"b"], db.multiget([b"a", b"b"])):
        assert v == ost.get(0, k), k
    e.close()


def test_multiget_hashjoin_overflow_fallback(olib):
    """Many versions of few keys make the candidate list exceed its cap
    (4*nq+1024): the hash-join must detect the overflow and retry via the
    per-query scan kernel with identical results."""
    e = ra.Engine(nshards=1)
    db = e.open(0)
    ost = oracle_ffi.Store(olib, 1)
    # 3 keys x 2000 versions each -> any query of them yields ~2000
    # candidates; 40 queries -> ~80000 >> 4*40+1024
    for i in range(2000):
        rep = (PyBatch().put(b"hot0", f"v{i}".encode())
               .put(b"hot1", f"w{i}".encode())
               .put(b"hot2", f"x{i}".encode()).data())
        assert db.handle_replicate_response(rep)
        assert ost.apply(0, rep)
    e.flush()
    probes = [f"hot{i % 3}".encode() for i in range(40)] + [b"cold"]
    for k, v in zip(probes, db.multiget(probes)):
        assert v == ost.get(0, k), k
    e.close()


def test_h2d_staged_corruption_recovery(olib):
    """Corruption inside an h2d-staged window: the staged path carries no
    per-update counts, so recovery must still poison the shard and roll
    back to the durable boundary (whole-group drop)."""
    e = ra.Engine(nshards=2)
    dbs = [e.open(s) for s in range(2)]
    good = [PyBatch().put(f"g{i}".encode(), b"v").data() for i in range(4)]
    bad = bytearray(PyBatch().put(b"x", b"y").data())
    bad[8] = 3
    blobs = [(0, good[0]), (1, good[1]), (0, bytes(bad)), (0, good[2]),
             (1, good[3])]
    used = sum(len(b) for _s, b in blobs)
    pin = e.pin_alloc(used + 64)
    descs = (ra.ffi.GraUpdateDesc * len(blobs))()
    off = 0
    # shard-grouped order within the window (shard 0 first)
    ordered = sorted(range(len(blobs)), key=lambda i: blobs[i][0])
    base = C.cast(pin, C.c_void_p).value
    for j, i in enumerate(ordered):
        s, b = blobs[i]
        C.memmove(base + off, b, len(b))
        descs[j].shard, descs[j].len, descs[j].off, descs[j].ts = \
            s, len(b), off, 0
        off += len(b)
    rep = e.upload(pin, used, descs, len(blobs))
    rep.tick_h2d(0, len(blobs))
    rep.sync()
    # shard 0: g0 applied; bad drops the REST of the shard-0 group
    # (no counts on the staged path -> whole-group truncation at the
    # first corrupt update; g0 precedes it)
    assert dbs[0].get(b"g0") == b"v"
    assert dbs[0].get(b"g2") is None
    assert dbs[0].latest_seq() == 1
    # shard 1 untouched by shard 0's corruption
    assert dbs[1].latest_seq() == 2
    assert dbs[1].get(b"g1") == b"v" and dbs[1].get(b"g3") == b"v"
    # fail-once then recovery on shard 0
    assert not dbs[0].handle_replicate_response(good[2])
    assert dbs[0].handle_replicate_response(good[2])
    e.flush()
    assert dbs[0].latest_seq() == 2 and dbs[0].get(b"g2") == b"v"
    e.close()
