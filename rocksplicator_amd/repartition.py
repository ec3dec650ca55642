"""Cross-shard repartition (BASELINE config #4): update blobs are re-homed
to their owner rank by an all-to-all over RCCL/xGMI before apply
(SURVEY §8e — the path's one real exchange step; a permutation, so
all-to-all is the semantically right collective and spreads traffic over
all 7 xGMI links).

Streams are deterministic per (source, owner) pair, so the receiver
REGENERATES the descriptors of what it will receive instead of exchanging
metadata — only blob bytes cross the wire.
"""
import ctypes as C

from . import ffi


def pair_seed(base, src, dst):
    return base + 7919 * src + 104729 * dst


def gen_chunk(src, dst, world, nshards, n_updates, key_len, val_len, kind,
              seed_base):
    """The deterministic update stream source rank `src` produces for owner
    rank `dst`'s local shards. Returns (raw_bytes, used, descs)."""
    arena, used, descs = ffi.gen_stream(
        nshards=nshards, n_updates=n_updates, key_len=key_len,
        val_len=val_len, kind=kind, seed=pair_seed(seed_base, src, dst))
    return arena, used, descs


def build_send(rank, world, nshards, tick_updates, key_len, val_len, kind,
               seed_base):
    """All chunks rank sends, concatenated in owner order.
    Returns (send_bytes: bytearray, in_splits: [bytes per owner])."""
    per_owner = tick_updates // world
    buf = bytearray()
    splits = []
    for w in range(world):
        arena, used, _ = gen_chunk(rank, w, world, nshards, per_owner,
                                   key_len, val_len, kind, seed_base)
        buf += bytes(arena)[:used]
        splits.append(used)
    return buf, splits


def expected_recv(rank, world, nshards, tick_updates, key_len, val_len, kind,
                  seed_base):
    """What this rank receives: per-source sizes (out_splits), the rebased
    descriptor array over the concatenated recv buffer, and (for tests) the
    expected bytes."""
    per_owner = tick_updates // world
    out_splits = []
    descs_all = []
    expect = bytearray()
    base = 0
    for s in range(world):
        arena, used, descs = gen_chunk(s, rank, world, nshards, per_owner,
                                       key_len, val_len, kind, seed_base)
        out_splits.append(used)
        expect += bytes(arena)[:used]
        for i in range(per_owner):
            d = descs[i]
            descs_all.append((d.shard, d.len, d.off + base, d.ts))
        base += used
    n = len(descs_all)
    cdescs = (ffi.GraUpdateDesc * n)()
    for i, (shard, ln, off, ts) in enumerate(descs_all):
        cdescs[i] = ffi.GraUpdateDesc(shard, ln, off, ts)
    return out_splits, cdescs, n, expect


def upload_dev(engine, dev_ptr, nbytes, cdescs, n):
    """Zero-copy replay over a device-resident arena (e.g. a torch cuda
    tensor that received the all-to-all output). Generator batches are all
    1-record (performance.cpp:139-142 shape), so counts are constant 1."""
    lib = engine.lib
    if not hasattr(lib, "_upload_dev_bound"):
        lib.gra_upload_dev.argtypes = [
            C.c_void_p, C.c_void_p, C.c_size_t, C.POINTER(ffi.GraUpdateDesc),
            C.c_uint64, C.POINTER(C.c_uint32), C.c_void_p,
        ]
        lib._upload_dev_bound = True
    counts = (C.c_uint32 * n)(*([1] * n))
    out = C.c_void_p()
    rc = lib.gra_upload_dev(engine.h, dev_ptr, nbytes, cdescs, n, counts,
                            C.byref(out))
    if rc != 0:
        raise RuntimeError(f"gra_upload_dev rc={rc}: {ffi.last_error(lib)}")
    return ffi.Replay(engine, out)
