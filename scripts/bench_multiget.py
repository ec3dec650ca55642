"""Point-read throughput over the device store (VERDICT r01 #7 benchmark):
one shard, 100K entries, batched gra_multiget queries. Run on a GPU box.

r01 baseline: ~1.1 M reads/s (linear scan, full key compare per entry).
r02: header-only kpref reject — target >=10x.
"""
import ctypes as C
import os
import random
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import rocksplicator_amd as ra  # noqa: E402

N_ENTRIES = 100_000
BATCH = 4096
ROUNDS = 40

e = ra.Engine(nshards=1, store_bytes=2 << 30)
db = e.open(0)
rng = random.Random(7)
keys = [f"user{i:012d}".encode() for i in range(N_ENTRIES)]

# load via replay (one big multi-update stream, 50-update batches)
lib = ra.load()
arena = bytearray()
descs = []
off = 0
i = 0
while i < N_ENTRIES:
    b = ra.Batch()
    for _ in range(min(50, N_ENTRIES - i)):
        b.put(keys[i], rng.randbytes(64))
        i += 1
    rep = b.data()
    descs.append((0, len(rep), off))
    arena += rep
    off += len(rep)
buf = (C.c_uint8 * (len(arena) + 64)).from_buffer_copy(bytes(arena) + b"\0" * 64)
ds = (ra.ffi.GraUpdateDesc * len(descs))()
for j, (s, ln, of) in enumerate(descs):
    ds[j].shard, ds[j].len, ds[j].off, ds[j].ts = s, ln, of, 0
r = e.upload(C.cast(buf, C.POINTER(C.c_uint8)), len(arena), ds, len(descs))
r.tick(0, len(descs))
r.sync()
print(f"loaded {N_ENTRIES} entries, latest_seq={db.latest_seq()}")

probe = [keys[rng.randrange(N_ENTRIES)] for _ in range(BATCH)]
miss = [f"nope{i:08d}".encode() for i in range(BATCH)]

# warmup + correctness (python helper path)
res = db.multiget(probe[:64], val_stride=128)
for k, v in zip(probe[:64], res):
    assert v == db.get(k), k


def prebuilt(qs, stride=128):
    kb = b"".join(qs)
    refs = (ra.ffi.GraKeyRef * len(qs))()
    o = 0
    for j, k in enumerate(qs):
        refs[j].off, refs[j].len = o, len(k)
        o += len(k)
    vals = C.create_string_buffer(len(qs) * stride)
    outs = (ra.ffi.GraGetResult * len(qs))()
    return kb, refs, vals, outs, stride


def run(label, qs, expect_found):
    kb, refs, vals, outs, stride = prebuilt(qs)
    lib.gra_multiget(db.h, len(qs), refs, kb, len(kb), vals, stride, outs)
    nf = sum(1 for j in range(len(qs)) if outs[j].status == 0)
    assert (nf == len(qs)) == expect_found, (label, nf)
    t0 = time.perf_counter()
    for _ in range(ROUNDS):
        rc = lib.gra_multiget(db.h, len(qs), refs, kb, len(kb), vals, stride,
                              outs)
        assert rc == 0
    dt = time.perf_counter() - t0
    print(f"{label}: {ROUNDS * len(qs) / dt / 1e6:.2f} M reads/s "
          f"({dt / ROUNDS * 1e3:.3f} ms/batch of {len(qs)})")


run("hit-heavy (C-ABI)", probe, True)
run("all-miss  (C-ABI)", miss, False)
run("hit-64k   (C-ABI)", [keys[rng.randrange(N_ENTRIES)]
                          for _ in range(65536)], True)
e.close()
