"""Diagnose the stale-tick-after-corruption replay scenario on a GPU box:
prints per-step seq state and failure counters so we can see whether the
corrupt tick's error path ran at all. Scratch tool, not a test."""
import ctypes as C
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

import rocksplicator_amd as ra  # noqa: E402
from pywb import PyBatch  # noqa: E402

e = ra.Engine(nshards=1)
db = e.open(0)
b1 = PyBatch().put(b"g1", b"v1").data()
bad = bytearray(PyBatch().put(b"bad", b"bad").data())
bad[8] = 2
b3 = PyBatch().put(b"g3", b"v3").data()
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

for t in range(3):
    rep.tick(t, 1)
    rep.sync()  # ingest after EVERY tick so we see per-tick state
    c = db.counters()
    print(f"after tick {t}: latest_seq={db.latest_seq()} "
          f"failures={c['apply_failures']} applied_recs_visible="
          f"{[db.get(k) for k in (b'g1', b'bad', b'g3')]}")

print("g3 after all:", db.get(b"g3"))
e.close()

# second run: ticks enqueued back-to-back, ONE sync at the end (the failing
# test's shape)
e = ra.Engine(nshards=1)
db = e.open(0)
rep = e.upload(C.cast(arena, C.POINTER(C.c_uint8)), used, descs, 3)
rep.tick(0, 1)
rep.tick(1, 1)
rep.tick(2, 1)
rep.sync()
c = db.counters()
print(f"batched: latest_seq={db.latest_seq()} failures={c['apply_failures']} "
      f"g1={db.get(b'g1')} g3={db.get(b'g3')}")
e.close()
