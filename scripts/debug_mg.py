"""Deterministic repro of the gpu_fuzz_soak multiget divergence (seed
0x50AC, round 1): on mismatch, dump everything relevant."""
import os
import random
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

import oracle_ffi  # noqa: E402
import rocksplicator_amd as ra  # noqa: E402
from pywb import PyBatch  # noqa: E402

rng = random.Random(0x50AC)
olib = oracle_ffi.load()
nshards = 12
e = ra.Engine(nshards=nshards, merge_op=1)
dbs = [e.open(s) for s in range(nshards)]
ost = oracle_ffi.Store(olib, nshards, merge_op=1)
keys = [f"key{i:04d}".encode() for i in range(300)]
one = (1).to_bytes(8, "little")

for _ in range(rng.randrange(100, 400)):
    s = rng.randrange(nshards)
    b = PyBatch()
    for _ in range(rng.randrange(1, 5)):
        r = rng.random()
        k = rng.choice(keys)
        if r < 0.30:
            b.put(k, rng.randbytes(rng.randrange(0, 300)))
        elif r < 0.45:
            b.merge(k, one)
        elif r < 0.55:
            b.delete(k)
        elif r < 0.60:
            b.single_delete(k)
        elif r < 0.65:
            lo, hi = sorted([rng.choice(keys), rng.choice(keys)])
            if lo != hi:
                b.delete_range(lo, hi)
        elif r < 0.75:
            b.cf_put(rng.randrange(1, 4), k, rng.randbytes(24))
        elif r < 0.80:
            b.cf_delete(rng.randrange(1, 4), k)
        elif r < 0.85:
            lo, hi = sorted([rng.choice(keys), rng.choice(keys)])
            if lo != hi:
                b.cf_delete_range(rng.randrange(1, 4), lo, hi)
        elif r < 0.92:
            b.log_data(rng.randbytes(rng.randrange(0, 40)))
        else:
            b.noop()
    rep = bytearray(b.data())
    if rng.random() < 0.08:
        rep[rng.randrange(len(rep))] ^= rng.randrange(1, 256)
    rep = bytes(rep)
    acc_e = dbs[s].handle_replicate_response(rep, 1)
    acc_o = ost.apply(s, rep, 1)
    assert acc_e
    if not acc_o:
        e.flush()
        probe = PyBatch().put(b"resync", b"1").data()
        ok = dbs[s].handle_replicate_response(probe, 1)
        if not ok:
            assert dbs[s].handle_replicate_response(probe, 1)
        assert ost.apply(s, probe, 1)
e.flush()
for s in range(nshards):
    assert dbs[s].latest_seq() == ost.latest_seq(s), s
    assert dbs[s].checksum() == olib.orc_shard_checksum(ost.h, s), s
s = rng.randrange(nshards)
probes = rng.sample(keys, 24)
probes += [(cf).to_bytes(4, "little") + rng.choice(keys) for cf in (1, 2, 3)]
mg = dbs[s].multiget(probes)
bad = 0
for k, v in zip(probes, mg):
    want = ost.get(s, k)
    host = dbs[s].get(k)
    if v != want or host != want:
        bad += 1
        print(f"DIVERGE shard={s} key={k!r}: multiget={v!r} host_get={host!r} "
              f"oracle={want!r}")
print(f"done: {bad} divergences on shard {s}")
if bad:
    # single-key multiget of the diverging keys (isolates batching effects)
    for k, v in zip(probes, mg):
        want = ost.get(s, k)
        if v != want:
            solo = dbs[s].multiget([k])
            print(f"  solo multiget {k!r} -> {solo[0]!r} (oracle {want!r})")
sys.exit(1 if bad else 0)
