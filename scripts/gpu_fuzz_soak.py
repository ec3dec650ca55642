"""GPU differential fuzz soak: random WAL-legal batches (all record
types incl. CF variants), random single-byte corruption, periodic poison
recovery and periodic full-store checksum comparison — engine vs oracle,
per shard, for a wall-clock budget. Run on a GPU box:

    python scripts/gpu_fuzz_soak.py [seconds] [seed] [plain|drain|concat]

Exit 0 = every probe agreed. Totals printed for the log.

Status: the `plain` mode ran clean at scale (profiles/r02/
gpu_fuzz_soak2.log: 827K batches / 40K checksum checks) and caught the
verbatim-Put semantics divergence on its first run. The `drain` and
`concat` variants were added late in round 2 and could not be executed:
the GPU pool lost three boxes in the setup phase (before the command
ran) on consecutive calls, closing gpurun for the round — the variants
are untested on hardware, flagged here rather than silently.
"""
import os
import random
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

import oracle_ffi  # noqa: E402
import rocksplicator_amd as ra  # noqa: E402
from pywb import PyBatch  # noqa: E402


def main(seconds=300, seed=0x50AC, mode="plain"):
    """mode: plain | drain (drain_host runs, arena-backed reads) |
    concat (merge_op 0)."""
    rng = random.Random(seed)
    olib = oracle_ffi.load()
    nshards = 12
    merge_op = 0 if mode == "concat" else 1
    e = ra.Engine(nshards=nshards, merge_op=merge_op,
                  drain_host=1 if mode == "drain" else 0)
    dbs = [e.open(s) for s in range(nshards)]
    ost = oracle_ffi.Store(olib, nshards, merge_op=merge_op)
    keys = [f"key{i:04d}".encode() for i in range(300)]
    one = (1).to_bytes(8, "little")
    deadline = time.monotonic() + seconds
    batches = mutations = checks = rounds = 0
    while time.monotonic() < deadline:
        rounds += 1
        if mode == "drain" and rounds % 20 == 0:
            # drain mode accumulates pinned host arenas per tick (runs own
            # them); bound the footprint by recycling the engine — which
            # also exercises the restart story mid-soak
            for db in dbs:
                db.close()
            e.close()
            ost = oracle_ffi.Store(olib, nshards, merge_op=merge_op)
            e = ra.Engine(nshards=nshards, merge_op=merge_op, drain_host=1)
            dbs = [e.open(s) for s in range(nshards)]
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
                mutations += 1
            rep = bytes(rep)
            acc_e = dbs[s].handle_replicate_response(rep, 1)
            acc_o = ost.apply(s, rep, 1)
            assert acc_e, "submit path never refuses pre-validation"
            batches += 1
            if not acc_o:
                # oracle rejected: engine detects at flush; resync the
                # poison signal (reference fail-once cadence)
                e.flush()
                probe = PyBatch().put(b"resync", b"1").data()
                ok = dbs[s].handle_replicate_response(probe, 1)
                if not ok:  # fail-once consumed; re-apply
                    assert dbs[s].handle_replicate_response(probe, 1)
                assert ost.apply(s, probe, 1)
        e.flush()
        for s in range(nshards):
            assert dbs[s].latest_seq() == ost.latest_seq(s), (rounds, s)
            assert dbs[s].checksum() == olib.orc_shard_checksum(ost.h, s), \
                (rounds, s)
            checks += 1
        # sampled multiget parity (device reads incl. cf-prefixed keys)
        s = rng.randrange(nshards)
        probes = rng.sample(keys, 24)
        probes += [(cf).to_bytes(4, "little") + rng.choice(keys)
                   for cf in (1, 2, 3)]
        for k, v in zip(probes, dbs[s].multiget(probes)):
            assert v == ost.get(s, k), (rounds, s, k)
    print(f"gpu fuzz soak clean [{mode}]: {rounds} rounds, {batches} "
          f"batches ({mutations} mutated), {checks} shard checksum checks, "
          f"seed {seed:#x}")
    e.close()
    return 0


if __name__ == "__main__":
    sys.exit(main(int(sys.argv[1]) if len(sys.argv) > 1 else 300,
                  int(sys.argv[2], 0) if len(sys.argv) > 2 else 0x50AC,
                  sys.argv[3] if len(sys.argv) > 3 else "plain"))
