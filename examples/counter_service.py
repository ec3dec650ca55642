#!/usr/bin/env python3
"""counter_service — the reference's worked example, on the GPU engine.

Mirrors examples/counter_service of the reference (counter_handler.cpp:
152-158 and 212-218: thrift setCounter/incrCounter doing WriteBatch
Put/Merge through ApplicationDB::Write with a custom merge operator, shards
replicated leader -> follower) over this framework's pieces: the
`Replicator` registry with LEADER/FOLLOWER roles and pull threads, u64-add
merge on device, mode-1 ACK'd writes, reads served from the follower, and
a full-store checksum comparing the two replicas at the end.

Run on a GPU box:  python examples/counter_service.py [n_ops]
(The thrift RPC front end is out of tier scope — this is the service's
data plane; swap `incr`/`get_counter` bodies into any RPC handler.)
"""
import os
import random
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import rocksplicator_amd as ra
from rocksplicator_amd import replicator as rp

NSHARDS = 8


def shard_of(name):          # the router's job in the reference
    return hash(name) % NSHARDS


def main(n_ops=20000):
    leader_eng = ra.Engine(nshards=NSHARDS, merge_op=ra.MERGE_U64ADD,
                           retain_log=1)
    follower_eng = ra.Engine(nshards=NSHARDS, merge_op=ra.MERGE_U64ADD)
    leader = rp.Replicator(leader_eng)
    follower = rp.Replicator(follower_eng)
    for s in range(NSHARDS):
        lrs = leader.add_db(f"counters{s:05d}", rp.LEADER)
        follower.add_db(f"counters{s:05d}", rp.FOLLOWER, upstream_db=lrs.db)

    def incr(name, delta):   # ≅ incrCounter (counter_handler.cpp:212-218)
        rep = ra.Batch().merge(name.encode(),
                               delta.to_bytes(8, "little")).data()
        leader.write(f"counters{shard_of(name):05d}", rep, mode=1)

    def set_counter(name, value):  # ≅ setCounter (:152-158)
        rep = ra.Batch().put(name.encode(),
                             value.to_bytes(8, "little")).data()
        leader.write(f"counters{shard_of(name):05d}", rep, mode=1)

    def get_counter(replicator, name):
        v = replicator.get(f"counters{shard_of(name):05d}").db.get(
            name.encode())
        return int.from_bytes(v, "little") if v else 0

    rng = random.Random(1)
    model = {}
    for i in range(n_ops):
        name = f"counter_{rng.randrange(64)}"
        if rng.random() < 0.1:
            v = rng.randrange(1 << 20)
            set_counter(name, v)
            model[name] = v
        else:
            d = rng.randrange(1, 100)
            incr(name, d)
            model[name] = model.get(name, 0) + d
    leader_eng.flush()

    # wait for the follower pull threads to drain (do NOT pull manually too:
    # two pullers racing the same `since` would double-apply)
    import time
    deadline = time.time() + 60
    for s in range(NSHARDS):
        lrs = leader.get(f"counters{s:05d}")
        frs = follower.get(f"counters{s:05d}")
        while (frs.db.latest_seq() < lrs.db.latest_seq()
               and time.time() < deadline):
            time.sleep(0.01)
        assert frs.db.latest_seq() == lrs.db.latest_seq(), s
    follower_eng.flush()

    bad = sum(1 for name, want in model.items()
              if get_counter(leader, name) != want
              or get_counter(follower, name) != want)
    csums_equal = all(
        leader.get(f"counters{s:05d}").db.checksum()
        == follower.get(f"counters{s:05d}").db.checksum()
        for s in range(NSHARDS))
    print(f"{n_ops} ops over {len(model)} counters, {NSHARDS} shards: "
          f"{'OK' if not bad and csums_equal else 'MISMATCH'} "
          f"(replica checksums {'equal' if csums_equal else 'DIFFER'})")
    print(rp.dump_stats_text(leader).splitlines()[0])
    follower.close()  # downstream first (pull threads hold upstream handles)
    leader.close()
    return 0 if not bad and csums_equal else 1


if __name__ == "__main__":
    sys.exit(main(int(sys.argv[1]) if len(sys.argv) > 1 else 20000))
