#!/usr/bin/env python3
"""thrift_follower — the drop-in story end to end, over the reference's
own wire format.

A "reference-shaped" leader serves `Replicator.replicate` (THeader +
compact protocol — what an unmodified rocksplicator leader speaks through
fbthrift's HeaderClientChannel), and a follower built on this framework
pulls from it exactly like ReplicatedDB::pullFromUpstream: request from
LatestSequenceNumber()+1, apply each Update via HandleReplicateResponse,
long-poll when caught up. The follower runs in drain-host mode, so every
applied tick's runs are streamed back into pinned host memtable arenas —
the north star's full loop: wire -> pinned staging -> HBM kernels ->
host memtable.

Run on a GPU box:  python examples/thrift_follower.py [n_writes]
"""
import os
import random
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import rocksplicator_amd as ra
from rocksplicator_amd import thrift_wire as tw
from rocksplicator_amd.replicator import pull_once

NSHARDS = 4


def main(n_writes=5000):
    rng = random.Random(11)
    leader = ra.Engine(nshards=NSHARDS, merge_op=ra.MERGE_U64ADD, retain_log=1)
    follower = ra.Engine(nshards=NSHARDS, merge_op=ra.MERGE_U64ADD,
                         drain_host=1)
    srv = tw.ThriftUpdateServer()
    ldbs, fdbs, remotes = [], [], []
    for s in range(NSHARDS):
        ldbs.append(leader.open(s))
        fdbs.append(follower.open(s))
        srv.register(f"shard{s}", ldbs[s])
        remotes.append(tw.ThriftRemoteUpstream("127.0.0.1", srv.port,
                                               f"shard{s}", max_wait_ms=200))

    one = (1).to_bytes(8, "little")
    t0 = time.perf_counter()
    for i in range(n_writes):
        s = rng.randrange(NSHARDS)
        b = ra.Batch()
        if rng.random() < 0.6:
            b.merge(f"ctr{rng.randrange(32)}".encode(), one)
        else:
            b.put(f"k{rng.randrange(200)}".encode(), rng.randbytes(96))
        ldbs[s].write_leader(b.data())
        if i % 64 == 0:
            srv.notify_write()
    srv.notify_write()

    # follower catch-up over the thrift wire
    for s in range(NSHARDS):
        while pull_once(remotes[s], fdbs[s]):
            pass
    follower.flush()
    dt = time.perf_counter() - t0

    ok = True
    for s in range(NSHARDS):
        if fdbs[s].latest_seq() != ldbs[s].latest_seq():
            ok = False
        if fdbs[s].checksum() != ldbs[s].checksum():
            ok = False
        for i in range(32):
            k = f"ctr{i}".encode()
            if fdbs[s].get(k) != ldbs[s].get(k):
                ok = False
    print(f"{n_writes} leader writes replicated over the thrift wire in "
          f"{dt:.2f}s; follower drained to host memtables: "
          f"{'OK (seqs, counters and store checksums equal)' if ok else 'MISMATCH'}")
    for r in remotes:
        r.close()
    srv.close()
    leader.close()
    follower.close()
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main(int(sys.argv[1]) if len(sys.argv) > 1 else 5000))
