#!/usr/bin/env python3
"""Long-running differential fuzz campaign (CPU-only, no GPU needed).

Drives the same differential properties the test suite checks, but at
campaign scale: random WriteBatch op streams through the three builders
(product / pywb / oracle round-trip), the oracle store vs the independent
event-log model, byte-mutation corruption handling, and snappy cross-impl
round-trips. The pytest suites run bounded versions; this runs until the
requested case count (default 20k; CI-ish) or forever with --cases 0.

Usage:  python scripts/fuzz_campaign.py [--cases N] [--seed S]
Exit 0 = no divergence found.
"""
import argparse
import os
import random
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

import oracle_ffi  # noqa: E402
from pywb import PyBatch  # noqa: E402
from test_store_model import ModelStore, KEYS  # noqa: E402

import rocksplicator_amd as ra  # noqa: E402


def rand_ops(rng, keys, n):
    ops = []
    for _ in range(n):
        kind = rng.choice(("put", "put", "merge", "delete", "single_delete",
                           "delete_range", "log_data", "cf_put", "cf_delete",
                           "cf_single_delete", "cf_merge", "cf_delete_range"))
        k = rng.choice(keys)
        if kind in ("put", "merge"):
            ops.append((kind, k, rng.randbytes(rng.randrange(0, 64))))
        elif kind == "delete_range":
            ops.append((kind, k, rng.choice(keys)))
        elif kind == "log_data":
            ops.append((kind, rng.randbytes(rng.randrange(0, 32)), b""))
        elif kind in ("cf_put", "cf_merge"):
            ops.append((kind, (rng.randrange(1, 4), k),
                        rng.randbytes(rng.randrange(0, 48))))
        elif kind in ("cf_delete", "cf_single_delete"):
            ops.append((kind, (rng.randrange(1, 4), k), b""))
        elif kind == "cf_delete_range":
            ops.append((kind, (rng.randrange(1, 4), k, rng.choice(keys)), b""))
        else:
            ops.append((kind, k, b""))
    return ops


def build(ops, seq=0):
    """Both builders (pywb + product gra_wb_*) for every op kind,
    byte-compared."""
    pb = PyBatch(seq=seq)
    gb = ra.Batch().set_seq(seq)
    for kind, a, b in ops:
        if kind in ("put", "merge"):
            getattr(pb, kind)(a, b)
            getattr(gb, kind)(a, b)
        elif kind == "delete_range":
            pb.delete_range(a, b)
            gb.delete_range(a, b)
        elif kind == "log_data":
            pb.log_data(a)
            gb.log_data(a)
        elif kind in ("cf_put", "cf_merge"):
            getattr(pb, kind)(a[0], a[1], b)
            getattr(gb, kind)(a[0], a[1], b)
        elif kind in ("cf_delete", "cf_single_delete"):
            getattr(pb, kind)(a[0], a[1])
            getattr(gb, kind)(a[0], a[1])
        elif kind == "cf_delete_range":
            pb.cf_delete_range(a[0], a[1], a[2])
            gb.cf_delete_range(a[0], a[1], a[2])
        else:
            getattr(pb, kind)(a)
            getattr(gb, kind)(a)
    return pb.data(), gb.data()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--cases", type=int, default=20000)
    ap.add_argument("--seed", type=lambda s: int(s, 0), default=0xF022)
    args = ap.parse_args()
    rng = random.Random(args.seed)
    lib = oracle_ffi.load()

    case = 0
    while args.cases == 0 or case < args.cases:
        case += 1
        keys = [rng.randbytes(rng.randrange(1, 24)) or b"k"
                for _ in range(rng.randrange(2, 8))] + [rng.choice(KEYS)]
        merge_op = rng.randrange(2)
        ost = oracle_ffi.Store(lib, 1, merge_op=merge_op)
        model = ModelStore(merge_op)
        for _batch in range(rng.randrange(1, 6)):
            ops = rand_ops(rng, keys, rng.randrange(1, 8))
            rep_py, rep_c = build(ops)
            assert rep_py == rep_c, ("builder divergence", case, ops)
            if rng.random() < 0.25:  # corrupt sometimes
                rep = bytearray(rep_c)
                if rep and rng.random() < 0.5:
                    rep[rng.randrange(len(rep))] ^= rng.randrange(1, 256)
                else:
                    rep = rep[:rng.randrange(len(rep))]
                before = ost.latest_seq(0)
                if not ost.apply(0, bytes(rep)):
                    assert ost.latest_seq(0) == before, ("seq leak", case)
                    continue
                # mutation happened to stay valid: mirror it in the model
                seq, cnt, recs = oracle_ffi.decode(lib, bytes(rep))
                model.latest += cnt
                # model can't interpret arbitrary mutated records; restart
                # both stores to keep them comparable
                ost = oracle_ffi.Store(lib, 1, merge_op=merge_op)
                model = ModelStore(merge_op)
                continue
            assert ost.apply(0, rep_c), ("valid batch rejected", case, ops)
            model.apply(ops)
        assert ost.latest_seq(0) == model.latest, ("latest_seq", case)
        probes = list(keys) + [b"\x00absent"]
        for k in keys:  # cf-namespaced probes (stored form = [cfLE4|key])
            for cf in (1, 2, 3):
                probes.append(ModelStore._cfkey(cf, k))
        for k in probes:
            got, want = ost.get(0, k), model.get(k)
            assert got == want, ("get divergence", case, k, got, want)
        if case % 5000 == 0:
            print(f"{case} cases clean", flush=True)
    print(f"campaign clean: {case} cases (seed {args.seed:#x})")
    return 0


if __name__ == "__main__":
    sys.exit(main())
