#!/usr/bin/env python3
"""bench.py — replicated-updates/sec on the MI355X apply path.

Workload (BASELINE.json configs[2], the configuration the metric is quoted
on): 1024 shards, 16 B keys / 1 KB values, Zipf-0.99 key skew, synthetic
deterministic streams. A "step" is one pipeline tick: decode + placement +
partition-copy of one batch of updates, inputs already resident in HBM
(replay path; the PCIe-inclusive staging rate is reported in DESIGN.md, never
as `value`).

Per the harness contract: N ranks = N GPUs (torchrun, one rank per GPU),
W untimed warmup steps, exactly K timed steps bracketed by barrier + device
sync on both sides, MAX time over ranks, rank 0 prints ONE JSON line.
`value` is whole-job updates/sec over all ranks. Shards are independent
per-shard streams sharded across ranks — weak scaling, no
data-path collective (the cross-shard repartition exchange is config #4,
a later row).
"""
import argparse
import ctypes as C
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

import rocksplicator_amd as ra  # noqa: E402

HBM_PEAK_GBPS = 8000.0  # gfx950 spec peak (MI355X_MICROARCH.md)


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=200)
    p.add_argument("--warmup", type=int, default=20)
    p.add_argument("--nshards", type=int, default=1024)
    p.add_argument("--key-len", type=int, default=16)
    p.add_argument("--val-len", type=int, default=1024)
    p.add_argument("--kind", type=int, default=1, help="0 uniform,1 zipf,2 mixed")
    p.add_argument("--tick-updates", type=int, default=819200)
    p.add_argument("--max-ticks-resident", type=int, default=6,
                   help="distinct ticks of data generated/uploaded; steps cycle over them")
    p.add_argument("--cpu-baseline", action="store_true", default=True)
    p.add_argument("--no-cpu-baseline", dest="cpu_baseline", action="store_false")
    p.add_argument("--cpu-sample-seconds", type=float, default=10.0)
    p.add_argument("--traffic-file", default=os.path.join(REPO, "profiles", "traffic.json"),
                   help="optional rocprofv3-derived per-launch HBM traffic (see profiles/)")
    p.add_argument("--h2d", action="store_true",
                   help="measure the PCIe-inclusive staging path instead (side report)")
    p.add_argument("--no-legs", action="store_true",
                   help="skip the h2d/drain side legs (clean-profile runs)")
    p.add_argument("--device-wrap", action="store_true",
                   help="TESTING ONLY: map rank devices modulo the visible "
                        "GPU count so a multi-rank run can be validated on "
                        "a 1-GPU box (never for real measurements)")
    p.add_argument("--snappy", action="store_true",
                   help="BASELINE config #5: Snappy-compressed payloads, "
                        "GPU decompress stage ahead of the decode walk")
    p.add_argument("--repartition", action="store_true",
                   help="BASELINE config #4: re-home update blobs to their "
                        "owner rank via RCCL all-to-all over xGMI inside "
                        "each timed step (at world=1: device self-copy, "
                        "same code path)")
    return p.parse_args()


def dist_env():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local = int(os.environ.get("LOCAL_RANK", str(rank)))
    return rank, world, local


def h2d_side_leg(eng, arena, used, descs, n_upd, args, steps=24, warmup=6):
    """the staging-inclusive leg, reported beside the
    HBM-resident headline on every default run. Blobs start in pinned host
    memory and are staged H2D inside every timed tick, double-buffered on a
    dedicated copy stream so tick N+1's PCIe copy overlaps tick N's kernels
    (the north star's 'pinned host ring buffers ... hipMemcpyAsync' path,
    rocksdb_wrapper.cpp:13-28 replaced end-to-end)."""
    ticks = max(1, min(2, n_upd // args.tick_updates))  # bound pinned memory
    n = ticks * args.tick_updates
    end = descs[n - 1].off + descs[n - 1].len
    pin = eng.pin_alloc(end)
    C.memmove(pin, arena, end)
    rep = eng.upload(pin, end, descs, n)

    def step(i):
        rep.tick_h2d((i % ticks) * args.tick_updates, args.tick_updates)

    for i in range(warmup):
        step(i)
    rep.sync()
    eng.stats_reset()
    t0 = time.perf_counter()
    for i in range(steps):
        step(warmup + i)
    rep.sync()
    wall = time.perf_counter() - t0
    st = eng.stats()
    blob_per_step = st.blob_bytes / max(st.ticks, 1)
    return {
        "value": steps * args.tick_updates / wall,
        "unit": "updates/s",
        "ms_per_step": wall * 1e3 / steps,
        "pcie_h2d_gbps": blob_per_step / (wall / steps) / 1e9,
        "h2d_copy_ms_per_step": st.h2d_ms / max(st.ticks, 1),
        "kernel_total_ms_per_step": st.total_ms / max(st.ticks, 1),
        "steps": steps,
        "note": "staging-inclusive side leg (pinned host -> HBM inside every "
                "timed tick, overlapped); headline value stays HBM-resident",
    }


def drain_side_leg(arena_p, descs, n_upd, args, steps=8, warmup=3):
    """the host-drain leg — apply into the device store ring
    AND stream every tick's headers+payload into pinned host arenas
    (k_drain on the copyout stream, overlapped with the next tick's
    kernels): the north star's literal 'drain sorted runs back to the host
    RocksDB memtables'. Reported beside the PCIe D2H bound."""
    ticks = max(1, min(2, n_upd // args.tick_updates))
    n = ticks * args.tick_updates
    end = descs[n - 1].off + descs[n - 1].len
    per_tick = end // ticks
    eng = ra.Engine(nshards=args.nshards, device=-1, store_ring=1,
                    drain_host=1,
                    store_bytes=min(per_tick * 14 + (1 << 30), 48 << 30))
    lib = ra.load()
    arena_need = per_tick + 48 * args.tick_updates + 4096
    rc = lib.gra_drain_prewarm(eng.h, arena_need, steps + warmup + 2)
    assert rc == 0, ra.ffi.last_error(lib)
    rep = eng.upload(arena_p, end, descs, n)
    for w in range(ticks):
        rep.prepare(w * args.tick_updates, args.tick_updates)

    def step(i):
        rep.tick((i % ticks) * args.tick_updates, args.tick_updates)

    for i in range(warmup):
        step(i)
    rep.sync()
    eng.stats_reset()
    t0 = time.perf_counter()
    for i in range(steps):
        step(warmup + i)
    rep.sync()
    wall = time.perf_counter() - t0
    st = eng.stats()
    d2h_bytes = st.payload_bytes + 24 * st.records  # arena bytes drained
    out = {
        "value": steps * args.tick_updates / wall,
        "unit": "updates/s",
        "ms_per_step": wall * 1e3 / steps,
        "d2h_gbps": d2h_bytes / wall / 1e9,
        "steps": steps,
        "note": "drain-host side leg: apply + k_drain of every tick's "
                "hdrs+payload into pinned host arenas on the copyout "
                "stream; host runs own the bytes (ring store recycles "
                "device side)",
    }
    eng.close()
    return out


def cpu_baseline_leg(raw_arena, used, descs, n, nshards, target_s, snappy=False):
    """Time the ORACLE applier (the CPU restatement of replicated_db.cpp:
    369-383 + rocksdb_wrapper.cpp:13-28) on the host cores — checker/baseline
    only, never the measured GPU path."""
    sys.path.insert(0, os.path.join(REPO, "tests"))
    import oracle_ffi

    olib = oracle_ffi.load()
    ncores = os.cpu_count() or 1
    # convert descs to oracle desc layout (identical struct layout)
    ods = (oracle_ffi.OrcUpdateDesc * n)()
    C.memmove(ods, descs, C.sizeof(oracle_ffi.OrcUpdateDesc) * n)

    bench_fn = (olib.orc_cpu_snappy_apply_bench if snappy
                else olib.orc_cpu_apply_bench)

    def run(sample_n, threads, fn):
        st = olib.orc_store_create(nshards, 0)
        try:
            secs = fn(st, C.cast(raw_arena, C.c_void_p), ods, sample_n, threads)
        finally:
            olib.orc_store_destroy(st)
        return secs

    # bounded sample: repeat full-stream applies until ~target_s of CPU work
    total_n, total_s, reps = 0, 0.0, 0
    sample_n = min(n, 2_000_000)
    while total_s < target_s and reps < 64:
        secs = run(sample_n, ncores, bench_fn)
        total_n += sample_n
        total_s += secs
        reps += 1
    rate = total_n / total_s if total_s > 0 else 0.0
    out = {
        "value": rate,
        "unit": "updates/s",
        "cores": ncores,
        "kind": "port",
        "sample": f"{reps}x{sample_n} updates of the same synthetic stream"
                  + (" (snappy decompress+apply)" if snappy else "") +
                  f", {ncores} threads, {total_s:.2f}s total",
    }
    if not snappy:
        # WAL-on leg (both variants reported); `value` stays the
        # faster WAL-less leg — conservative for the GPU/CPU ratio
        wal_secs = run(sample_n, ncores, olib.orc_cpu_apply_bench_wal)
        out["wal_on_value"] = sample_n / wal_secs if wal_secs > 0 else 0.0
    return out


def run_repartition(args, rank, world, local, dist):
    """Config #4 path: per timed step, an all-to-all re-homes one tick of
    update blobs to their owner ranks over RCCL/xGMI, then the owner applies
    them from the received device buffer (zero-copy)."""
    import torch

    from rocksplicator_amd import repartition as rp

    torch.cuda.set_device(local)
    seed_base = 0xB0CC5EED
    send_bytes, in_splits = rp.build_send(
        rank, world, args.nshards, args.tick_updates, args.key_len,
        args.val_len, args.kind, seed_base)
    out_splits, cdescs, n_recv, _ = rp.expected_recv(
        rank, world, args.nshards, args.tick_updates, args.key_len,
        args.val_len, args.kind, seed_base)
    send_t = torch.frombuffer(bytearray(send_bytes), dtype=torch.uint8).cuda()
    recv_t = torch.empty(sum(out_splits) + 16, dtype=torch.uint8, device="cuda")

    eng = ra.Engine(nshards=args.nshards, device=-1, store_ring=1,
                    store_bytes=min(int(sum(out_splits) * 2.6) + (1 << 30), 48 << 30))
    rep = rp.upload_dev(eng, recv_t.data_ptr(), sum(out_splits), cdescs, n_recv)

    def exchange():
        if dist is not None:
            import torch.distributed as td
            td.all_to_all_single(recv_t[:sum(out_splits)], send_t,
                                 out_splits, in_splits)
        else:  # world 1: self-exchange, same downstream path
            recv_t[:len(send_bytes)].copy_(send_t)
        torch.cuda.synchronize()

    def one_step(_i):
        exchange()
        rep.tick(0, n_recv)

    for i in range(args.warmup):
        one_step(i)
    rep.sync()
    eng.stats_reset()
    if dist:
        dist.barrier()
    t0 = time.perf_counter()
    for i in range(args.steps):
        one_step(i)
    rep.sync()
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    if dist:
        dist.barrier()
    wall = t1 - t0
    if dist:
        tw = torch.tensor([wall], dtype=torch.float64)
        dist.all_reduce(tw, op=dist.ReduceOp.MAX)
        wall = float(tw.item())
    return eng, wall, n_recv


def main():
    args = parse_args()
    rank, world, local = dist_env()
    if args.gpus > 1 and world == 1:
        # invoked without torchrun: self-launch one rank per GPU
        import subprocess
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={args.gpus}", "--master-addr", "127.0.0.1",
               "--master-port", "29711", os.path.abspath(__file__),
               *[a for a in sys.argv[1:]]]
        raise SystemExit(subprocess.call(cmd))
    dist = None
    if world > 1:
        import torch.distributed as td
        # gloo for control-plane barriers; nccl (RCCL) group for the
        # repartition all-to-all when enabled
        td.init_process_group(backend="nccl" if args.repartition else "gloo")
        dist = td

    if args.repartition:
        eng, wall, n_recv = run_repartition(args, rank, world, local, dist)
        st = eng.stats()
        value = args.steps * n_recv * world / wall
        result = {
            "metric": "replicated updates/sec (Put+Delete) at 1024 shards",
            "value": value,
            "unit": "updates/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": wall * 1e3 / args.steps,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "u8",
            "data": "synthetic",
            "config": {
                "workload": f"{args.nshards}shards/gpu_{args.key_len}Bkey_"
                            f"{args.val_len}Bval_repartition",
                "nshards_per_gpu": args.nshards,
                "tick_updates": args.tick_updates,
                "path": "rccl-alltoall-repartition + hbm apply",
                "parallelism": f"shard-parallel x{world} + all-to-all",
            },
            "kernels_ms_per_tick": {
                "copy": st.copy_ms / max(st.ticks, 1),
                "total": st.total_ms / max(st.ticks, 1),
            },
            # secondary leg by design: the exchange step of
            # config #4. The headline line (no --repartition) carries the
            # roofline and cpu_baseline objects; this one measures the
            # RCCL-all-to-all + zero-copy apply path.
            "secondary": True,
        }
        if rank == 0:
            print(json.dumps(result))
        eng.close()
        if dist:
            dist.destroy_process_group()
        return

    # --- generate the replay stream (untimed) ---
    n_ticks_data = min(args.warmup + args.steps, args.max_ticks_resident)
    n_upd = args.tick_updates * n_ticks_data
    worst = n_upd * (23 + args.key_len + args.val_len + 16) + 4096
    arena = (C.c_uint8 * worst)()
    arena_p = C.cast(arena, C.POINTER(C.c_uint8))
    descs = (ra.ffi.GraUpdateDesc * n_upd)()
    g = ra.ffi.GraGenOpts(args.nshards, args.key_len, args.val_len, args.kind,
                          1 << 24, 0.99, 0xB0CC5EED + 1000 * rank,
                          1 if args.snappy else 0, 0)
    used = C.c_size_t()
    rc = ra.load().gra_gen_stream(C.byref(g), n_upd, arena_p, worst,
                                  C.byref(used), descs, 0)
    assert rc == 0, f"gen rc={rc}"
    used = used.value
    blob_bytes_total = used

    if args.device_wrap:
        import torch
        local = local % max(1, torch.cuda.device_count())

    # --- engine + upload (untimed; inputs land in HBM) ---
    store_bytes = min(int(used * 1.3) + (1 << 30), 48 << 30)
    # staging must hold one tick's blob window (the h2d staged leg copies a
    # whole tick per step, double-buffered on the device side)
    staging_bytes = used // n_ticks_data + (64 << 20)
    eng = ra.Engine(nshards=args.nshards, device=local, store_ring=1,
                    store_bytes=store_bytes, staging_bytes=staging_bytes)
    if args.snappy:
        # transport compression (config #5): compress on host (untimed),
        # the GPU decompress stage runs inside every timed tick
        lib = ra.load()
        comp_cap = used + used // 4 + 64 * n_upd
        comp = (C.c_uint8 * comp_cap)()
        cdescs = (ra.ffi.GraUpdateDesc * n_upd)()
        ulens = (C.c_uint32 * n_upd)()
        comp_used = C.c_size_t()
        rc = lib.gra_snappy_compress_stream(
            arena_p, descs, n_upd, C.cast(comp, C.POINTER(C.c_uint8)),
            comp_cap, C.byref(comp_used), cdescs, ulens, os.cpu_count() or 8)
        assert rc == 0, f"compress rc={rc}"
        counts = (C.c_uint32 * n_upd)(*([1] * n_upd))
        out = C.c_void_p()
        rc = lib.gra_upload_snappy(eng.h, C.cast(comp, C.POINTER(C.c_uint8)),
                                   comp_used.value, cdescs, n_upd, ulens,
                                   counts, C.byref(out))
        assert rc == 0, ra.ffi.last_error(lib)
        rep = ra.ffi.Replay(eng, out)
    elif args.h2d:
        pin = eng.pin_alloc(used)
        C.memmove(pin, arena, used)
        rep = eng.upload(pin, used, descs, n_upd)
    else:
        rep = eng.upload(arena_p, used, descs, n_upd)

    # pre-build every window's device-cached plan (setup, untimed): a cold
    # window inside the timed region costs ~2 ms of hipMalloc/H2D
    for w in range(n_ticks_data):
        rep.prepare(w * args.tick_updates, args.tick_updates)

    def one_step(i):
        first = (i % n_ticks_data) * args.tick_updates
        if args.h2d:
            rep.tick_h2d(first, args.tick_updates)
        else:
            rep.tick(first, args.tick_updates)

    # --- warmup ---
    for i in range(args.warmup):
        one_step(i)
    rep.sync()
    eng.stats_reset()

    # --- timed region ---
    if dist:
        dist.barrier()
    t0 = time.perf_counter()
    for i in range(args.steps):
        one_step(args.warmup + i)
    rep.sync()  # hipStreamSynchronize + run-registry ingest
    t1 = time.perf_counter()
    if dist:
        dist.barrier()
    wall = t1 - t0

    # max over ranks
    if dist:
        import torch
        tw = torch.tensor([wall], dtype=torch.float64)
        dist.all_reduce(tw, op=dist.ReduceOp.MAX)
        wall = float(tw.item())

    st = eng.stats()
    upd_per_step = args.tick_updates
    total_updates = args.steps * upd_per_step * world
    value = total_updates / wall

    # roofline for the dominant kernel (partition-copy), HIP-event timed on
    # the engine stream. Algorithmic bytes per launch: raw payload read +
    # 16B-aligned payload write + 32 B of copy-task reads per record
    # (DESIGN.md states the per-unit figure).
    payload_raw = st.payload_bytes  # aligned payload written
    copy_read = st.blob_bytes  # upper-bound read = blob bytes (incl. headers)
    launches = max(st.ticks, 1)
    alg_bytes_per_launch = (payload_raw + copy_read + 32 * st.records) / launches
    copy_ms_per_launch = st.copy_ms / launches if st.copy_ms else None
    achieved = (alg_bytes_per_launch / (copy_ms_per_launch * 1e-3) / 1e9
                if copy_ms_per_launch else None)
    traffic = None
    if os.path.exists(args.traffic_file):
        try:
            tf = json.load(open(args.traffic_file))
            wk = tf.get("workload", {})
            if (wk.get("val_len") == args.val_len
                    and wk.get("nshards") == args.nshards
                    and wk.get("tick_updates") in (None, args.tick_updates)):
                traffic = tf.get("hbm_bytes_per_copy_launch")
        except Exception:
            pass

    result = {
        "metric": "replicated updates/sec (Put+Delete) at 1024 shards",
        "value": value,
        "unit": "updates/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": wall * 1e3 / args.steps,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,  # reference publishes no number
        "dtype": "u8",
        "data": "synthetic",
        "config": {
            "workload": f"{args.nshards}shards_{args.key_len}Bkey_"
                        f"{args.val_len}Bval_" +
                        {0: "uniform", 1: "zipf0.99", 2: "mixed70/20/10"}[args.kind]
                        + ("_snappy" if args.snappy else ""),
            "nshards": args.nshards,
            "key_len": args.key_len,
            "val_len": args.val_len,
            "tick_updates": args.tick_updates,
            "path": "h2d-staged" if args.h2d else "hbm-resident",
            "parallelism": f"shard-parallel x{world}",
        },
        "roofline": {
            "bound": "hbm",
            "achieved": achieved,
            "peak": HBM_PEAK_GBPS,
            "unit": "GB/s",
            "frac": (achieved / HBM_PEAK_GBPS) if achieved else None,
            "traffic": traffic,
        },
        "kernels_ms_per_tick": {
            "h2d": st.h2d_ms / launches,
            "snappy": st.snappy_ms / launches,
            "decode": st.decode_ms / launches,
            "scan": st.scan_ms / launches,
            "emit": st.emit_ms / launches,
            "copy": st.copy_ms / launches,
            "rundesc_d2h": st.runfix_ms / launches,
            "total": st.total_ms / launches,
        },
        "gpu_bytes": {"blob": st.blob_bytes, "payload": st.payload_bytes,
                      "records": st.records},
    }

    if (rank == 0 and world == 1 and not args.h2d and not args.snappy
            and not args.no_legs):
        # staging-inclusive side line on every headline run
        result["h2d_staged"] = h2d_side_leg(eng, arena, used, descs, n_upd, args)
        # host-drain side line; separate engine so the
        # drain mode never touches the headline engine's state
        result["drain_host"] = drain_side_leg(
            C.cast(arena, C.POINTER(C.c_uint8)), descs, n_upd, args)

    if rank == 0 and world == 1 and args.cpu_baseline:  # contract: N=1 only
        if args.snappy:
            result["cpu_baseline"] = cpu_baseline_leg(
                comp, comp_used.value, cdescs, n_upd, args.nshards,
                args.cpu_sample_seconds, snappy=True)
        else:
            result["cpu_baseline"] = cpu_baseline_leg(
                arena, used, descs, n_upd, args.nshards, args.cpu_sample_seconds)
        if result["cpu_baseline"]["value"]:
            result["gpu_vs_cpu"] = value / result["cpu_baseline"]["value"]

    if rank == 0:
        print(json.dumps(result))
    eng.close()
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
