#!/bin/bash
# rocprofv3 evidence for the apply-path kernels (run on the GPU box).
# Pass 1: kernel trace + stats (per-kernel durations).
# Pass 2/3: HBM traffic counters — FETCH_SIZE and WRITE_SIZE cost 3/2 TCC
# slots so they need separate --pmc passes (MI355X_MICROARCH.md §rocprofv3);
# never combined with trace domains (gpurun refuses that combination).
set -e
cd /tmp && export TMPDIR=/tmp
REPO=${GRAFT_REPO_ROOT:-/root/repo}
OUT=$REPO/gpurun_out/prof
mkdir -p "$OUT"
BENCH="python $REPO/bench.py --steps 24 --warmup 6 --max-ticks-resident 4 --no-cpu-baseline --no-legs"
SNAPPY_BENCH="python $REPO/bench.py --snappy --kind 2 --steps 4 --warmup 2 --max-ticks-resident 2 --no-cpu-baseline"

rocprofv3 --output-format csv --kernel-trace --stats -d "$OUT/stats" -o run -- $BENCH > "$OUT/bench_stats.json" 2> "$OUT/stats.log"
rocprofv3 --output-format csv --pmc FETCH_SIZE -d "$OUT/fetch" -o run -- $BENCH > /dev/null 2> "$OUT/fetch.log"
rocprofv3 --output-format csv --pmc WRITE_SIZE -d "$OUT/write" -o run -- $BENCH > /dev/null 2> "$OUT/write.log"

rocprofv3 --output-format csv --kernel-trace --stats -d "$OUT/snappy" -o run -- $SNAPPY_BENCH > /dev/null 2> "$OUT/snappy.log"
python3 "$REPO/scripts/summarize_prof.py" "$OUT"
