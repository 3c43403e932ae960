#!/bin/bash
# rocprofv3 evidence for the roofline claims (run on the GPU box via
# gpurun; summaries are copied into profiles/ and committed).
#
# Pass 1: kernel trace + stats (per-kernel durations — must agree with
#         bench.py's HIP-event scan timing).
# Pass 2/3: PMC counters, SEPARATE passes (TCC has 4 slots; FETCH_SIZE
#         costs 3, WRITE_SIZE 2 — they cannot share a pass), and NEVER
#         combined with -s/-r/hip-trace (gpurun refuses; suspected node
#         crasher).
# gfx950 FETCH_SIZE gotcha (MI355X_MICROARCH.md §HBM): wide coalesced
# streaming reads are reported at 1/2 the true bytes — calibrate before
# quoting absolutes.
set -ex
ROOT="${GRAFT_REPO_ROOT:-/root/repo}"
OUT="$ROOT/gpurun_out/prof"
mkdir -p "$OUT"
cd /tmp && export TMPDIR=/tmp

BENCH_ARGS="${BENCH_ARGS:---steps 5 --warmup 2 --cpu-baseline 0}"

rocprofv3 --kernel-trace --stats --output-format csv -d "$OUT/trace" -- \
    python "$ROOT/bench.py" $BENCH_ARGS > "$OUT/bench_trace.json" 2> "$OUT/trace.log"

rocprofv3 --pmc FETCH_SIZE --output-format csv -d "$OUT/pmc_fetch" -- \
    python "$ROOT/bench.py" $BENCH_ARGS > /dev/null 2> "$OUT/pmc_fetch.log" || true

rocprofv3 --pmc WRITE_SIZE --output-format csv -d "$OUT/pmc_write" -- \
    python "$ROOT/bench.py" $BENCH_ARGS > /dev/null 2> "$OUT/pmc_write.log" || true

echo "profile artifacts in $OUT"
find "$OUT" -name "*stats*" -o -name "*.csv" | head -20
