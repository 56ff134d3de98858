#!/bin/bash
# Round-2 GPU call #2: re-validate after the SQ/SKV kernel extension,
# A/B v2-vs-v3 bench, capture rocprof stats. All rocprof raw output goes
# to /tmp; only the small stats CSVs are copied back (the 64 MiB
# gpurun_out merge limit ate call #1's results).
set -x
mkdir -p gpurun_out/r2

NXDT_ATTN_V3=1 timeout 900 python -m pytest tests -m gpu -q \
    > gpurun_out/r2/pytest_gpu.log 2>&1
tail -3 gpurun_out/r2/pytest_gpu.log

timeout 420 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/r2/bench_base.json 2>&1
NXDT_ATTN_V3=1 timeout 420 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/r2/bench_v3.json 2>&1

cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
timeout 420 rocprofv3 --kernel-trace --stats -d /tmp/prof_v2 -- \
    python bench.py --steps 3 --warmup 2 > gpurun_out/r2/prof_v2.log 2>&1 || true
NXDT_ATTN_V3=1 timeout 420 rocprofv3 --kernel-trace --stats -d /tmp/prof_v3 -- \
    python bench.py --steps 3 --warmup 2 > gpurun_out/r2/prof_v3.log 2>&1 || true
# stats CSVs only (raw traces are 10s of MiB)
find /tmp/prof_v2 /tmp/prof_v3 -name '*stats*.csv' \
    -exec cp --parents {} gpurun_out/r2/ \; 2>/dev/null || true
du -sh gpurun_out/ || true
tail -1 gpurun_out/r2/bench_base.json gpurun_out/r2/bench_v3.json 2>/dev/null
