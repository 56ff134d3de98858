#!/bin/bash
# Round-2 GPU call #3: kernel-level diagnosis.
#  - rocprof kernel stats (CSV this time) for the v2 bench: where did the
#    ZeRO-1 rework's expected win go, what are the top kernels now
#  - A/B v3-forward-only (v3 fwd was predicted faster; full v3 was slower
#    end-to-end, suspect the backward)
#  - microbench fwd/bwd attention kernels in isolation
set -x
mkdir -p gpurun_out/r2

cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof_v2 -- \
    python bench.py --steps 3 --warmup 2 > gpurun_out/r2/prof_v2.log 2>&1 || true
find /tmp/prof_v2 -name '*.csv' | head -5
for f in $(find /tmp/prof_v2 -name '*kernel_stats*.csv'); do
  cp "$f" gpurun_out/r2/kernel_stats_v2.csv
done

NXDT_ATTN_V3=fwd timeout 420 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/r2/bench_v3fwd.json 2>&1

timeout 600 python tools/bench_attn_kernels.py > gpurun_out/r2/attn_micro.log 2>&1 || true

tail -1 gpurun_out/r2/bench_v3fwd.json
tail -30 gpurun_out/r2/attn_micro.log
