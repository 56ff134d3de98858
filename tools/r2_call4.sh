#!/bin/bash
# Round-2 GPU call #4: grouped-GEMM MoE validation + the profile/A-B runs
# that call #3 lost to a stale .so (rope signature change).
set -x
mkdir -p gpurun_out/r2

timeout 900 python -m pytest tests/test_gpu_kernels.py -q \
    > gpurun_out/r2/pytest_gpu4.log 2>&1
tail -3 gpurun_out/r2/pytest_gpu4.log

timeout 420 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/r2/bench_base4.json 2>&1
NXDT_ATTN_V3=fwd timeout 420 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/r2/bench_v3fwd.json 2>&1

cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof_v2 -- \
    python bench.py --steps 3 --warmup 2 > gpurun_out/r2/prof_v2.log 2>&1 || true
for f in $(find /tmp/prof_v2 -name '*kernel_stats*.csv'); do
  cp "$f" gpurun_out/r2/kernel_stats_v2.csv
done
tail -1 gpurun_out/r2/bench_base4.json gpurun_out/r2/bench_v3fwd.json 2>/dev/null
