#!/bin/bash
# Round-2 final validation: full GPU suite (v3 battery enabled), smoke,
# final bench, kernel-stats profile for the record.
set -x
mkdir -p gpurun_out/final

NXDT_ATTN_V3=1 timeout 900 python -m pytest tests -m gpu -q \
    > gpurun_out/final/pytest_gpu.log 2>&1
tail -2 gpurun_out/final/pytest_gpu.log

timeout 300 python __graft_entry__.py smoke > gpurun_out/final/smoke.log 2>&1
tail -2 gpurun_out/final/smoke.log

timeout 300 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/final/bench.json 2>&1
tail -1 gpurun_out/final/bench.json

cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/pf -- \
    python bench.py --steps 3 --warmup 2 > gpurun_out/final/prof.log 2>&1 || true
for f in $(find /tmp/pf -name '*kernel_stats*.csv'); do
  cp "$f" gpurun_out/final/kernel_stats_final.csv
done
du -sh gpurun_out/
