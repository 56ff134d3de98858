#!/bin/bash
# Round-2 GPU call #1 (re-run after container loss): validate the ZeRO-1
# grad-hook rework, validate + A/B the v3 attention kernels, record a
# profile. Budget-focused subset of r2_gpu_validation.sh.
set -x
mkdir -p gpurun_out/r2

# 0) full GPU suite including the gated v3 tests
NXDT_ATTN_V3=1 timeout 900 python -m pytest tests -m gpu -q \
    > gpurun_out/r2/pytest_gpu.log 2>&1

# 1) bench base (v2 attention, new ZeRO-1) — round-1 was 12,269 tok/s
timeout 420 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/r2/bench_base.json 2>&1

# 2) bench with v3 attention
NXDT_ATTN_V3=1 timeout 420 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/r2/bench_v3.json 2>&1

# 3) kernel profile of the better path (assume v3 pending check; profile both briefly)
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/r2/prof_v2 -- \
    python bench.py --steps 3 --warmup 2 > gpurun_out/r2/prof_v2.log 2>&1 || true
NXDT_ATTN_V3=1 timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/r2/prof_v3 -- \
    python bench.py --steps 3 --warmup 2 > gpurun_out/r2/prof_v3.log 2>&1 || true

tail -3 gpurun_out/r2/bench_*.json
tail -15 gpurun_out/r2/pytest_gpu.log
