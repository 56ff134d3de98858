#!/bin/bash
# Round-2 GPU call #5: moe-dgrad fix + bwd fast-path validation, then
# TunableOp tuning at the REAL bench GEMM shapes (16384 tokens — the
# committed CSV was tuned at 8192, so the bench's M=16384 GEMMs run on
# heuristics), then bench A/B with the new selections.
set -x
mkdir -p gpurun_out/r2

NXDT_ATTN_V3=1 timeout 600 python -m pytest tests/test_gpu_kernels.py -q \
    > gpurun_out/r2/pytest_gpu5.log 2>&1
tail -3 gpurun_out/r2/pytest_gpu5.log

timeout 300 python tools/bench_attn_kernels.py > gpurun_out/r2/attn_micro5.log 2>&1 || true
tail -6 gpurun_out/r2/attn_micro5.log

timeout 300 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/r2/bench_pre_tune.json 2>&1
tail -1 gpurun_out/r2/bench_pre_tune.json

PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
timeout 720 python tools/tune_gemms.py --tp 1 8 --tokens 16384 \
    --csv tunableop/tunableop_gfx950.csv > gpurun_out/r2/tune.log 2>&1 || true
tail -5 gpurun_out/r2/tune.log
cp tunableop/tunableop_gfx950.csv gpurun_out/r2/tunableop_gfx950.csv 2>/dev/null || true

timeout 300 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/r2/bench_post_tune.json 2>&1
tail -1 gpurun_out/r2/bench_post_tune.json
