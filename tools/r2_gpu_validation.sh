#!/bin/bash
# Round-2 GPU validation battery — run via:
#   /usr/local/graft/bin/gpurun --timeout 1800 -- 'bash tools/r2_gpu_validation.sh'
# Each phase writes into gpurun_out/ so results merge back even if a later
# phase dies. Phases are ordered cheapest-first; comment out what you
# don't need.
set -x
mkdir -p gpurun_out/r2

# 0a) distributed-backward parity on the device (gloo spawn; both ranks
# share the GPU — catches device-specific grad bugs the CPU run can't)
timeout 600 python tools/check_grad_parity.py --layout tp2_sp \
    > gpurun_out/r2/grad_parity.log 2>&1 || true
timeout 600 python tools/check_grad_parity.py --layout lora_tp2_sp \
    >> gpurun_out/r2/grad_parity.log 2>&1 || true

# 0) baseline sanity: full GPU suite + current bench number
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/r2/pytest_gpu.log 2>&1
timeout 600 python bench.py --steps 6 --warmup 2 > gpurun_out/r2/bench_base.json 2>&1

# 1) v3 (T12 32x32) attention kernels: numerics first, then A/B
NXDT_ATTN_V3=1 timeout 600 python -m pytest tests/test_gpu_kernels.py -q -m gpu \
    > gpurun_out/r2/pytest_v3.log 2>&1
NXDT_ATTN_V3=1 timeout 600 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/r2/bench_v3.json 2>&1

# 2) SP comm/GEMM overlap (needs >1 GPU; skipped on 1-GPU boxes)
NGPU=$(rocm-smi --showid 2>/dev/null | grep -c "^GPU" || echo 1)
if [ "$NGPU" -ge 2 ]; then
  for C in 0 2 4; do
    NXDT_SP_OVERLAP=$C timeout 600 python -m torch.distributed.run --nnodes=1 \
      --nproc-per-node 2 --master-addr 127.0.0.1 bench.py --gpus 2 --steps 6 \
      --warmup 2 > gpurun_out/r2/bench_sp_overlap_$C.json 2>&1
  done
fi

# 3) TunableOp: extend coverage to the TP=8-shard GEMM shapes
timeout 900 python tools/tune_gemms.py --tp 8 \
    --csv gpurun_out/r2/tunableop_tp8.csv > gpurun_out/r2/tune.log 2>&1 || true

# 4) kernel profile for the record
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/r2/prof -- \
    python bench.py --steps 3 --warmup 2 > gpurun_out/r2/prof.log 2>&1 || true

tail -5 gpurun_out/r2/*.log gpurun_out/r2/*.json 2>/dev/null
