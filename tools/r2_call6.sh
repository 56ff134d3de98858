#!/bin/bash
# Round-2 GPU call #6: validate the fused delta kernel, then PMC counters
# on the attention kernels (VALU/MFMA mix, waits, LDS conflicts) to rank
# the next fwd/bwd optimizations.
set -x
mkdir -p gpurun_out/r2

timeout 420 python -m pytest tests/test_gpu_kernels.py -q -k "flash or moe" \
    > gpurun_out/r2/pytest_gpu6.log 2>&1
tail -3 gpurun_out/r2/pytest_gpu6.log

timeout 300 python tools/bench_attn_kernels.py --iters 10 > gpurun_out/r2/attn_micro6.log 2>&1 || true
grep -E "v2|v3|drift" gpurun_out/r2/attn_micro6.log

cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
timeout 420 rocprofv3 \
  --pmc SQ_WAVE_CYCLES,SQ_WAIT_ANY,SQ_WAIT_INST_ANY,SQ_INSTS_VALU,SQ_INSTS_MFMA,SQ_LDS_BANK_CONFLICT,SQ_LDS_IDX_ACTIVE,SQ_VALU_MFMA_BUSY_CYCLES \
  --output-format csv -d /tmp/pmc -- \
  python tools/bench_attn_kernels.py --iters 3 > gpurun_out/r2/pmc.log 2>&1 || true
find /tmp/pmc -name '*.csv' | head
for f in $(find /tmp/pmc -name '*counter_collection*.csv'); do
  cp "$f" gpurun_out/r2/pmc_attn.csv
done
ls -la gpurun_out/r2/pmc_attn.csv 2>/dev/null || find /tmp/pmc -name '*.csv' -exec cp {} gpurun_out/r2/ \;
du -sh gpurun_out/
