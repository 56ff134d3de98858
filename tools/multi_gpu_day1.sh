#!/bin/bash
# First session on a REAL multi-GPU MI355X node — the measurement plan
# ROADMAP.md defers to. Everything here is correctness-tested on
# CPU/gloo (world 2-4) but has never executed over RCCL/xGMI: run
# cheapest-first, stop at the first failure, keep the logs.
#
#   bash tools/multi_gpu_day1.sh 2>&1 | tee multi_gpu_day1.log
set -x
NGPU=$(rocm-smi --showid 2>/dev/null | grep -c "GPU" || echo 8)
export HSA_ENABLE_IPC_MODE_LEGACY=0

run() { # world, extra bench args...
  local W=$1; shift
  python -m torch.distributed.run --nnodes=1 --nproc-per-node "$W" \
    --master-addr 127.0.0.1 --master-port 29400 bench.py --gpus "$W" "$@"
}

# 1) 2-GPU: TP2+SP correctness signal (loss finite, no hang) + first
#    multi-GPU number
timeout 600 run 2 --steps 4 --warmup 2

# 2) distributed GPU parity: TP2 grad parity ON DEVICE (catches RCCL
#    view-tensor / async-collective bugs the gloo runs cannot)
timeout 600 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29401 tools/check_grad_parity.py \
  --layout tp2_sp || true

# 3) ring attention on 2 real GPUs (zigzag CP2) — the one major path
#    with zero hardware evidence; compare against the single-GPU kernel
timeout 600 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29402 -m pytest \
  tests/test_context_parallel.py -q -k ring || true

# 4) the headline: TP=8 (then 4, 2 for the scaling curve)
for W in 8 4 2; do
  [ "$NGPU" -ge "$W" ] && timeout 900 run "$W" --steps 6 --warmup 2
done

# 5) SP overlap A/B at TP=8 (flip the default only on a win)
for C in 0 2 4 8; do
  NXDT_SP_OVERLAP=$C timeout 600 run 8 --steps 4 --warmup 2
done

# 6) RCCL tuning sweep for the xGMI clique
for ALGO in "" Ring Tree; do
  NCCL_ALGO=$ALGO timeout 600 run 8 --steps 4 --warmup 2
done
NCCL_MIN_NCHANNELS=32 timeout 600 run 8 --steps 4 --warmup 2

# 7) rocprof of the TP=8 steady state (collectives + kernels)
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/p8 -- \
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
  --master-addr 127.0.0.1 --master-port 29403 bench.py --gpus 8 --steps 3 --warmup 2
