#!/bin/bash
# Cluster environment resolution (reference examples/train_setup.sh):
# SLURM / single-node detection, rendezvous, RCCL-over-xGMI env.

if [ -n "${SLURM_JOB_ID:-}" ]; then
    NNODES=${SLURM_NNODES:-1}
    NODE_RANK=${SLURM_NODEID:-0}
    MASTER_ADDR=$(scontrol show hostnames "$SLURM_JOB_NODELIST" | head -n1)
    # restart-count log dirs (reference train_setup.sh:27-29)
    export RESTART_COUNT=${SLURM_RESTART_COUNT:-0}
else
    NNODES=${NNODES:-1}
    NODE_RANK=${NODE_RANK:-0}
    MASTER_ADDR=${MASTER_ADDR:-127.0.0.1}
fi
MASTER_PORT=${MASTER_PORT:-41000}

DISTRIBUTED_ARGS="--nnodes $NNODES --node-rank $NODE_RANK \
--master-addr $MASTER_ADDR --master-port $MASTER_PORT"

# MI355X / RCCL environment (replaces the reference's EFA/NEURON_* block)
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}
export TORCH_BLAS_PREFER_HIPBLASLT=1
export NCCL_MIN_NCHANNELS=${NCCL_MIN_NCHANNELS:-32}
# inter-node (when RDMA NICs present): uncomment / adjust
# export NCCL_IB_HCA=...
export GPU_MAX_HW_QUEUES=${GPU_MAX_HW_QUEUES:-4}
