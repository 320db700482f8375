#!/usr/bin/env bash
# Single/multi-node launch for Llama pretraining on MI355X nodes
# (parity target: reference scripts/train.sh, re-targeted from EFA/NCCL
# to RCCL over xGMI intra-node).
set -euo pipefail

MODEL_ARGS="${MODEL_ARGS:-
--model_variant=llama2_7b
--sharding_strategy=hsdp
--batch_size=2
--seq_length=4096
--use_dummy_dataset=true
--report_interval=50
}"

NNODES=${SLURM_NNODES:-1}
NODE_RANK=${SLURM_NODEID:-0}
NPROC=${NPROC_PER_NODE:-8}
MASTER_ADDR=${MASTER_ADDR:-127.0.0.1}
MASTER_PORT=${MASTER_PORT:-29500}

# dmabuf IPC is the only supported mode on this driver stack
export HSA_ENABLE_IPC_MODE_LEGACY=0
export TORCH_NCCL_ASYNC_ERROR_HANDLING=1

torchrun \
    --nnodes="$NNODES" \
    --node_rank="$NODE_RANK" \
    --nproc_per_node="$NPROC" \
    --master_addr="$MASTER_ADDR" \
    --master_port="$MASTER_PORT" \
    main_training_llama.py ${MODEL_ARGS}
