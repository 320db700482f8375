#!/usr/bin/env bash
# Speculator training launch (parity: reference scripts/train_speculator.sh).
set -euo pipefail

SPEC_ARGS="${SPEC_ARGS:-
--model_variant=llama2_7b
--model_path=/fsx/base_model_ckpt
--sharding_strategy=tp
--tp_size=8
--batch_size=8
--seq_length=4096
--n_speculator_heads=3
--speculator_width=4096
--stage2_start_step=15000
--use_dummy_dataset=true
}"

export HSA_ENABLE_IPC_MODE_LEGACY=0
torchrun --nnodes=1 --nproc_per_node="${NPROC_PER_NODE:-8}" \
    --master_addr=127.0.0.1 --master_port="${MASTER_PORT:-29500}" \
    speculator/train_speculator.py ${SPEC_ARGS}
