"""fms_fsdp_amd — MI355X-native LLM pretraining framework.

A from-scratch re-design of the capabilities of foundation-model-stack/
fms-fsdp for AMD Instinct MI355X (gfx950): PyTorch-ROCm orchestration,
hand-written CDNA4 HIP kernels for every hot op, our own FSDP/HSDP/DDP
sharded-training runtime over RCCL/xGMI, a checkpointable & rescalable
streaming dataloader, distributed checkpointing, and HF export.
"""

import torch as _torch  # noqa: F401  (loads libtorch/hip shared objects that _C links against)

__version__ = "0.1.0"
