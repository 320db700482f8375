"""Training configuration for the MI355X-native pretraining engine.

Mirrors the capability surface of the reference's flat config dataclass
(reference: fms_fsdp/config/training.py:5-74) while adding MI355X-specific
runtime knobs (all-gather bucket sizing for the 7-link xGMI clique,
reshard_after_forward control exploiting 288 GB HBM3E).
"""

from dataclasses import dataclass
from typing import Optional, Union


@dataclass
class train_config:
    # model
    model_variant: str = "7b"
    ckpt_load_path: str = "/fsx/output/ckpt"
    ckpt_save_path: str = "/fsx/output/ckpt"

    # dataset and dataloader
    use_dummy_dataset: bool = False
    data_path: str = "/fsx/data"
    file_type: str = "arrow"
    col_name: str = "tokens"
    tokenizer_path: str = "/fsx/tokenizer"
    datasets: str = "lang=en/dataset=commoncrawl,lang=en/dataset=webhose,lang=en/dataset=github_clean,lang=de/dataset=wikipedia,lang=es/dataset=wikipedia,lang=fr/dataset=wikipedia,lang=ja/dataset=wikipedia,lang=pt/dataset=wikipedia,lang=en/dataset=wikimedia,lang=en/dataset=uspto,lang=en/dataset=pubmedcentral,lang=en/dataset=arxiv,lang=en/dataset=stackexchange"
    weights: str = "7725,500,550,28,17,22,25,8,100,500,175,250,100"
    seq_length: int = 4096
    vocab_size: int = 32000
    bos_token: Optional[int] = None
    eos_token: int = 0
    bol_token: Optional[int] = None
    eol_token: Optional[int] = None
    strip_tokens: str = ""
    logical_shards: int = 1024
    num_workers: int = 1

    # fsdp policies
    sharding_strategy: str = "hsdp"           # fsdp | hsdp | ddp
    fsdp_activation_checkpointing: bool = False
    selective_checkpointing: Union[float, str] = 1  # fraction of blocks to checkpoint
    mixed_precision: bool = True
    low_cpu_fsdp: bool = False

    # training spec
    batch_size: int = 2
    num_steps: int = 1000000
    training_stage: str = ""
    learning_rate: float = 3e-4
    grad_clip_thresh: float = 1.0
    seed: int = 2023

    # continued training spec
    resuming_dataset: bool = False

    # profiling
    use_profiler: bool = False
    profiler_rank0_only: bool = True

    # logging
    report_interval: int = 100
    # write checkpoint files from a background thread (D2H snapshot is
    # still synchronous; metadata.pth written last = validity marker)
    async_checkpoint: bool = False
    checkpoint_interval: int = 10000
    tracker: Optional[str] = None             # None | "wandb" | "aim"
    tracker_dir: str = "/fsx/aim_logs/llama"
    tracker_project_name: str = "llama"
    tracker_run_id: Optional[str] = None

    # compile — the MI355X engine replaces torch.compile with hand-fused
    # HIP kernels; the flag is accepted for config compatibility and ignored.
    use_torch_compile: bool = False

    # speculator training
    tp_size: int = 8
    model_arch: str = "embedllama"
    model_path: str = "/path/to/model/"
    n_speculator_heads: int = 3
    speculator_width: int = 4096
    speculator_tie_weights: bool = True
    speculator_scale_input: bool = True
    stage2_start_step: int = 15000
    stage2_prompt_length: int = 64
    stage2_batch_size: int = 96
    stage2_seq_length: int = 256

    # ---- MI355X-native runtime knobs (no reference equivalent) ----
    # Keep a unit's gathered bf16 params alive between forward and backward
    # (288 GB HBM3E makes this the right default for <=13B models); set True
    # to re-gather in backward like classic FULL_SHARD (needed for 70B).
    reshard_after_forward: Union[bool, str] = "auto"
    # How many units ahead to prefetch all-gathers on the comm stream.
    prefetch_lookahead: int = 1
    # low-precision reduce-scatter of grads (bf16) with fp32 master accum;
    # "policy" defers to the mp_policy triple below
    reduce_dtype: str = "policy"
    # Mixed-precision policy (reference mixed_precision.py:5-27 triples):
    #   auto         -> bf16 when mixed_precision and the device supports
    #                   it, fp16 otherwise; fp32 when not mixed_precision
    #   bf16         -> bfSixteen        (param bf16, reduce bf16)
    #   bf16_working -> bfSixteen_working (param fp32, reduce bf16)
    #   fp16         -> fpSixteen        (param fp16, reduce fp16,
    #                   dynamic loss scaling enabled automatically)
    #   fp32         -> fp32_policy
    mp_policy: str = "auto"
