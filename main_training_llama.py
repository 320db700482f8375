"""Llama pretraining entry point (parity: reference main_training_llama.py).

Launch: torchrun --nproc-per-node N main_training_llama.py --key=value ...
(one process per GPU over RCCL/xGMI). Without torchrun runs single-process.
"""

import math
import os
import sys

import torch
import torch.distributed as dist
from fms_fsdp_amd.config import train_config, update_config, get_model_config
from fms_fsdp_amd.models import Llama, LlamaBlock
from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
from fms_fsdp_amd.parallel.policies import (apply_selective_ac,
                                            get_mixed_precision_dtypes,
                                            get_sharding_strategy,
                                            resolve_reshard_after_forward)
from fms_fsdp_amd.data import get_data_loader, get_dummy_loader
from fms_fsdp_amd.utils.checkpointing import Checkpointer
from fms_fsdp_amd.utils.train import (LambdaLR, get_profiler, get_tracker,
                                      setup, setup_environ_flags, train)


def parse_cli(argv):
    """--key=value / --key value pairs -> kwargs (replaces fire.Fire)."""
    kwargs = {}
    i = 0
    while i < len(argv):
        a = argv[i]
        if not a.startswith("--"):
            raise ValueError(f"unexpected arg {a}")
        a = a[2:]
        if "=" in a:
            k, v = a.split("=", 1)
        else:
            k = a
            i += 1
            v = argv[i] if i < len(argv) else "true"
        kwargs[k.replace("-", "_")] = _coerce(v)
        i += 1
    return kwargs


def _coerce(v):
    if isinstance(v, bool):
        return v
    if v.lower() in ("true", "false"):
        return v.lower() == "true"
    for t in (int, float):
        try:
            return t(v)
        except ValueError:
            pass
    return v


def main(**kwargs):
    cfg = train_config()
    update_config(cfg, **kwargs)

    torch.manual_seed(cfg.seed)
    if "RANK" in os.environ and not dist.is_initialized():
        setup()
    rank = dist.get_rank() if dist.is_initialized() else 0
    world_size = dist.get_world_size() if dist.is_initialized() else 1
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    setup_environ_flags()

    if rank == 0:
        print(f"--> running with these configs {cfg}")

    # model
    model_config = get_model_config(cfg.model_variant)
    model_config.src_vocab_size = max(model_config.src_vocab_size, cfg.vocab_size)
    if cfg.low_cpu_fsdp:
        # model stays on meta; ShardedModel streams materialization one
        # unit at a time (no rank ever holds the full unsharded model —
        # reference: param_init.py:9-18 + main_training_llama.py:59-65)
        with torch.device("meta"):
            model = Llama(model_config)
        torch.manual_seed(cfg.seed)  # identical init on every rank pre-broadcast
    else:
        model = Llama(model_config)
        model.reset_parameters()

    if rank == 0:
        print(f"--> model has {model.param_count() / 1e6:.2f}M params")

    # data
    if cfg.use_dummy_dataset:
        cfg.vocab_size = model_config.src_vocab_size
        train_loader = get_dummy_loader(cfg, rank, world_size)
    else:
        train_loader = get_data_loader(cfg, rank, world_size)

    # sharded runtime
    param_dtype, reduce_dtype = get_mixed_precision_dtypes(cfg)
    model = ShardedModel(
        model, LlamaBlock,
        sharding_strategy=get_sharding_strategy(cfg),
        param_dtype=param_dtype, reduce_dtype=reduce_dtype,
        reshard_after_forward=resolve_reshard_after_forward(
            cfg, sum(p.numel() for p in model.parameters())),
        prefetch_lookahead=cfg.prefetch_lookahead,
    )

    # selective AC (reference: main_training_llama.py:99-102)
    if cfg.fsdp_activation_checkpointing:
        if rank == 0:
            print(f"--> applying selective AC p={cfg.selective_checkpointing}")
        apply_selective_ac(model, LlamaBlock, cfg.selective_checkpointing)

    optimizer = ShardedAdamW(model, lr=cfg.learning_rate, betas=(0.9, 0.95),
                             weight_decay=0.1)

    checkpointer = Checkpointer(cfg.ckpt_save_path, 1000, cfg.sharding_strategy,
                                rank, local_rank,
                                async_save=cfg.async_checkpoint)
    model, optimizer, _, start_step, tokens_seen, is_resuming = checkpointer.load(
        model, optimizer,
        None if cfg.use_dummy_dataset else train_loader,
        path=os.path.join(cfg.ckpt_load_path, "checkpoints/")
        if not os.path.isfile(cfg.ckpt_load_path) else cfg.ckpt_load_path,
        strict=False,
        is_compiled=False,
    )

    # LR schedule: warmup + cosine, or linear anneal for the annealing stage
    # (reference: main_training_llama.py:137-148)
    if cfg.training_stage == "annealing":
        schedule = lambda x: 1 - x / cfg.num_steps
    else:
        warmup_interval = min(2000, cfg.num_steps // 20) or 1
        schedule = lambda x: min(
            1 - (1 - min(x, warmup_interval) / warmup_interval) ** 2,
            0.1 + 0.5 * (1 - 0.1) * (1 + math.cos(min(x, cfg.num_steps) / cfg.num_steps * math.pi)),
        )
    scheduler = LambdaLR(optimizer, lambda x: schedule(x + start_step))

    profiler = get_profiler(cfg, rank)
    tracker = get_tracker(cfg, rank)

    if rank == 0:
        print(f"--> starting training for {cfg.num_steps} steps")
    train(cfg, model, local_rank, rank, train_loader, optimizer, scheduler,
          profiler, checkpointer, start_step, tokens_seen, tracker)

    if dist.is_initialized():
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main(**parse_cli(sys.argv[1:]))
