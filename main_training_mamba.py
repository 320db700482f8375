"""Mamba2 (hybrid) pretraining entry point (parity: reference
main_training_mamba.py). Same skeleton as main_training_llama with the
MambaBlock wrapping unit."""

import math
import os
import sys

import torch
import torch.distributed as dist

from fms_fsdp_amd.config import train_config, update_config, get_model_config
from fms_fsdp_amd.models.mamba import MambaBlock, MambaConfig, MambaLMHeadModel
from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
from fms_fsdp_amd.parallel.policies import (apply_selective_ac,
                                            get_mixed_precision_dtypes,
                                            get_sharding_strategy,
                                            resolve_reshard_after_forward)
from fms_fsdp_amd.data import get_data_loader, get_dummy_loader
from fms_fsdp_amd.utils.checkpointing import Checkpointer
from fms_fsdp_amd.utils.train import (LambdaLR, get_profiler, get_tracker,
                                      setup, setup_environ_flags, train)
from main_training_llama import parse_cli


def main(**kwargs):
    cfg = train_config()
    cfg.model_variant = "mamba_9.8b"
    cfg.learning_rate = 3e-4
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

    mcfg = MambaConfig.from_dict(get_model_config(cfg.model_variant))
    model = MambaLMHeadModel(mcfg)
    model.reset_parameters()
    if rank == 0:
        print(f"--> mamba model has {model.param_count() / 1e6:.2f}M params")

    if cfg.use_dummy_dataset:
        cfg.vocab_size = mcfg.vocab_size
        train_loader = get_dummy_loader(cfg, rank, world_size)
    else:
        train_loader = get_data_loader(cfg, rank, world_size)

    param_dtype, reduce_dtype = get_mixed_precision_dtypes(cfg)
    model = ShardedModel(
        model, MambaBlock,
        sharding_strategy=get_sharding_strategy(cfg),
        param_dtype=param_dtype, reduce_dtype=reduce_dtype,
        reshard_after_forward=resolve_reshard_after_forward(
            cfg, sum(p.numel() for p in model.parameters())),
        prefetch_lookahead=cfg.prefetch_lookahead,
    )
    if cfg.fsdp_activation_checkpointing:
        apply_selective_ac(model, MambaBlock, cfg.selective_checkpointing)

    optimizer = ShardedAdamW(model, lr=cfg.learning_rate, betas=(0.9, 0.95),
                             weight_decay=0.1)
    checkpointer = Checkpointer(cfg.ckpt_save_path, 1000, cfg.sharding_strategy,
                                rank, local_rank,
                                async_save=cfg.async_checkpoint)
    model, optimizer, _, start_step, tokens_seen, _ = checkpointer.load(
        model, optimizer,
        None if cfg.use_dummy_dataset else train_loader,
        path=os.path.join(cfg.ckpt_load_path, "checkpoints/")
        if not os.path.isfile(cfg.ckpt_load_path) else cfg.ckpt_load_path)

    warmup_interval = min(2000, cfg.num_steps // 20) or 1
    schedule = lambda x: min(
        1 - (1 - min(x, warmup_interval) / warmup_interval) ** 2,
        0.1 + 0.5 * (1 - 0.1) * (1 + math.cos(min(x, cfg.num_steps) / cfg.num_steps * math.pi)))
    scheduler = LambdaLR(optimizer, lambda x: schedule(x + start_step))

    profiler = get_profiler(cfg, rank)
    tracker = get_tracker(cfg, rank)
    train(cfg, model, local_rank, rank, train_loader, optimizer, scheduler,
          profiler, checkpointer, start_step, tokens_seen, tracker)

    if dist.is_initialized():
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main(**parse_cli(sys.argv[1:]))
