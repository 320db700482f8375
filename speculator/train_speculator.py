"""Speculator training entry point (parity target: reference
speculator/train_speculator.py).

Trains an MLPSpeculator against a frozen base Llama:
- base model sharding: fsdp | hsdp (via the ShardedModel runtime) or tp
  (column/row-parallel with RCCL all-reduce, parallel/tp.py) — reference
  builds a 2-D dp x tp mesh (:128-142 there);
- the speculator itself is data-parallel NO_SHARD (reference :197-212);
- two-stage loss/LR schedule (train_speculator_utils).
"""

import math
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from fms_fsdp_amd.config import train_config, update_config, get_model_config
from fms_fsdp_amd.models import Llama, LlamaBlock
from fms_fsdp_amd.models.speculator import MLPSpeculator
from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
from fms_fsdp_amd.parallel.tp import tp_shard_llama
from fms_fsdp_amd.data import get_data_loader, get_dummy_loader
from fms_fsdp_amd.utils.checkpointing import Checkpointer, consolidate_checkpoint
from fms_fsdp_amd.utils.train import (LambdaLR, get_profiler, setup,
                                      setup_environ_flags)
from main_training_llama import parse_cli
from speculator.train_speculator_utils import train_speculator


def main(**kwargs):
    cfg = train_config()
    cfg.seq_length = 4096
    cfg.sharding_strategy = "tp"
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
    device = torch.device("cuda", local_rank) if torch.cuda.is_available() \
        else torch.device("cpu")

    # base model (reference supports llama / gpt_bigcode / mixtral bases,
    # train_speculator_utils.py:430-523 there)
    mcfg = get_model_config(cfg.model_variant)
    arch = cfg.model_arch.lower()
    if arch in ("embedgptbigcode", "gpt_bigcode"):
        from fms_fsdp_amd.models.gpt_bigcode import GPTBigCode
        model = GPTBigCode(mcfg)
    elif arch in ("embedmixtral", "mixtral"):
        from fms_fsdp_amd.models.mixtral import Mixtral
        model = Mixtral(mcfg)
    else:
        model = Llama(mcfg)
    model.reset_parameters()
    if cfg.model_path and os.path.exists(cfg.model_path):
        if os.path.isfile(cfg.model_path):
            sd = torch.load(cfg.model_path, map_location="cpu",
                            weights_only=False)
        else:
            sd = consolidate_checkpoint(cfg.model_path)
        model.load_state_dict({k.replace("_orig_mod.", ""): v
                               for k, v in sd.items()}, strict=False)
    model = model.to(device)
    if torch.cuda.is_available():
        model = model.bfloat16()
    model.eval()
    for p in model.parameters():
        p.requires_grad_(False)

    # parallel layout for the frozen base model
    tp_group = None
    if cfg.sharding_strategy == "tp" and world_size > 1 and arch not in (
            "embedllama", "llama"):
        if rank == 0:
            print("WARNING: TP base-model sharding implemented for llama; "
                  f"running {arch} replicated")
        cfg.sharding_strategy = "fsdp"
    if cfg.sharding_strategy == "tp" and world_size > 1:
        # 2-D dp x tp device mesh for the frozen base model (reference:
        # train_speculator.py:128-142 builds base_model_mesh ("dp","tp")
        # plus a flat 1-D speculator mesh; our speculator uses the world
        # group via ShardedModel "ddp", which IS that flat mesh)
        tp_size = min(cfg.tp_size, world_size)
        from torch.distributed.device_mesh import init_device_mesh
        base_model_mesh = init_device_mesh(
            "cuda" if torch.cuda.is_available() else "cpu",
            (world_size // tp_size, tp_size), mesh_dim_names=("dp", "tp"))
        tp_group = base_model_mesh["tp"].get_group()
        model = tp_shard_llama(model, tp_group)
    elif cfg.sharding_strategy in ("fsdp", "hsdp") and world_size > 1:
        # frozen replicated weights would also work; keep them whole per
        # rank (288 GB HBM) — sharding is for the trainable speculator.
        pass

    # speculator (data-parallel, NO_SHARD like the reference)
    speculator = MLPSpeculator(
        mcfg.emb_dim, cfg.speculator_width, mcfg.src_vocab_size,
        cfg.n_speculator_heads, tie_weights=cfg.speculator_tie_weights,
        scale_input=cfg.speculator_scale_input)
    speculator.reset_parameters()
    speculator = speculator.to(device)

    sharded_spec = ShardedModel(
        speculator, MLPSpeculator, sharding_strategy="ddp",
        param_dtype=torch.bfloat16 if torch.cuda.is_available()
        else torch.float32, device=device)
    if rank == 0:
        print(f"--> speculator has {speculator.param_count() / 1e6:.1f}M params")

    if cfg.use_dummy_dataset:
        cfg.vocab_size = mcfg.src_vocab_size
        train_loader = get_dummy_loader(cfg, rank, world_size)
    else:
        train_loader = get_data_loader(cfg, rank, world_size)

    optimizer = ShardedAdamW(sharded_spec, lr=cfg.learning_rate,
                             betas=(0.9, 0.95), weight_decay=0.1)
    checkpointer = Checkpointer(cfg.ckpt_save_path, 1000, "ddp", rank,
                                local_rank)
    _, _, _, start_step, tokens_seen, _ = checkpointer.load(
        sharded_spec, optimizer, None, path=cfg.ckpt_load_path)

    # two-stage LR schedule (reference train_speculator.py:261-300)
    warmup = 150
    s2_start = cfg.stage2_start_step

    def schedule(x):
        if x <= s2_start:
            return min(x / warmup,
                       0.5 + 0.5 * math.cos(min(x, s2_start) / s2_start * math.pi))
        x2 = x - s2_start
        return min(x2 / warmup,
                   0.1 + 0.5 * (1 - 0.1) *
                   (1 + math.cos(min(x2, cfg.num_steps - s2_start)
                                 / (cfg.num_steps - s2_start) * math.pi)))

    scheduler = LambdaLR(optimizer, lambda x: schedule(x + start_step))
    profiler = get_profiler(cfg, rank)

    # expose clip through the sharded wrapper
    sharded_spec.n_predict = speculator.n_predict

    train_speculator(cfg, model, _wrap_spec(sharded_spec, speculator),
                     local_rank, rank, world_size, train_loader, optimizer,
                     scheduler, checkpointer, start_step, tokens_seen,
                     profiler, tp_group)

    if dist.is_initialized():
        dist.barrier()
        dist.destroy_process_group()


def _wrap_spec(sharded_spec, speculator):
    """The train loop calls speculator(state, inds) and clip_grad_norm_;
    route both through the ShardedModel wrapper."""
    sharded_spec.n_predict = speculator.n_predict
    return sharded_spec


if __name__ == "__main__":
    main(**parse_cli(sys.argv[1:]))
