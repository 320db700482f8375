"""Train loop, process-group setup, profiler factory, trackers.

Parity target: reference fms_fsdp/utils/train_utils.py:21-271 (steady-state
loop with interval-aggregated stats via one all_reduce, throughput and peak
memory reporting, checkpoint cadence, torch.profiler factory). MI355X
notes: torch.profiler rides roctracer on ROCm; the NCCL backend is RCCL.
"""

import os
import time
from dataclasses import asdict
from datetime import timedelta

import torch
import torch.distributed as dist


class LambdaLR:
    """Minimal LambdaLR for ShardedAdamW (same semantics as
    torch.optim.lr_scheduler.LambdaLR: lr = initial_lr * fn(epoch))."""

    def __init__(self, optimizer, lr_lambda, last_epoch=-1):
        self.optimizer = optimizer
        self.lr_lambda = lr_lambda
        self.last_epoch = last_epoch
        for g in optimizer.param_groups:
            g.setdefault("initial_lr", g["lr"])
        self.step()

    def step(self):
        self.last_epoch += 1
        self._last_lr = []
        for g in self.optimizer.param_groups:
            g["lr"] = g["initial_lr"] * self.lr_lambda(self.last_epoch)
            self._last_lr.append(g["lr"])

    def get_last_lr(self):
        return self._last_lr

    def state_dict(self):
        return {"last_epoch": self.last_epoch}

    def load_state_dict(self, sd):
        self.last_epoch = sd["last_epoch"]


def setup():
    """init_process_group on RCCL (reference: train_utils.py:183-184)."""
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend, timeout=timedelta(hours=1))


def setup_environ_flags():
    os.environ["TORCH_SHOW_CPP_STACKTRACES"] = str(1)
    os.environ["TORCH_NCCL_ASYNC_ERROR_HANDLING"] = str(1)
    # dmabuf IPC is the only supported mode on this driver stack
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")


def get_profiler(cfg, rank):
    """torch.profiler factory (reference: train_utils.py:256-271)."""
    if not cfg.use_profiler:
        return None
    if cfg.profiler_rank0_only and rank != 0:
        return None
    return torch.profiler.profile(
        activities=[torch.profiler.ProfilerActivity.CPU,
                    torch.profiler.ProfilerActivity.CUDA],
        schedule=torch.profiler.schedule(wait=1, warmup=2, active=3, repeat=1),
        on_trace_ready=torch.profiler.tensorboard_trace_handler("profile_traces"),
        profile_memory=True,
        with_stack=False,
        record_shapes=True,
    )


def get_tracker(cfg, rank):
    if not cfg.tracker or rank != 0:
        return None
    if cfg.tracker not in ("wandb", "aim"):
        raise ValueError(f"tracker {cfg.tracker} not supported (wandb|aim)")
    if cfg.tracker == "wandb":
        try:
            import wandb
        except ImportError:
            print("WARNING: wandb not installed; tracking disabled")
            return None
        run = wandb.init(project=cfg.tracker_project_name, dir=cfg.tracker_dir,
                         resume="allow", id=cfg.tracker_run_id)
        run.config.update(asdict(cfg))

        class _W:
            def log(self, d, step):
                wandb.log(d, step=step)
        return _W()
    if cfg.tracker == "aim":
        try:
            from aim import Run
        except ImportError:
            print("WARNING: aim not installed; tracking disabled")
            return None
        run = Run(experiment=cfg.tracker_project_name, repo=cfg.tracker_dir,
                  run_hash=cfg.tracker_run_id)
        run["hparams"] = asdict(cfg)

        class _A:
            def log(self, d, step):
                for k, v in d.items():
                    run.track(v, name=k, step=step)
        return _A()


def train(cfg, model, local_rank, rank, train_loader, optimizer, scheduler,
          profiler=None, checkpointer=None, start_step=0, n_tok=0,
          tracker=None, scaler=None):
    """Steady-state training loop (reference: train_utils.py:21-180).
    `scaler`: DynamicGradScaler — required for the fp16 policy, a no-op
    pass-through otherwise (built automatically from cfg when None)."""
    from fms_fsdp_amd.parallel import DynamicGradScaler
    from fms_fsdp_amd.parallel.policies import needs_loss_scaling
    if scaler is None:
        scaler = DynamicGradScaler(enabled=needs_loss_scaling(cfg))
    model.train()
    world = dist.get_world_size() if dist.is_initialized() else 1
    device = torch.device("cuda", local_rank) if torch.cuda.is_available() \
        else torch.device("cpu")

    ddp_stats = torch.zeros(3, device=device)  # loss sum, gnorm sum, count
    start = time.time()
    loop_start = time.time()
    elapsed_tokens = 0
    train_result = {}

    for batch_idx, (inp, label) in enumerate(train_loader, start=start_step + 1):
        if batch_idx > cfg.num_steps:
            break
        inp = inp.to(device, non_blocking=True)
        label = label.to(device, non_blocking=True)

        optimizer.zero_grad()
        loss = model(inp, labels=label)
        scaler.scale_loss(loss).backward()
        ddp_stats[0] += loss.detach().float()
        gnorm, _ = scaler.clip_and_step(model, optimizer,
                                        cfg.grad_clip_thresh)
        # (short-circuit keeps the default path free of host syncs)
        if not scaler.enabled or bool(torch.isfinite(gnorm)):
            ddp_stats[1] += gnorm.detach().float()
        ddp_stats[2] += 1
        scheduler.step()

        if profiler:
            profiler.step()

        if batch_idx % cfg.report_interval == 0:
            if dist.is_initialized():
                dist.all_reduce(ddp_stats, op=dist.ReduceOp.SUM)
            train_loss = (ddp_stats[0] / ddp_stats[2]).item()
            train_gnorm = (ddp_stats[1] / ddp_stats[2]).item()
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            elapsed = time.time() - loop_start
            elapsed_tokens += cfg.report_interval * world * cfg.batch_size * cfg.seq_length
            tok_per_gpu_sec = (cfg.report_interval * cfg.batch_size
                               * cfg.seq_length) / elapsed
            if rank == 0:
                mem = (torch.cuda.max_memory_allocated() / 2**30
                       if torch.cuda.is_available() else 0)
                overall = elapsed_tokens / (time.time() - start) / world
                tokens_per_day = elapsed_tokens / (time.time() - start) * 86400
                print(f"step {batch_idx}: loss={train_loss:.4f} "
                      f"gnorm={train_gnorm:.3f} lr={scheduler.get_last_lr()[0]:.2e} "
                      f"tok/s/gpu={tok_per_gpu_sec:,.0f} "
                      f"(overall {overall:,.0f}) "
                      f"tokens_seen={n_tok + elapsed_tokens:,} "
                      f"tok/day={tokens_per_day:,.0f} "
                      f"peak_mem={mem:.1f}GiB")
                train_result = {"loss": train_loss, "gnorm": train_gnorm,
                                "tok_per_gpu_sec": tok_per_gpu_sec}
                if tracker:
                    tracker.log({"loss": train_loss, "grad_norm": train_gnorm,
                                 "tok_per_gpu_sec": tok_per_gpu_sec,
                                 "lr": scheduler.get_last_lr()[0],
                                 "tokens_seen": n_tok + elapsed_tokens},
                                step=batch_idx)
            ddp_stats.zero_()
            if torch.cuda.is_available():
                torch.cuda.reset_peak_memory_stats()
            loop_start = time.time()

        if checkpointer is not None and batch_idx % cfg.checkpoint_interval == 0:
            tokens_seen = n_tok + (batch_idx - start_step) * world \
                * cfg.batch_size * cfg.seq_length
            checkpointer.save(batch_idx, model, optimizer, train_loader,
                              tokens_seen=tokens_seen)

    if checkpointer is not None:
        checkpointer.wait()   # join any in-flight async checkpoint write
    if rank == 0:
        total = time.time() - start
        print(f"training done: {total:.1f}s")
    return train_result
