"""Speculator training internals (parity target: reference
speculator/train_speculator_utils.py).

Stage 1 (steps <= stage2_start_step): teacher-forced — the frozen base
model produces hidden states for ground-truth text in parallel; head i at
position t predicts token t+2+i (reference stage1_loss, utils:122-171).

Stage 2: the batch is reshaped into many short prompts; the base model
GENERATES continuations with its KV cache, and the speculator learns to
match the base model's own sampled behavior (reference stage2_loss,
utils:175-242).

On-demand checkpointing by touching `<ckpt_save_path>/do_ckpt`
(reference utils:246-260).
"""

import os
import time

import torch
import torch.distributed as dist
import torch.nn.functional as F


def _tp_chunk(t, tp_group):
    """TP mode runs the frozen base on the all-gathered (tp*b, ...) batch;
    each rank trains its speculator only on ITS b-sized slice (reference
    utils:158-162, 224-232 — omitting this mismatches the batch dims)."""
    if tp_group is None or dist.get_world_size(tp_group) == 1:
        return t
    return t.chunk(dist.get_world_size(tp_group))[dist.get_rank(tp_group)]


def stage1_loss(cfg, model, speculator, base_model_input, inp, ddp_stats,
                tp_group=None):
    with torch.no_grad():
        _, embeds = model(base_model_input[:, : -speculator.n_predict - 1],
                          include_embeds=True)
        embeds = _tp_chunk(embeds, tp_group)
    preds = speculator(embeds.detach(), inp[:, 1:])
    losses = []
    for i in range(preds.size(0)):
        targ = inp[:, i + 2: preds.size(2) + i + 2]
        loss = F.cross_entropy(preds[i].reshape(-1, preds.size(3)),
                               targ.long().reshape(-1))
        losses.append(loss)
        ddp_stats[2 + i] += loss.item()
    return sum(losses), ddp_stats, inp.numel()


def stage2_loss(cfg, model, speculator, base_model_input, inp, ddp_stats,
                tp_group=None):
    n = speculator.n_predict
    with torch.no_grad():
        grow = cfg.stage2_batch_size // cfg.batch_size
        assert cfg.stage2_prompt_length * grow <= cfg.seq_length, \
            "batch too small for stage-2 partition"
        prompts = base_model_input[:, : cfg.stage2_prompt_length * grow] \
            .reshape(base_model_input.size(0) * grow, cfg.stage2_prompt_length)
        targs, embeds = model.generate(prompts, cfg.stage2_seq_length,
                                       do_sample=True, include_embeds=True)
        targs = _tp_chunk(targs, tp_group)
        embeds = _tp_chunk(embeds, tp_group)
        gen = targs[:, -cfg.stage2_seq_length:]           # generated tokens
        state = embeds[:, : cfg.stage2_seq_length - n]
    preds = speculator(state.detach(), gen[:, :-1].detach())
    losses = []
    for i in range(preds.size(0)):
        targ = gen[:, i + 1: preds.size(2) + i + 1]
        loss = F.cross_entropy(preds[i].reshape(-1, preds.size(3)),
                               targ.long().reshape(-1))
        losses.append(loss)
        ddp_stats[2 + i] += loss.item()
    return sum(losses), ddp_stats, gen.numel()


def do_ckpt(ckpt_save_path, reset=False):
    f = os.path.join(ckpt_save_path, "do_ckpt")
    if not os.path.exists(f):
        return False
    if reset:
        with open(f, "w") as fd:
            fd.write("0")
        return False
    with open(f) as fd:
        return fd.read().strip() == "1"


def train_speculator(cfg, model, speculator, local_rank, rank, world_size,
                     train_loader, optimizer, scheduler, checkpointer,
                     start_step=0, n_tok=0, profiler=None, tp_group=None):
    """Steady-state speculator train loop (reference utils:263-427)."""
    model.eval()
    speculator.train()
    device = torch.device("cuda", local_rank) if torch.cuda.is_available() \
        else torch.device("cpu")

    ddp_stats = torch.zeros(2 + speculator.n_predict, device=device)
    start = time.time()
    loop_start = time.time()
    elapsed_tokens = 0
    for batch_idx, (inp, _) in enumerate(train_loader, start=start_step + 1):
        if batch_idx > cfg.num_steps:
            break
        inp = inp.to(device, non_blocking=True)

        # TP: the frozen base model runs on the whole tp-group batch
        if tp_group is not None:
            bmi = torch.empty(inp.size(0) * dist.get_world_size(tp_group),
                              inp.size(1), dtype=inp.dtype, device=device)
            dist.all_gather_into_tensor(bmi, inp, group=tp_group)
        else:
            bmi = inp

        optimizer.zero_grad()
        if batch_idx <= cfg.stage2_start_step:
            loss, ddp_stats, step_tok = stage1_loss(
                cfg, model, speculator, bmi, inp, ddp_stats, tp_group)
        else:
            loss, ddp_stats, step_tok = stage2_loss(
                cfg, model, speculator, bmi, inp, ddp_stats, tp_group)
        loss.backward()
        ddp_stats[0] += speculator.clip_grad_norm_(cfg.grad_clip_thresh) \
            if hasattr(speculator, "clip_grad_norm_") else \
            torch.nn.utils.clip_grad_norm_(speculator.parameters(),
                                           cfg.grad_clip_thresh)
        ddp_stats[1] += 1
        optimizer.step()
        scheduler.step()
        elapsed_tokens += step_tok * world_size

        if profiler:
            profiler.step()

        if batch_idx % cfg.report_interval == 0:
            if dist.is_initialized():
                dist.all_reduce(ddp_stats, op=dist.ReduceOp.SUM)
            interval = time.time() - loop_start
            if rank == 0:
                head_losses = [round((ddp_stats[2 + i] / ddp_stats[1]).item(), 4)
                               for i in range(speculator.n_predict)]
                print(f"step {batch_idx}: gnorm="
                      f"{(ddp_stats[0] / ddp_stats[1]).item():.3f} "
                      f"lr={scheduler.get_last_lr()[0]:.2e} "
                      f"head_losses={head_losses} "
                      f"tok/s={elapsed_tokens / (time.time() - start):,.0f}")
            ddp_stats.zero_()
            loop_start = time.time()

        on_demand = do_ckpt(cfg.ckpt_save_path)
        if batch_idx % cfg.checkpoint_interval == 0 or on_demand:
            if on_demand:
                do_ckpt(cfg.ckpt_save_path, reset=True)
            checkpointer.save(batch_idx, speculator, optimizer, train_loader,
                              tokens_seen=n_tok + elapsed_tokens)
    return ddp_stats
