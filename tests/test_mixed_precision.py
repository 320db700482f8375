"""Mixed-precision policies (reference mixed_precision.py:5-27) and the
dynamic loss scaler for fp16."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from fms_fsdp_amd.config import train_config
from fms_fsdp_amd.models import Llama, LlamaBlock, LlamaConfig
from fms_fsdp_amd.parallel import (DynamicGradScaler, ShardedAdamW,
                                   ShardedModel)
from fms_fsdp_amd.parallel.policies import (get_mixed_precision_dtypes,
                                            needs_loss_scaling,
                                            resolve_mp_policy)

PORT = 29720


def test_policy_resolution():
    cfg = train_config()
    assert resolve_mp_policy(cfg) in ("bf16", "fp16")  # auto + mixed
    cfg.mixed_precision = False
    assert resolve_mp_policy(cfg) == "fp32"
    cfg.mp_policy = "bf16_working"
    assert get_mixed_precision_dtypes(cfg) == (torch.float32, torch.bfloat16)
    cfg.mp_policy = "fp16"
    assert get_mixed_precision_dtypes(cfg) == (torch.float16, torch.float16)
    assert needs_loss_scaling(cfg)
    cfg.mp_policy = "bf16"
    assert not needs_loss_scaling(cfg)
    cfg.reduce_dtype = "fp32"  # explicit override of the policy triple
    assert get_mixed_precision_dtypes(cfg) == (torch.bfloat16, torch.float32)
    cfg.mp_policy = "nope"
    with pytest.raises(ValueError):
        resolve_mp_policy(cfg)


def test_scaler_backoff_and_growth():
    sc = DynamicGradScaler(enabled=True, init_scale=8.0, growth_interval=2)

    class FakeModel:
        def __init__(self, norm):
            self.norm = norm
            self._clip_coef = None

        def clip_grad_norm_(self, max_norm):
            n = torch.tensor(self.norm)
            self._clip_coef = torch.clamp(max_norm / (n + 1e-6), max=1.0)
            return n

    class FakeOpt:
        stepped = 0

        def step(self):
            FakeOpt.stepped += 1

    opt = FakeOpt()
    # overflow: skipped, scale halves
    _, stepped = sc.clip_and_step(FakeModel(float("inf")), opt, 1.0)
    assert not stepped and sc.scale == 4.0 and FakeOpt.stepped == 0
    # good steps: scale doubles after growth_interval consecutive
    m = FakeModel(2.0)
    _, s1 = sc.clip_and_step(m, opt, 1.0)
    # unscale folded into coef: true norm = 2/4 = 0.5 < 1 => coef = 1/scale
    assert s1 and abs(m._clip_coef.item() - 1 / 4.0) < 1e-6
    sc.clip_and_step(FakeModel(2.0), opt, 1.0)
    assert sc.scale == 8.0 and FakeOpt.stepped == 2
    # round-trip
    sd = sc.state_dict()
    sc2 = DynamicGradScaler()
    sc2.load_state_dict(sd)
    assert sc2.scale == sc.scale


def _mk(policy, seed=0):
    torch.manual_seed(seed)
    cfg = LlamaConfig(src_vocab_size=128, emb_dim=64, nheads=4, kvheads=2,
                      nlayers=2, max_expected_seq_len=64)
    m = Llama(cfg)
    m.reset_parameters()
    tc = train_config()
    tc.mp_policy = policy
    pd, rd = get_mixed_precision_dtypes(tc)
    sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                      param_dtype=pd, reduce_dtype=rd)
    return sm, ShardedAdamW(sm, lr=1e-3), tc


@pytest.mark.parametrize("policy", ["fp16", "bf16_working", "fp32"])
def test_policy_trains_cpu(policy):
    """Each policy runs end-to-end on CPU: finite losses, params update,
    dtype layout as declared."""
    from fms_fsdp_amd.utils.train import train

    sm, opt, tc = _mk(policy)
    pd, rd = get_mixed_precision_dtypes(tc)
    for u in sm.all_units:
        assert u.flat_param.dtype == pd
        assert u.grad_shard.dtype == rd
        assert u.master_shard.dtype == torch.float32
    g = torch.Generator().manual_seed(1)
    loader = [(torch.randint(0, 128, (2, 32), generator=g),
               torch.randint(0, 128, (2, 32), generator=g))
              for _ in range(3)]
    tc.num_steps = 3
    tc.report_interval = 10
    tc.checkpoint_interval = 10 ** 9
    before = [u.master_shard.clone() for u in sm.all_units]
    train(tc, sm, 0, 0, loader, opt, _const_sched(opt))
    for u, b in zip(sm.all_units, before):
        assert not torch.equal(u.master_shard, b), u.name
        assert torch.isfinite(u.master_shard).all()


def _const_sched(opt):
    from fms_fsdp_amd.utils.train import LambdaLR
    return LambdaLR(opt, lambda x: 1.0)


def _worker(rank, world, policy, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(PORT + {"fp16": 1, "bf16_working": 2}[policy])
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        sm, opt, tc = _mk(policy)
        scaler = DynamicGradScaler(enabled=needs_loss_scaling(tc),
                                   init_scale=4.0)
        g = torch.Generator().manual_seed(42)
        x = torch.randint(0, 128, (2, 32), generator=g)
        y = torch.randint(0, 128, (2, 32), generator=g)
        losses = []
        for _ in range(3):
            opt.zero_grad()
            loss = sm(x, labels=y)
            scaler.scale_loss(loss).backward()
            scaler.clip_and_step(sm, opt, 1.0)
            losses.append(loss.item())
        if rank == 0:
            q.put(("ok", losses))
    except Exception as e:
        q.put(("err", f"{type(e).__name__}: {e}"))
        raise
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("policy", ["fp16", "bf16_working"])
def test_policy_world2_matches_single(policy):
    """Identical data on both ranks: the sharded run must match the
    single-process trajectory under the same policy (reduce in the
    policy's reduce_dtype is numerically transparent when both ranks
    hold identical grads)."""
    sm, opt, tc = _mk(policy)
    scaler = DynamicGradScaler(enabled=needs_loss_scaling(tc), init_scale=4.0)
    g = torch.Generator().manual_seed(42)
    x = torch.randint(0, 128, (2, 32), generator=g)
    y = torch.randint(0, 128, (2, 32), generator=g)
    ref = []
    for _ in range(3):
        opt.zero_grad()
        loss = sm(x, labels=y)
        scaler.scale_loss(loss).backward()
        scaler.clip_and_step(sm, opt, 1.0)
        ref.append(loss.item())

    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.spawn(_worker, args=(2, policy, q), nprocs=2, join=True)
    status, got = q.get()
    assert status == "ok", got
    for a, b in zip(ref, got):
        assert abs(a - b) < 2e-3, (ref, got)


def test_fp16_overflow_skip_and_recover():
    """An injected overflow step must be SKIPPED (params/moments
    untouched, scale backed off) and training must continue normally
    afterwards — the full scaler path through the sharded runtime."""
    sm, opt, tc = _mk("fp16", seed=3)
    scaler = DynamicGradScaler(enabled=True, init_scale=8.0)
    g = torch.Generator().manual_seed(5)
    x = torch.randint(0, 128, (2, 32), generator=g)
    y = torch.randint(0, 128, (2, 32), generator=g)

    # one normal step
    opt.zero_grad()
    scaler.scale_loss(sm(x, labels=y)).backward()
    _, stepped = scaler.clip_and_step(sm, opt, 1.0)
    assert stepped
    before = [u.master_shard.clone() for u in sm.all_units]
    moments = [u.exp_avg.clone() for u in sm.all_units]
    step_count = opt.step_count

    # inject an overflow into the grads
    opt.zero_grad()
    scaler.scale_loss(sm(x, labels=y)).backward()
    sm.all_units[0].grad_shard.view(-1)[0] = float("inf")
    gnorm, stepped = scaler.clip_and_step(sm, opt, 1.0)
    assert not stepped and not bool(torch.isfinite(gnorm))
    assert scaler.scale == 4.0                       # backed off
    for u, b, m in zip(sm.all_units, before, moments):
        assert torch.equal(u.master_shard, b), u.name # update skipped
        assert torch.equal(u.exp_avg, m), u.name      # moments untouched
    assert opt.step_count == step_count               # bias corr. frozen

    # recovery: next clean step applies normally
    opt.zero_grad()
    loss = sm(x, labels=y)
    scaler.scale_loss(loss).backward()
    _, stepped = scaler.clip_and_step(sm, opt, 1.0)
    assert stepped and torch.isfinite(loss)
    assert not torch.equal(sm.all_units[0].master_shard, before[0])
