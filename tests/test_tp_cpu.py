"""TP correctness (gloo, world 2): the tp-sharded frozen base model must
produce the same logits as the unsharded model."""

import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29695"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from fms_fsdp_amd.models import Llama, LlamaConfig
        from fms_fsdp_amd.parallel.tp import tp_shard_llama
        torch.manual_seed(0)
        cfg = LlamaConfig(src_vocab_size=64, emb_dim=32, nheads=4, kvheads=2,
                          nlayers=2, max_expected_seq_len=64)
        m = Llama(cfg)
        m.reset_parameters()
        m.eval()
        x = torch.randint(0, 64, (2, 16),
                          generator=torch.Generator().manual_seed(5))
        with torch.no_grad():
            ref = m(x)
        tp_shard_llama(m, dist.group.WORLD)
        with torch.no_grad():
            got = m(x)
        err = (ref - got).abs().max().item()
        if rank == 0:
            q.put(("ok", err))
    except Exception as e:
        q.put(("err", f"{type(e).__name__}: {e}"))
        raise
    finally:
        dist.destroy_process_group()


def test_tp_matches_unsharded():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.spawn(_worker, args=(2, q), nprocs=2, join=True)
    status, err = q.get()
    assert status == "ok", err
    assert err < 1e-4, err


def _loop_worker(rank, world, tmpdir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29696"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from fms_fsdp_amd.config import train_config
        from fms_fsdp_amd.models import Llama, LlamaConfig
        from fms_fsdp_amd.models.speculator import MLPSpeculator
        from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
        from fms_fsdp_amd.parallel.tp import tp_shard_llama
        from fms_fsdp_amd.utils.checkpointing import Checkpointer
        from fms_fsdp_amd.utils.train import LambdaLR
        from speculator.train_speculator_utils import train_speculator

        torch.manual_seed(0)
        mcfg = LlamaConfig(src_vocab_size=64, emb_dim=32, nheads=4,
                           kvheads=2, nlayers=2, max_expected_seq_len=128)
        base = Llama(mcfg)
        base.reset_parameters()
        base.eval()
        for p in base.parameters():
            p.requires_grad_(False)
        tp_shard_llama(base, dist.group.WORLD)

        torch.manual_seed(0)
        spec = MLPSpeculator(emb_dim=32, inner_dim=32, vocab_size=64,
                             n_predict=2)
        spec.reset_parameters()
        sspec = ShardedModel(spec, MLPSpeculator, sharding_strategy="ddp",
                             param_dtype=torch.float32)
        sspec.n_predict = spec.n_predict
        opt = ShardedAdamW(sspec, lr=1e-3)
        sched = LambdaLR(opt, lambda x: 1.0)

        cfg = train_config()
        cfg.num_steps = 5
        cfg.stage2_start_step = 3   # steps 1-3 stage-1, 4-5 stage-2
        cfg.batch_size = 2
        cfg.seq_length = 64
        cfg.stage2_batch_size = 8
        cfg.stage2_prompt_length = 8
        cfg.stage2_seq_length = 24
        cfg.report_interval = 2
        cfg.checkpoint_interval = 1000
        cfg.ckpt_save_path = tmpdir

        # per-rank DIFFERENT data (the dp dimension of the tp layout)
        g = torch.Generator().manual_seed(100 + rank)
        loader = [(torch.randint(0, 64, (2, 64), generator=g), None)
                  for _ in range(cfg.num_steps)]
        ck = Checkpointer(tmpdir, 2, "ddp", rank, rank)
        stats = train_speculator(cfg, base, sspec, 0, rank, world, loader,
                                 opt, sched, ck, tp_group=dist.group.WORLD)
        shards = [u.master_shard.clone() for u in sspec.all_units]
        q.put((f"ok{rank}", [s.sum().item() for s in shards],
               [float(s.abs().max()) for s in shards]))
    except Exception as e:
        q.put(("err", f"{type(e).__name__}: {e}", None))
        raise
    finally:
        dist.destroy_process_group()


def test_tp_speculator_training_loop(tmp_path):
    """World-2 TP end-to-end: stage-1 AND stage-2 steps of the actual
    train_speculator loop run with tp_group (the all-gather + per-rank
    embeds chunk path), and the speculator stays identical across ranks
    (its grads all-reduce over the full world, reference speculator
    mesh)."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.spawn(_loop_worker, args=(2, str(tmp_path), q), nprocs=2, join=True)
    res = {}
    for _ in range(2):
        tag, sums, maxes = q.get()
        assert tag != "err", sums
        res[tag] = (sums, maxes)
    assert res["ok0"] == res["ok1"], "speculator diverged across dp ranks"
    assert all(abs(x) > 0 for x in res["ok0"][1]), "speculator never trained"


def _entry_worker(rank, world, tmpdir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29697"
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from speculator import train_speculator as ts
        ts.main(model_variant="llama2_125m", use_dummy_dataset=True,
                batch_size=1, seq_length=128, num_steps=2,
                report_interval=1, checkpoint_interval=100,
                mixed_precision=False, n_speculator_heads=2,
                speculator_width=64, stage2_start_step=10,
                model_path="/nonexistent", ckpt_save_path=tmpdir,
                ckpt_load_path=tmpdir, vocab_size=256, learning_rate=1e-4,
                sharding_strategy="tp", tp_size=2)
        if rank == 0:
            q.put(("ok", None))
    except Exception as e:
        q.put(("err", f"{type(e).__name__}: {e}"))
        raise
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_speculator_entry_tp_world2(tmp_path):
    """The full train_speculator entry under sharding_strategy=tp at
    world 2: dp x tp device-mesh construction (reference
    train_speculator.py:128-142), tp_shard_llama, and the TP loop."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.spawn(_entry_worker, args=(2, str(tmp_path), q), nprocs=2, join=True)
    status, err = q.get()
    assert status == "ok", err
