"""Multi-process (gloo, world_size=2) tests of the sharded runtime —
coverage the reference lacks entirely (SURVEY.md §4 'the new framework must
add real distributed tests')."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

PORT = 29650


def _single_process_reference(seed=0, steps=3, lr=1e-3):
    from fms_fsdp_amd.models import Llama, LlamaBlock, LlamaConfig
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
    torch.manual_seed(seed)
    cfg = LlamaConfig(src_vocab_size=128, emb_dim=64, nheads=4, kvheads=2,
                      nlayers=3, max_expected_seq_len=64)
    m = Llama(cfg)
    m.reset_parameters()
    sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                      param_dtype=torch.float32)
    opt = ShardedAdamW(sm, lr=lr)
    g = torch.Generator().manual_seed(42)
    x = torch.randint(0, 128, (2, 32), generator=g)
    y = torch.randint(0, 128, (2, 32), generator=g)
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        loss = sm(x, labels=y)
        loss.backward()
        sm.clip_grad_norm_(1.0)
        opt.step()
        losses.append(loss.item())
    return losses


def _worker(rank, world, port, strat, reshard, q, steps=3):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from fms_fsdp_amd.models import Llama, LlamaBlock, LlamaConfig
        from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
        torch.manual_seed(0)
        cfg = LlamaConfig(src_vocab_size=128, emb_dim=64, nheads=4, kvheads=2,
                          nlayers=3, max_expected_seq_len=64)
        m = Llama(cfg)
        m.reset_parameters()
        sm = ShardedModel(m, LlamaBlock, sharding_strategy=strat,
                          param_dtype=torch.float32, reshard_after_forward=reshard,
                          intra_node_size=2)
        opt = ShardedAdamW(sm, lr=1e-3)
        g = torch.Generator().manual_seed(42)
        x = torch.randint(0, 128, (2, 32), generator=g)
        y = torch.randint(0, 128, (2, 32), generator=g)
        losses = []
        for _ in range(steps):
            opt.zero_grad()
            loss = sm(x, labels=y)
            loss.backward()
            sm.clip_grad_norm_(1.0)
            opt.step()
            losses.append(loss.item())
        if rank == 0:
            q.put(("ok", losses))
    except Exception as e:  # pragma: no cover
        q.put(("err", f"{type(e).__name__}: {e}"))
        raise
    finally:
        dist.destroy_process_group()


def _run_world2(strat, reshard, port):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.spawn(_worker, args=(2, port, strat, reshard, q), nprocs=2, join=True)
    status, payload = q.get()
    assert status == "ok", payload
    return payload


@pytest.mark.parametrize("strat,reshard,port", [
    ("fsdp", False, PORT + 1),
    ("fsdp", True, PORT + 2),
    ("hsdp", False, PORT + 3),
    ("ddp", False, PORT + 4),
])
def test_world2_matches_single_process(strat, reshard, port):
    """Identical batches on both ranks => identical training trajectory to
    a single process (sharding must be numerically transparent)."""
    ref = _single_process_reference()
    got = _run_world2(strat, reshard, port)
    for a, b in zip(ref, got):
        assert abs(a - b) < 1e-5, (strat, ref, got)


def test_world8_fsdp_matches_single_process():
    """8-rank FSDP over gloo — the CPU proxy for the driver's single-node
    8-GPU scaling run (one shard group of 8, the exact topology bench.py
    sees at N=8). Trajectory must match a single process."""
    ref = _single_process_reference()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.spawn(_worker, args=(8, PORT + 30, "fsdp", False, q), nprocs=8,
             join=True)
    status, payload = q.get()
    assert status == "ok", payload
    for a, b in zip(ref, payload):
        assert abs(a - b) < 1e-5, (ref, payload)


def _ckpt_worker(rank, world, port, tmpdir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from fms_fsdp_amd.models import Llama, LlamaBlock, LlamaConfig
        from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
        from fms_fsdp_amd.utils.checkpointing import Checkpointer
        torch.manual_seed(0)
        cfg = LlamaConfig(src_vocab_size=128, emb_dim=64, nheads=4, kvheads=2,
                          nlayers=2, max_expected_seq_len=64)
        m = Llama(cfg)
        m.reset_parameters()
        sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                          param_dtype=torch.float32)
        opt = ShardedAdamW(sm, lr=1e-3)
        g = torch.Generator().manual_seed(7)
        x = torch.randint(0, 128, (2, 16), generator=g)
        y = torch.randint(0, 128, (2, 16), generator=g)
        for _ in range(2):
            opt.zero_grad()
            sm(x, labels=y).backward()
            opt.step()
        ck = Checkpointer(tmpdir, 3, "fsdp", rank, rank)
        ck.save(2, sm, opt, None, tokens_seen=123)
        if rank == 0:
            torch.save([u.master_shard.clone() for u in sm.all_units],
                       os.path.join(tmpdir, "rank0_shards.pth"))
            q.put(("ok", None))
    except Exception as e:
        q.put(("err", f"{type(e).__name__}: {e}"))
        raise
    finally:
        dist.destroy_process_group()


def test_checkpoint_save_and_reshard_load(tmp_path):
    """Save at world=2, reload single-process (world=1): master shards must
    reassemble exactly (rescalable checkpoint, SURVEY.md hard-part 5)."""
    tmpdir = str(tmp_path)
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.spawn(_ckpt_worker, args=(2, PORT + 10, tmpdir, q), nprocs=2, join=True)
    status, payload = q.get()
    assert status == "ok", payload
    rank0_shards = torch.load(os.path.join(tmpdir, "rank0_shards.pth"),
                              weights_only=False)

    from fms_fsdp_amd.models import Llama, LlamaBlock, LlamaConfig
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
    from fms_fsdp_amd.utils.checkpointing import Checkpointer, get_latest
    torch.manual_seed(0)
    cfg = LlamaConfig(src_vocab_size=128, emb_dim=64, nheads=4, kvheads=2,
                      nlayers=2, max_expected_seq_len=64)
    m = Llama(cfg)
    m.reset_parameters()
    sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                      param_dtype=torch.float32)
    opt = ShardedAdamW(sm, lr=1e-3)
    ck = Checkpointer(tmpdir, 3, "fsdp", 0, 0)
    _, _, _, step, tokens, resuming = ck.load(sm, opt, None, path="")
    assert step == 2 and tokens == 123 and resuming
    # world=1 master shard must equal the concatenation of the world=2 halves
    for u, old0 in zip(sm.all_units, rank0_shards):
        half = old0.numel()
        assert torch.equal(u.master_shard[:half], old0), u.name


def test_async_checkpoint_roundtrip(tmp_path):
    """async_save=True: files written by a background thread, metadata
    last; wait() joins; the reloaded state is identical to sync-saved."""
    from fms_fsdp_amd.models import Llama, LlamaBlock, LlamaConfig
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
    from fms_fsdp_amd.utils.checkpointing import Checkpointer
    torch.manual_seed(3)
    cfg = LlamaConfig(src_vocab_size=128, emb_dim=64, nheads=4, kvheads=2,
                      nlayers=2, max_expected_seq_len=64)
    m = Llama(cfg)
    m.reset_parameters()
    sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                      param_dtype=torch.float32)
    opt = ShardedAdamW(sm, lr=1e-3)
    x = torch.randint(0, 128, (1, 64))
    y = torch.randint(0, 128, (1, 64))
    opt.zero_grad()
    sm(x, labels=y).backward()
    opt.step()
    ck = Checkpointer(str(tmp_path), 3, "fsdp", 0, 0, async_save=True)
    out = ck.save(7, sm, opt, None, tokens_seen=42)
    ck.wait()
    assert os.path.exists(os.path.join(out, "metadata.pth"))
    shards = [u.master_shard.clone() for u in sm.all_units]

    # mutate, then reload: state must come back bit-exact
    opt.zero_grad()
    sm(x, labels=y).backward()
    opt.step()
    ck2 = Checkpointer(str(tmp_path), 3, "fsdp", 0, 0)
    _, _, _, step, tokens, resuming = ck2.load(sm, opt, None, path="")
    assert step == 7 and tokens == 42 and resuming
    for u, ref in zip(sm.all_units, shards):
        assert torch.equal(u.master_shard, ref), u.name


def test_consolidate_checkpoint(tmp_path):
    """consolidate_checkpoint reconstructs exact full params offline."""
    from fms_fsdp_amd.models import Llama, LlamaBlock, LlamaConfig
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
    from fms_fsdp_amd.utils.checkpointing import Checkpointer, consolidate_checkpoint
    torch.manual_seed(3)
    cfg = LlamaConfig(src_vocab_size=128, emb_dim=64, nheads=4, kvheads=2,
                      nlayers=2, max_expected_seq_len=64)
    m = Llama(cfg)
    m.reset_parameters()
    ref = {n: p.detach().clone() for n, p in m.named_parameters()}
    sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                      param_dtype=torch.float32)
    opt = ShardedAdamW(sm, lr=1e-3)
    ck = Checkpointer(str(tmp_path), 3, "fsdp", 0, 0)
    out = ck.save(1, sm, opt, None)
    sd = consolidate_checkpoint(out)
    for n, t in ref.items():
        assert torch.allclose(sd[n], t.float(), atol=1e-6), n


def test_world4_hsdp_two_node_simulation():
    """world=4 as 2 'nodes' x 2 'GPUs' (intra_node_size=2): exercises the
    shard-intra + replicate-inter path incl. the inter-node grad
    all-reduce; must match the single-process trajectory."""
    ref = _single_process_reference()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.spawn(_worker, args=(4, PORT + 20, "hsdp", False, q), nprocs=4,
             join=True)
    status, got = q.get()
    assert status == "ok", got
    for a, b in zip(ref, got):
        assert abs(a - b) < 1e-5, (ref, got)


def _small_cfg():
    from fms_fsdp_amd.models import LlamaConfig
    return LlamaConfig(src_vocab_size=128, emb_dim=64, nheads=4, kvheads=2,
                       nlayers=3, max_expected_seq_len=64)


def test_meta_streamed_init_matches_eager():
    """ShardedModel built from a meta-device model (unit-streamed
    materialization, reference low_cpu_fsdp/param_init.py:9-18) must
    produce bit-identical shards to the eager path under the same seed."""
    from fms_fsdp_amd.models import Llama, LlamaBlock
    from fms_fsdp_amd.parallel import ShardedModel

    m_eager = Llama(_small_cfg())
    torch.manual_seed(11)   # seed right before init: the RNG draws must
    m_eager.reset_parameters()  # line up with the streamed path's
    sm_eager = ShardedModel(m_eager, LlamaBlock, sharding_strategy="fsdp",
                            param_dtype=torch.float32)

    with torch.device("meta"):
        m_meta = Llama(_small_cfg())
    assert all(p.is_meta for p in m_meta.parameters())
    torch.manual_seed(11)
    sm_meta = ShardedModel(m_meta, LlamaBlock, sharding_strategy="fsdp",
                           param_dtype=torch.float32)

    for ue, um in zip(sm_eager.all_units, sm_meta.all_units):
        assert ue.name == um.name
        assert torch.equal(ue.master_shard, um.master_shard), ue.name
    # rope tables were rebuilt, not left as to_empty garbage
    assert torch.equal(m_eager.rot_emb.cos, m_meta.rot_emb.cos)
    # params are views into the flat buffers (originals freed)
    for u in sm_meta.all_units:
        for p, off in zip(u.params, u.offsets):
            assert p.data_ptr() == u.flat_param[off:off + p.numel()].data_ptr()
    # direct-wgrad routing flags survived the to_empty param swap
    n_direct = sum(1 for p in m_meta.parameters()
                   if getattr(p, "_direct_wgrad", False))
    n_direct_eager = sum(1 for p in m_eager.parameters()
                         if getattr(p, "_direct_wgrad", False))
    assert n_direct == n_direct_eager and n_direct > 0


def test_meta_streamed_trains_like_eager():
    """A few optimizer steps starting from streamed-meta init match the
    eager-init trajectory exactly (single process)."""
    from fms_fsdp_amd.models import Llama, LlamaBlock
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW

    def run(meta):
        if meta:
            with torch.device("meta"):
                m = Llama(_small_cfg())
        else:
            m = Llama(_small_cfg())
            torch.manual_seed(5)
            m.reset_parameters()
        torch.manual_seed(5)
        sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                          param_dtype=torch.float32)
        opt = ShardedAdamW(sm, lr=1e-3)
        g = torch.Generator().manual_seed(42)
        x = torch.randint(0, 128, (2, 32), generator=g)
        y = torch.randint(0, 128, (2, 32), generator=g)
        losses = []
        for _ in range(3):
            opt.zero_grad()
            loss = sm(x, labels=y)
            loss.backward()
            sm.clip_grad_norm_(1.0)
            opt.step()
            losses.append(loss.item())
        return losses

    assert run(meta=False) == run(meta=True)


def test_incomplete_checkpoint_falls_back_to_older(tmp_path):
    """A checkpoint interrupted mid-write (missing .done shard marker)
    must be skipped by load() in favor of the previous complete one
    (advisor finding: metadata-present-but-shards-missing)."""
    from fms_fsdp_amd.models import Llama, LlamaBlock
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
    from fms_fsdp_amd.utils.checkpointing import Checkpointer
    torch.manual_seed(3)
    m = Llama(_small_cfg())
    m.reset_parameters()
    sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                      param_dtype=torch.float32)
    opt = ShardedAdamW(sm, lr=1e-3)
    x = torch.randint(0, 128, (1, 64))
    y = torch.randint(0, 128, (1, 64))
    ck = Checkpointer(str(tmp_path), 5, "fsdp", 0, 0)

    opt.zero_grad(); sm(x, labels=y).backward(); opt.step()
    ck.save(1, sm, opt, None, tokens_seen=10)
    good = [u.master_shard.clone() for u in sm.all_units]

    opt.zero_grad(); sm(x, labels=y).backward(); opt.step()
    out2 = ck.save(2, sm, opt, None, tokens_seen=20)
    # simulate a crash that left metadata but a truncated shard write
    os.remove(os.path.join(out2, "model_0_of_1.pth.done"))

    opt.zero_grad(); sm(x, labels=y).backward(); opt.step()
    _, _, _, step, tokens, resuming = ck.load(sm, opt, None, path="")
    assert step == 1 and tokens == 10 and resuming
    for u, ref in zip(sm.all_units, good):
        assert torch.equal(u.master_shard, ref), u.name


def _make_tiny_corpus(root):
    import pyarrow as pa
    schema = pa.schema([pa.field("tokens", pa.uint32())])
    rel = "dataset_1/shard_0.arrow"
    path = os.path.join(root, rel)
    os.makedirs(os.path.dirname(path), exist_ok=True)
    with pa.ipc.new_file(path, schema) as w:
        for i in range(200):
            w.write(pa.record_batch(
                [pa.array(range(i * 100, i * 100 + 100), pa.uint32())],
                schema=schema))
    os.makedirs(os.path.join(root, "meta"), exist_ok=True)
    with open(os.path.join(root, "meta", "counts.csv"), "w") as f:
        f.write("dataset/filename,documents,tokens\n")
        f.write(f"data/{rel},200,20000\n")


def test_checkpointer_resume_real_pipeline(tmp_path):
    """End-to-end resume THROUGH Checkpointer with a real dataset
    pipeline: save mid-stream, rebuild everything, load() must restore
    the loader via load_from_path and continue token-exact (advisor
    high-severity finding: the old path crashed on a flat state dict)."""
    from fms_fsdp_amd.config import train_config
    from fms_fsdp_amd.data import get_data_loader
    from fms_fsdp_amd.models import Llama, LlamaBlock
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
    from fms_fsdp_amd.utils.checkpointing import Checkpointer

    data_root = str(tmp_path / "data")
    os.makedirs(data_root, exist_ok=True)
    _make_tiny_corpus(data_root)
    save_root = str(tmp_path / "save")

    def mk_cfg():
        cfg = train_config()
        cfg.data_path = data_root
        cfg.datasets = "dataset_1"
        cfg.weights = "1"
        cfg.seq_length = 64
        cfg.vocab_size = 32000
        cfg.batch_size = 2
        cfg.num_workers = 0        # state lives in-process => exact resume
        cfg.checkpoint_interval = 10**9  # CheckpointDataset never auto-saves
        cfg.ckpt_save_path = save_root
        cfg.ckpt_load_path = save_root
        cfg.eos_token = 0
        cfg.bos_token = None
        cfg.logical_shards = 8
        return cfg

    def mk_model_opt(seed=3):
        from fms_fsdp_amd.models import LlamaConfig
        torch.manual_seed(seed)
        # corpus token values go up to ~20100: vocab must cover them
        m = Llama(LlamaConfig(src_vocab_size=20608, emb_dim=64, nheads=4,
                              kvheads=2, nlayers=2, max_expected_seq_len=64))
        m.reset_parameters()
        sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                          param_dtype=torch.float32)
        return sm, ShardedAdamW(sm, lr=1e-3)

    sm, opt = mk_model_opt()
    dl = get_data_loader(mk_cfg(), 0, 1)
    it = iter(dl)
    ck = Checkpointer(save_root, 3, "fsdp", 0, 0)
    for step in range(1, 6):
        x, y = next(it)
        opt.zero_grad()
        sm(x.long(), labels=y.long()).backward()
        opt.step()
    ck.save(5, sm, opt, dl, tokens_seen=5 * 2 * 64)
    expect = [next(it) for _ in range(4)]      # the continuation

    sm2, opt2 = mk_model_opt(seed=99)          # different init, then load
    dl2 = get_data_loader(mk_cfg(), 0, 1)
    ck2 = Checkpointer(save_root, 3, "fsdp", 0, 0)
    _, _, dl2, step, tokens, resuming = ck2.load(sm2, opt2, dl2, path="")
    assert step == 5 and resuming
    for u, u2 in zip(sm.all_units, sm2.all_units):
        assert torch.equal(u.master_shard, u2.master_shard), u.name
    got = [next(iter_dl2) for iter_dl2 in [iter(dl2)] for _ in range(4)]
    for (ex, ey), (gx, gy) in zip(expect, got):
        assert torch.equal(ex, gx) and torch.equal(ey, gy), \
            "resume not token-exact"


def _worker70(rank, world, port, strat, intra, q, steps=3):
    """70B-execution-path proxy: reshard_after_forward=True (transient
    flat params AND grads) + selective AC + streamed meta init, at the
    8-rank topology of the driver's single-node scaling run."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from fms_fsdp_amd.models import Llama, LlamaBlock
        from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
        from fms_fsdp_amd.parallel.policies import apply_selective_ac
        with torch.device("meta"):
            m = Llama(_small_cfg())
        apply_selective_ac(m, LlamaBlock, "1/2")
        torch.manual_seed(0)
        sm = ShardedModel(m, LlamaBlock, sharding_strategy=strat,
                          param_dtype=torch.float32,
                          reshard_after_forward=True,
                          intra_node_size=intra)
        opt = ShardedAdamW(sm, lr=1e-3)
        g = torch.Generator().manual_seed(42)
        x = torch.randint(0, 128, (2, 32), generator=g)
        y = torch.randint(0, 128, (2, 32), generator=g)
        losses = []
        for _ in range(steps):
            opt.zero_grad()
            loss = sm(x, labels=y)
            loss.backward()
            sm.clip_grad_norm_(1.0)
            opt.step()
            losses.append(loss.item())
        # transient buffers really were freed after backward
        for u in sm.units:
            assert u.flat_param.untyped_storage().size() == 0, u.name
            assert u.flat_grad.untyped_storage().size() == 0, u.name
        if rank == 0:
            q.put(("ok", losses))
    except Exception as e:
        q.put(("err", f"{type(e).__name__}: {e}"))
        raise
    finally:
        dist.destroy_process_group()


def _ref70():
    """Single-process trajectory with the same init path (meta+seed)."""
    from fms_fsdp_amd.models import Llama, LlamaBlock
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
    with torch.device("meta"):
        m = Llama(_small_cfg())
    torch.manual_seed(0)
    sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                      param_dtype=torch.float32)
    opt = ShardedAdamW(sm, lr=1e-3)
    g = torch.Generator().manual_seed(42)
    x = torch.randint(0, 128, (2, 32), generator=g)
    y = torch.randint(0, 128, (2, 32), generator=g)
    losses = []
    for _ in range(3):
        opt.zero_grad()
        loss = sm(x, labels=y)
        loss.backward()
        sm.clip_grad_norm_(1.0)
        opt.step()
        losses.append(loss.item())
    return losses


@pytest.mark.parametrize("strat,intra,port", [
    ("fsdp", None, PORT + 61),
    ("hsdp", 4, PORT + 62),   # 2 'nodes' x 4 'GPUs'
])
def test_world8_70b_execution_path(strat, intra, port):
    """World-8 gloo: meta-streamed init + reshard_after_forward +
    selective AC + (fsdp | hsdp 2x4) must match the single-process
    trajectory — the code-level readiness check for the 70B row
    (VERDICT weak #6)."""
    ref = _ref70()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.spawn(_worker70, args=(8, port, strat, intra, q), nprocs=8, join=True)
    status, got = q.get()
    assert status == "ok", got
    for a, b in zip(ref, got):
        assert abs(a - b) < 1e-5, (ref, got)


def _ckpt8_worker(rank, world, port, tmpdir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from fms_fsdp_amd.models import Llama, LlamaBlock
        from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
        from fms_fsdp_amd.utils.checkpointing import Checkpointer
        torch.manual_seed(0)
        m = Llama(_small_cfg())
        m.reset_parameters()
        sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                          param_dtype=torch.float32,
                          reshard_after_forward=True)
        opt = ShardedAdamW(sm, lr=1e-3)
        g = torch.Generator().manual_seed(42)
        x = torch.randint(0, 128, (2, 32), generator=g)
        y = torch.randint(0, 128, (2, 32), generator=g)
        for _ in range(2):
            opt.zero_grad()
            sm(x, labels=y).backward()
            sm.clip_grad_norm_(1.0)
            opt.step()
        Checkpointer(tmpdir, 3, "fsdp", rank, rank).save(2, sm, opt, None)
        if rank == 0:
            q.put(("ok", None))
    except Exception as e:
        q.put(("err", f"{type(e).__name__}: {e}"))
        raise
    finally:
        dist.destroy_process_group()


def _resume2_worker(rank, world, port, tmpdir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from fms_fsdp_amd.models import Llama, LlamaBlock
        from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
        from fms_fsdp_amd.utils.checkpointing import Checkpointer
        torch.manual_seed(77)   # different init: must be overwritten
        m = Llama(_small_cfg())
        m.reset_parameters()
        sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                          param_dtype=torch.float32)
        opt = ShardedAdamW(sm, lr=1e-3)
        ck = Checkpointer(tmpdir, 3, "fsdp", rank, rank)
        _, _, _, step, _, resuming = ck.load(sm, opt, None, path="")
        assert step == 2 and resuming
        g = torch.Generator().manual_seed(42)
        x = torch.randint(0, 128, (2, 32), generator=g)
        y = torch.randint(0, 128, (2, 32), generator=g)
        opt.zero_grad()
        loss = sm(x, labels=y)
        loss.backward()
        sm.clip_grad_norm_(1.0)
        opt.step()
        if rank == 0:
            q.put(("ok", loss.item()))
    except Exception as e:
        q.put(("err", f"{type(e).__name__}: {e}"))
        raise
    finally:
        dist.destroy_process_group()


def test_world8_save_reshard2_resume(tmp_path):
    """Save at world=8, resume at world=2 (8->2 shard resharding incl.
    optimizer moments), continue a step: loss must equal the
    uninterrupted single-process step-3 loss."""
    ref3 = _single_process_reference()[2]
    tmpdir = str(tmp_path)
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.spawn(_ckpt8_worker, args=(8, PORT + 63, tmpdir, q), nprocs=8,
             join=True)
    status, _ = q.get()
    assert status == "ok", _
    q2 = ctx.SimpleQueue()
    mp.spawn(_resume2_worker, args=(2, PORT + 64, tmpdir, q2), nprocs=2,
             join=True)
    status, loss3 = q2.get()
    assert status == "ok", loss3
    assert abs(loss3 - ref3) < 1e-5, (loss3, ref3)


def _reshard3_worker(rank, world, port, tmpdir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from fms_fsdp_amd.models import Llama, LlamaBlock
        from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
        from fms_fsdp_amd.utils.checkpointing import Checkpointer
        torch.manual_seed(0)
        m = Llama(_small_cfg())
        m.reset_parameters()
        sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                          param_dtype=torch.float32)
        opt = ShardedAdamW(sm, lr=1e-3)
        ck = Checkpointer(tmpdir, 3, "fsdp", rank, rank)
        _, _, _, step, _, resuming = ck.load(sm, opt, None, path="")
        assert step == 2 and resuming
        g = torch.Generator().manual_seed(42)
        x = torch.randint(0, 128, (2, 32), generator=g)
        y = torch.randint(0, 128, (2, 32), generator=g)
        opt.zero_grad()
        loss = sm(x, labels=y)
        loss.backward()
        sm.clip_grad_norm_(1.0)
        opt.step()
        if rank == 0:
            q.put(("ok", loss.item()))
    except Exception as e:
        q.put(("err", f"{type(e).__name__}: {e}"))
        raise
    finally:
        dist.destroy_process_group()


def test_save2_resume3_odd_world(tmp_path):
    """NON-DIVISOR resharding: save at world=2, resume at world=3 (the
    per-unit flat padding differs between shard worlds — offsets are
    S-independent so the reshard must still be exact)."""
    ref3 = _single_process_reference()[2]
    tmpdir = str(tmp_path)
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    # _ckpt8_worker trains the SAME model/recipe as the single-process
    # reference (3 layers, clipped) before saving
    mp.spawn(_ckpt8_worker, args=(2, PORT + 65, tmpdir, q), nprocs=2,
             join=True)
    status, _ = q.get()
    assert status == "ok", _
    q2 = ctx.SimpleQueue()
    mp.spawn(_reshard3_worker, args=(3, PORT + 66, tmpdir, q2), nprocs=3,
             join=True)
    status, loss3 = q2.get()
    assert status == "ok", loss3
    assert abs(loss3 - ref3) < 1e-5, (loss3, ref3)


def _hsdp_ckpt_worker(rank, world, port, tmpdir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["LOCAL_WORLD_SIZE"] = "2"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from fms_fsdp_amd.models import Llama, LlamaBlock
        from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
        from fms_fsdp_amd.utils.checkpointing import Checkpointer

        class FakeLoader:
            class DS:
                def state_dict(self):
                    return {"pos": 7}
            dataset = DS()

        torch.manual_seed(0)
        m = Llama(_small_cfg())
        m.reset_parameters()
        sm = ShardedModel(m, LlamaBlock, sharding_strategy="hsdp",
                          param_dtype=torch.float32, intra_node_size=2)
        opt = ShardedAdamW(sm, lr=1e-3)
        g = torch.Generator().manual_seed(42)
        x = torch.randint(0, 128, (2, 32), generator=g)
        y = torch.randint(0, 128, (2, 32), generator=g)
        opt.zero_grad()
        sm(x, labels=y).backward()
        opt.step()
        Checkpointer(tmpdir, 3, "hsdp", rank, rank).save(
            1, sm, opt, FakeLoader())
        if rank == 0:
            q.put(("ok", None))
    except Exception as e:
        q.put(("err", f"{type(e).__name__}: {e}"))
        raise
    finally:
        dist.destroy_process_group()


def test_hsdp_checkpoint_write_dedup(tmp_path):
    """HSDP at 2 'nodes' x 2 'GPUs': only the replicate-rank-0 shard
    group writes model/optim shards (reference checkpointing_utils.py
    :137-141), while EVERY rank writes its loader state."""
    tmpdir = str(tmp_path)
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.spawn(_hsdp_ckpt_worker, args=(4, PORT + 67, tmpdir, q), nprocs=4,
             join=True)
    status, _ = q.get()
    assert status == "ok", _
    files = sorted(os.listdir(os.path.join(tmpdir, "checkpoints",
                                           "step_1_ckp")))
    models = [f for f in files if f.startswith("model_") and
              f.endswith(".pth")]
    loaders = [f for f in files if f.startswith("loader_state_")]
    assert models == ["model_0_of_2.pth", "model_1_of_2.pth"], files
    assert len(loaders) == 4, files
    assert "metadata.pth" in files
