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
