"""Tensor parallelism for the frozen speculator base model.

Parity target: the reference's TP path for `get_model(...,
distributed_strategy="tp")` (speculator/train_speculator.py:150-160 there):
column-parallel QKV / gate-up projections, row-parallel output
projections with an RCCL all-reduce over the tp group (2 all-reduces per
block forward, SURVEY.md §2.3 fms-TP row). The base model is frozen
(no_grad), so the all-reduces need no autograd handling.

xGMI note: with tp inside the 8-GPU clique the all-reduced activation is
(b, s, h) bf16 (~64 MB at b2 s4096 h4096) — RCCL multi-ring over the 7
point-to-point links.
"""

import torch
import torch.distributed as dist

from fms_fsdp_amd.models.llama import Attention, Llama, SwiGLU


def _slice_rows(w, rank, ws):
    n = w.shape[0] // ws
    return w[rank * n:(rank + 1) * n].clone()


def _slice_cols(w, rank, ws):
    n = w.shape[1] // ws
    return w[:, rank * n:(rank + 1) * n].clone()


def tp_shard_llama(model: Llama, tp_group) -> Llama:
    """Shard a (fully materialized) Llama across the tp group in place."""
    ws = dist.get_world_size(tp_group)
    rank = dist.get_rank(tp_group)
    if ws == 1:
        return model
    cfg = model.config
    assert cfg.nheads % ws == 0 and cfg.kvheads % ws == 0, \
        f"nheads/kvheads must divide tp size {ws}"
    assert cfg.hidden_dim % ws == 0
    hd = cfg.head_dim

    def reduce_hook(module, args, out):
        dist.all_reduce(out, group=tp_group)
        return out

    for block in model.layers:
        attn: Attention = block.attn
        nq, nkv = attn.nheads * hd, attn.kvheads * hd
        w = attn.qkv.weight.data
        qw, kw, vw = w[:nq], w[nq:nq + nkv], w[nq + nkv:]
        attn.qkv.weight.data = torch.cat(
            [_slice_rows(qw, rank, ws), _slice_rows(kw, rank, ws),
             _slice_rows(vw, rank, ws)], dim=0)
        attn.proj.weight.data = _slice_cols(attn.proj.weight.data, rank, ws)
        attn.nheads //= ws
        attn.kvheads //= ws
        attn._disable_fused_residual = True  # all-reduce rides the module hook
        attn.proj.register_forward_hook(reduce_hook)

        mlp: SwiGLU = block.mlp
        gw = mlp.wg1.weight.data[:mlp.hidden_dim]
        uw = mlp.wg1.weight.data[mlp.hidden_dim:]
        mlp.wg1.weight.data = torch.cat(
            [_slice_rows(gw, rank, ws), _slice_rows(uw, rank, ws)], dim=0)
        mlp.w2.weight.data = _slice_cols(mlp.w2.weight.data, rank, ws)
        mlp.hidden_dim //= ws
        mlp._disable_fused_residual = True
        mlp.w2.register_forward_hook(reduce_hook)
    return model
