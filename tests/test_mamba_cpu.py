import pytest
import torch

from fms_fsdp_amd.models.mamba import (MambaBlock, MambaConfig,
                                       MambaLMHeadModel, ssd_chunked)


def naive_ssd(x, dt, A, B, C):
    """Sequential state-space recurrence (gold reference):
    S_t = exp(dt_t A) S_{t-1} + dt_t B_t (x) x_t ;  y_t = C_t . S_t"""
    b, l, h, p = x.shape
    g, n = B.shape[2], B.shape[3]
    rep = h // g
    Bh = B.repeat_interleave(rep, dim=2)
    Ch = C.repeat_interleave(rep, dim=2)
    S = torch.zeros(b, h, n, p, dtype=torch.float64)
    ys = []
    for t in range(l):
        decay = torch.exp(dt[:, t] * A).to(torch.float64)  # (b,h)
        S = decay[:, :, None, None] * S + torch.einsum(
            "bhn,bhp->bhnp", Bh[:, t].double() * dt[:, t, :, None].double(),
            x[:, t].double())
        ys.append(torch.einsum("bhn,bhnp->bhp", Ch[:, t].double(), S))
    return torch.stack(ys, dim=1).float()


@pytest.mark.parametrize("l,chunk", [(64, 16), (128, 32), (96, 96)])
def test_ssd_chunked_matches_recurrence(l, chunk):
    torch.manual_seed(0)
    b, h, p, g, n = 2, 4, 8, 2, 16
    x = torch.randn(b, l, h, p)
    dt = torch.rand(b, l, h) * 0.5
    A = -torch.rand(h) * 2
    B = torch.randn(b, l, g, n)
    C = torch.randn(b, l, g, n)
    y = ssd_chunked(x, dt, A, B, C, chunk)
    ref = naive_ssd(x, dt, A, B, C)
    err = (y - ref).abs().max() / ref.abs().max()
    assert err < 1e-4, err.item()


def test_ssd_backward_flows():
    torch.manual_seed(1)
    b, l, h, p, g, n = 1, 32, 2, 4, 1, 8
    x = torch.randn(b, l, h, p, requires_grad=True)
    dt = torch.rand(b, l, h, requires_grad=True)
    A = (-torch.rand(h)).clone().detach().requires_grad_()
    B = torch.randn(b, l, g, n, requires_grad=True)
    C = torch.randn(b, l, g, n, requires_grad=True)
    y = ssd_chunked(x, dt, A, B, C, 16)
    y.sum().backward()
    for t in (x, dt, A, B, C):
        assert t.grad is not None and torch.isfinite(t.grad).all()


def tiny_cfg(**kw):
    d = dict(d_model=64, n_layer=2, vocab_size=96, d_state=16, headdim=16,
             chunk_size=32, attn_layer_idx=[1],
             attn_cfg={"num_heads": 1, "num_heads_kv": 1, "head_dim": 64,
                       "rotary_emb_dim": 16}, d_intermediate=128)
    d.update(kw)
    return MambaConfig(**d)


def test_mamba_model_forward_backward():
    torch.manual_seed(0)
    m = MambaLMHeadModel(tiny_cfg())
    m.reset_parameters()
    x = torch.randint(0, 96, (2, 32))
    y = torch.randint(0, 96, (2, 32))
    loss = m(x, labels=y)
    assert torch.isfinite(loss)
    loss.backward()
    for n_, p_ in m.named_parameters():
        assert p_.grad is not None and torch.isfinite(p_.grad).all(), n_


def test_mamba_vocab_padding():
    cfg = MambaConfig(d_model=32, n_layer=1, vocab_size=50277, headdim=16,
                      d_state=16)
    assert cfg.vocab_size % 16 == 0


def test_mamba_sharded_training_step():
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
    torch.manual_seed(0)
    m = MambaLMHeadModel(tiny_cfg())
    m.reset_parameters()
    sm = ShardedModel(m, MambaBlock, sharding_strategy="fsdp",
                      param_dtype=torch.float32)
    opt = ShardedAdamW(sm, lr=1e-3)
    x = torch.randint(0, 96, (2, 32))
    y = torch.randint(0, 96, (2, 32))
    losses = []
    for _ in range(4):
        opt.zero_grad()
        loss = sm(x, labels=y)
        loss.backward()
        sm.clip_grad_norm_(1.0)
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses


def test_mamba_registry_config():
    from fms_fsdp_amd.config import get_model_config
    mcfg = MambaConfig.from_dict(get_model_config("mamba_9.8b"))
    assert mcfg.d_model == 4096 and mcfg.n_layer == 32
    assert mcfg.attn_layer_idx == [9, 18, 27]
    assert mcfg.vocab_size % 16 == 0
