import pytest
import torch

from fms_fsdp_amd.models import Llama, LlamaBlock, LlamaConfig
from fms_fsdp_amd.parallel.policies import apply_selective_ac


def tiny_cfg(**kw):
    d = dict(src_vocab_size=64, emb_dim=32, nheads=4, kvheads=2, nlayers=3,
             max_expected_seq_len=64)
    d.update(kw)
    return LlamaConfig(**d)


def test_forward_shapes():
    m = Llama(tiny_cfg())
    m.reset_parameters()
    x = torch.randint(0, 64, (2, 16))
    logits = m(x)
    assert logits.shape == (2, 16, 64)
    loss = m(x, labels=torch.randint(0, 64, (2, 16)))
    assert loss.ndim == 0 and torch.isfinite(loss)


def test_loss_matches_eager_ce():
    """linear_cross_entropy == F.cross_entropy over full logits."""
    torch.manual_seed(0)
    m = Llama(tiny_cfg())
    m.reset_parameters()
    x = torch.randint(0, 64, (2, 16))
    y = torch.randint(0, 64, (2, 16))
    y[0, :3] = -100
    loss = m(x, labels=y)
    logits = m(x)
    ref = torch.nn.functional.cross_entropy(
        logits.view(-1, 64).float(), y.view(-1), ignore_index=-100)
    assert torch.allclose(loss, ref, atol=1e-5), (loss, ref)


def test_gqa_broadcast():
    m = Llama(tiny_cfg(kvheads=1))
    m.reset_parameters()
    out = m(torch.randint(0, 64, (1, 8)))
    assert out.shape == (1, 8, 64)


def test_causality():
    """Changing a future token must not change past logits."""
    torch.manual_seed(1)
    m = Llama(tiny_cfg())
    m.reset_parameters()
    m.eval()
    x = torch.randint(0, 64, (1, 16))
    l1 = m(x)
    x2 = x.clone()
    x2[0, -1] = (x2[0, -1] + 1) % 64
    l2 = m(x2)
    assert torch.allclose(l1[0, :-1], l2[0, :-1], atol=1e-5)
    assert not torch.allclose(l1[0, -1], l2[0, -1], atol=1e-5)


def test_backward_produces_grads():
    m = Llama(tiny_cfg())
    m.reset_parameters()
    loss = m(torch.randint(0, 64, (2, 8)), labels=torch.randint(0, 64, (2, 8)))
    loss.backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n


# ---- selective AC: exact placement parity with the reference algorithm
# (reference tests/test_selective_ac.py:12-64 patterns) ----

@pytest.mark.parametrize("p,expected", [
    (0, [False] * 15),
    (1 / 100, [False] * 15),
    (1 / 5, [False, False, True, False, False] * 3),
    (1 / 3, [False, True, False] * 5),
    (1 / 2, [True, False] * 7 + [True]),
    (3 / 5, [True, False, True, False, True] * 3),
    (2 / 3, [True, False, True] * 5),
    (1, [True] * 15),
    (5 / 3, [True] * 15),
    ("1/3", [False, True, False] * 5),
])
def test_selective_ac_pattern(p, expected):
    cfg = tiny_cfg(nlayers=15)
    m = Llama(cfg)
    pattern = apply_selective_ac(m, LlamaBlock, p)
    assert pattern == expected


def test_ac_numerics_match():
    """AC on all blocks gives identical loss+grads to no AC."""
    torch.manual_seed(0)
    m = Llama(tiny_cfg())
    m.reset_parameters()
    x = torch.randint(0, 64, (2, 16))
    y = torch.randint(0, 64, (2, 16))
    loss1 = m(x, labels=y)
    loss1.backward()
    g1 = {n: p.grad.clone() for n, p in m.named_parameters()}
    m.zero_grad()
    apply_selective_ac(m, LlamaBlock, 1)
    loss2 = m(x, labels=y)
    loss2.backward()
    assert torch.allclose(loss1, loss2, atol=1e-6)
    for n, p in m.named_parameters():
        assert torch.allclose(g1[n], p.grad, atol=1e-6), n
