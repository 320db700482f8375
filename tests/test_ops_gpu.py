"""GPU numerics tests: every HIP kernel vs the plain-PyTorch fp32
reference of the same op (SURVEY.md §4 test strategy). All @pytest.mark.gpu."""

import pytest
import torch

from fms_fsdp_amd.ops import reference

pytestmark = pytest.mark.gpu


def dev():
    return torch.device("cuda:0")


def relerr(a, b):
    a, b = a.float(), b.float()
    return ((a - b).abs().max() / b.abs().max().clamp(min=1e-6)).item()


@pytest.mark.parametrize("rows,H", [(128, 256), (512, 4096), (64, 5120)])
def test_rmsnorm_fwd_bwd(rows, H):
    torch.manual_seed(0)
    from fms_fsdp_amd import _C
    x = torch.randn(rows, H, device=dev(), dtype=torch.bfloat16)
    w = torch.randn(H, device=dev(), dtype=torch.bfloat16)
    y, rinv = _C.rmsnorm_fwd(x, w, 1e-6)
    ref = reference.rmsnorm(x, w, 1e-6)
    assert relerr(y, ref) < 2e-2

    # backward vs autograd of fp32 reference
    xf = x.float().requires_grad_()
    wf = w.float().requires_grad_()
    yf = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-6) * wf
    dy = torch.randn_like(yf)
    yf.backward(dy)
    dx, dw = _C.rmsnorm_bwd(dy.bfloat16(), x, w, rinv, None)
    assert relerr(dx, xf.grad) < 5e-2
    assert relerr(dw, wf.grad) < 5e-2


def test_rope():
    torch.manual_seed(1)
    from fms_fsdp_amd import _C
    b, s, h, kvh, d = 2, 128, 8, 4, 128
    q = torch.randn(b, s, h, d, device=dev(), dtype=torch.bfloat16)
    k = torch.randn(b, s, kvh, d, device=dev(), dtype=torch.bfloat16)
    t = torch.arange(s, dtype=torch.float32)
    inv = 1.0 / (10000.0 ** (torch.arange(0, d, 2).float() / d))
    fr = torch.outer(t, inv)
    cos, sin = fr.cos().to(dev()), fr.sin().to(dev())
    qo, ko = _C.rope_fwd(q, k, cos, sin, False)
    qr, kr = reference.rope_apply(q, k, cos, sin)
    assert relerr(qo, qr) < 2e-2
    assert relerr(ko, kr) < 2e-2
    # conj rotation is the exact adjoint/inverse
    qb, kb = _C.rope_fwd(qo, ko, cos, sin, True)
    assert relerr(qb, q) < 3e-2


def test_swiglu():
    torch.manual_seed(2)
    from fms_fsdp_amd import _C
    gu = torch.randn(512, 2048, device=dev(), dtype=torch.bfloat16)
    h = _C.swiglu_fwd(gu)
    ref = reference.swiglu(gu)
    assert relerr(h, ref) < 2e-2
    guf = gu.float().requires_grad_()
    g, u = guf.chunk(2, -1)
    out = torch.nn.functional.silu(g) * u
    dy = torch.randn_like(out)
    out.backward(dy)
    dgu = _C.swiglu_bwd(dy.bfloat16(), gu)
    assert relerr(dgu, guf.grad) < 5e-2


@pytest.mark.parametrize("V", [1024, 32000, 128256])
def test_cross_entropy(V):
    torch.manual_seed(3)
    from fms_fsdp_amd import _C
    n = 256
    logits = (torch.randn(n, V, device=dev()) * 3).bfloat16()
    labels = torch.randint(0, V, (n,), device=dev())
    labels[:17] = -100
    count = (labels != -100).sum().clamp(min=1).float()
    lf = logits.float().requires_grad_()
    ref = torch.nn.functional.cross_entropy(lf, labels, ignore_index=-100)
    ref.backward()
    loss_sum = torch.zeros((), device=dev())
    work = logits.clone()
    _C.ce_fwd_bwd(work, labels, loss_sum, count, -100)
    loss = loss_sum / count
    assert abs(loss.item() - ref.item()) / ref.item() < 1e-2
    assert relerr(work, lf.grad) < 5e-2


def test_adamw_matches_torch():
    torch.manual_seed(4)
    from fms_fsdp_amd import _C
    n = 4096
    p0 = torch.randn(n, device=dev())
    g = torch.randn(n, device=dev())
    p_ref = torch.nn.Parameter(p0.clone())
    opt = torch.optim.AdamW([p_ref], lr=1e-3, betas=(0.9, 0.95), eps=1e-8,
                            weight_decay=0.1)
    p = p0.clone()
    m = torch.zeros(n, device=dev())
    v = torch.zeros(n, device=dev())
    for step in range(1, 4):
        p_ref.grad = g.clone()
        opt.step()
        _C.adamw(p, g, m, v, float(step), 1e-3, 0.9, 0.95, 1e-8, 0.1, None, None)
    assert relerr(p, p_ref.detach()) < 1e-4


def test_sq_norm():
    from fms_fsdp_amd import _C
    t = torch.randn(10000, device=dev())
    out = torch.zeros((), device=dev())
    _C.sq_norm_accum(t[:9996], out)  # multiple of 4
    assert abs(out.item() - t[:9996].pow(2).sum().item()) < 1.0


@pytest.mark.parametrize("b,s,h,kvh,d", [
    (1, 128, 2, 2, 128),
    (2, 256, 4, 2, 128),
    (1, 512, 8, 1, 64),
])
def test_attention_fwd(b, s, h, kvh, d):
    torch.manual_seed(5)
    from fms_fsdp_amd import _C
    q = torch.randn(b, s, h, d, device=dev(), dtype=torch.bfloat16)
    k = torch.randn(b, s, kvh, d, device=dev(), dtype=torch.bfloat16)
    v = torch.randn(b, s, kvh, d, device=dev(), dtype=torch.bfloat16)
    o, lse = _C.attn_fwd(q, k, v)
    ref = reference.attention_causal(q, k, v)
    err = relerr(o, ref)
    assert err < 3e-2, err
    # lse check vs manual fp32
    qt = q.transpose(1, 2).float()
    kt = k.transpose(1, 2).float()
    if kvh != h:
        kt = kt.repeat_interleave(h // kvh, dim=1)
    scores = qt @ kt.transpose(-1, -2) / (d ** 0.5)
    mask = torch.full((s, s), float("-inf"), device=dev()).triu(1)
    lse_ref = (scores + mask).logsumexp(-1)
    assert relerr(lse, lse_ref) < 3e-2


@pytest.mark.parametrize("b,s,h,kvh,d", [
    (1, 128, 2, 2, 128),
    (2, 256, 4, 2, 128),
    (1, 256, 4, 4, 64),
    (2, 1024, 12, 12, 64),
])
def test_attention_bwd(b, s, h, kvh, d):
    torch.manual_seed(6)
    from fms_fsdp_amd import _C
    q = torch.randn(b, s, h, d, device=dev(), dtype=torch.bfloat16)
    k = torch.randn(b, s, kvh, d, device=dev(), dtype=torch.bfloat16)
    v = torch.randn(b, s, kvh, d, device=dev(), dtype=torch.bfloat16)
    do = torch.randn(b, s, h, d, device=dev(), dtype=torch.bfloat16)

    qf = q.float().requires_grad_()
    kf = k.float().requires_grad_()
    vf = v.float().requires_grad_()
    of = torch.nn.functional.scaled_dot_product_attention(
        qf.transpose(1, 2), kf.transpose(1, 2), vf.transpose(1, 2),
        is_causal=True, enable_gqa=(kvh != h)).transpose(1, 2)
    of.backward(do.float())

    o, lse = _C.attn_fwd(q, k, v)
    dq, dk, dv = _C.attn_bwd(do, q, k, v, o, lse, None)
    assert relerr(dq, qf.grad) < 6e-2
    assert relerr(dk, kf.grad) < 6e-2
    assert relerr(dv, vf.grad) < 6e-2


def test_model_gpu_step():
    """End-to-end: tiny llama fwd+bwd+step on GPU through the HIP path."""
    from fms_fsdp_amd.models import Llama, LlamaBlock, LlamaConfig
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
    torch.manual_seed(0)
    cfg = LlamaConfig(src_vocab_size=512, emb_dim=256, nheads=2, kvheads=2,
                      nlayers=2, max_expected_seq_len=256)
    with torch.device(dev()):
        m = Llama(cfg)
        m.reset_parameters()
    sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                      param_dtype=torch.bfloat16)
    opt = ShardedAdamW(sm, lr=1e-3)
    x = torch.randint(0, 512, (2, 256), device=dev())
    y = torch.randint(0, 512, (2, 256), device=dev())
    losses = []
    for _ in range(5):
        opt.zero_grad()
        loss = sm(x, labels=y)
        loss.backward()
        sm.clip_grad_norm_(1.0)
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses
    assert all(l == l for l in losses), losses


def test_model_gpu_step_selective_ac():
    """Selective AC recompute + direct-wgrad (linear_flat) + the fused
    qkv path together on GPU: AC re-runs the block forward during
    backward; each weight's wgrad must still fire exactly once."""
    from fms_fsdp_amd.models import Llama, LlamaBlock, LlamaConfig
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
    from fms_fsdp_amd.parallel.policies import apply_selective_ac
    torch.manual_seed(0)
    cfg = LlamaConfig(src_vocab_size=512, emb_dim=256, nheads=2, kvheads=2,
                      nlayers=4, max_expected_seq_len=256)
    with torch.device(dev()):
        m = Llama(cfg)
        m.reset_parameters()
    apply_selective_ac(m, LlamaBlock, 0.5)
    sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                      param_dtype=torch.bfloat16)
    opt = ShardedAdamW(sm, lr=1e-3)
    x = torch.randint(0, 512, (2, 256), device=dev())
    y = torch.randint(0, 512, (2, 256), device=dev())
    losses = []
    for _ in range(5):
        opt.zero_grad()
        loss = sm(x, labels=y)
        loss.backward()
        sm.clip_grad_norm_(1.0)
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses
    assert all(l == l for l in losses), losses


def test_speculator_base_models_gpu():
    """GPT-BigCode (MQA kvh=1) and Mixtral (dense top-2 MoE) forward +
    backward through the HIP kernel paths on GPU (speculator base-model
    families; CPU parity tests cover the math, this pins the GPU path)."""
    from fms_fsdp_amd.config import get_model_config
    from fms_fsdp_amd.models.gpt_bigcode import GPTBigCode
    from fms_fsdp_amd.models.mixtral import Mixtral
    torch.manual_seed(5)
    for ctor, name in ((GPTBigCode, "gpt_bigcode_test"),
                       (Mixtral, "mixtral_test")):
        with torch.device(dev()):
            m = ctor(get_model_config(name))
            m.reset_parameters()
        m = m.bfloat16()
        x = torch.randint(0, 256, (2, 128), device=dev())
        loss = m(x, labels=torch.randint(0, 256, (2, 128), device=dev()))
        loss.float().backward()
        assert torch.isfinite(loss), name
        for p in m.parameters():
            assert p.grad is None or torch.isfinite(p.grad).all(), name


def test_causal_conv1d():
    torch.manual_seed(7)
    from fms_fsdp_amd import _C
    b, l, C, W = 2, 64, 256, 4
    x = torch.randn(b, l, C, device=dev(), dtype=torch.bfloat16)
    w = torch.randn(C, W, device=dev(), dtype=torch.bfloat16)
    bias = torch.randn(C, device=dev())
    y = _C.cconv_fwd(x, w, bias)
    ref = reference.causal_conv1d(x, w.float(), bias)
    assert relerr(y, ref) < 2e-2

    xf = x.float().requires_grad_()
    wf = w.float().requires_grad_()
    bfp = bias.clone().requires_grad_()
    yr = reference.causal_conv1d(xf, wf, bfp)
    dy = torch.randn(b, l, C, device=dev())  # contiguous (b,l,C)
    yr.backward(dy)
    dx, dw, db = _C.cconv_bwd(dy.bfloat16().contiguous(), x, w, bias)
    assert relerr(dx, xf.grad) < 5e-2
    assert relerr(dw, wf.grad) < 5e-2
    assert relerr(db, bfp.grad) < 5e-2


def test_add_rmsnorm():
    torch.manual_seed(8)
    from fms_fsdp_amd import _C
    rows, H = 256, 4096
    x = torch.randn(rows, H, device=dev(), dtype=torch.bfloat16)
    r = torch.randn(rows, H, device=dev(), dtype=torch.bfloat16)
    w = torch.randn(H, device=dev(), dtype=torch.bfloat16)
    y, s, rinv = _C.add_rmsnorm_fwd(x, r, w, 1e-6)
    sref = (x.float() + r.float()).bfloat16()
    yref = reference.rmsnorm(sref, w, 1e-6)
    assert relerr(s, sref) < 1e-2
    assert relerr(y, yref) < 2e-2
    # fused-dextra backward == plain backward + add
    dy = torch.randn(rows, H, device=dev(), dtype=torch.bfloat16)
    ds = torch.randn(rows, H, device=dev(), dtype=torch.bfloat16)
    dx1, dw1 = _C.rmsnorm_bwd(dy, s, w, rinv, ds)
    dx0, dw0 = _C.rmsnorm_bwd(dy, s, w, rinv, None)
    assert relerr(dx1, (dx0.float() + ds.float())) < 2e-2
    assert relerr(dw1, dw0) < 1e-4


def test_mamba_gpu_step():
    """Hybrid mamba (SSD + attn + conv kernels) fwd+bwd+step on GPU."""
    from fms_fsdp_amd.config import get_model_config
    from fms_fsdp_amd.models.mamba import (MambaBlock, MambaConfig,
                                           MambaLMHeadModel)
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
    torch.manual_seed(0)
    mc = MambaConfig.from_dict(get_model_config("mamba_test"))
    with torch.device(dev()):
        m = MambaLMHeadModel(mc)
        m.reset_parameters()
    sm = ShardedModel(m, MambaBlock, sharding_strategy="fsdp",
                      param_dtype=torch.bfloat16)
    opt = ShardedAdamW(sm, lr=1e-3)
    x = torch.randint(0, mc.vocab_size, (2, 128), device=dev())
    y = torch.randint(0, mc.vocab_size, (2, 128), device=dev())
    losses = []
    for _ in range(4):
        opt.zero_grad()
        loss = sm(x, labels=y)
        loss.backward()
        sm.clip_grad_norm_(1.0)
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] and all(l == l for l in losses), losses


def test_checkpoint_roundtrip_gpu(tmp_path):
    """Save + reload a trained sharded model on GPU; training state must
    be bit-identical (master shards, moments) and training must continue."""
    from fms_fsdp_amd.models import Llama, LlamaBlock, LlamaConfig
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
    from fms_fsdp_amd.utils.checkpointing import Checkpointer
    torch.manual_seed(0)
    cfg = LlamaConfig(src_vocab_size=512, emb_dim=256, nheads=2, kvheads=2,
                      nlayers=2, max_expected_seq_len=256)

    def build():
        with torch.device(dev()):
            m = Llama(cfg)
            m.reset_parameters()
        sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                          param_dtype=torch.bfloat16)
        return sm, ShardedAdamW(sm, lr=1e-3)

    sm, opt = build()
    x = torch.randint(0, 512, (2, 256), device=dev())
    y = torch.randint(0, 512, (2, 256), device=dev())
    for _ in range(2):
        opt.zero_grad()
        sm(x, labels=y).backward()
        sm.clip_grad_norm_(1.0)
        opt.step()
    ck = Checkpointer(str(tmp_path), 3, "fsdp", 0, 0)
    ck.save(2, sm, opt, None, tokens_seen=99)

    sm2, opt2 = build()
    ck2 = Checkpointer(str(tmp_path), 3, "fsdp", 0, 0)
    _, _, _, step, tokens, resuming = ck2.load(sm2, opt2, None, path="")
    assert step == 2 and tokens == 99 and resuming
    for u1, u2 in zip(sm.all_units, sm2.all_units):
        assert torch.equal(u1.master_shard, u2.master_shard), u1.name
        assert torch.equal(u1.exp_avg, u2.exp_avg), u1.name
    # continues training identically
    opt.zero_grad(); l1 = sm(x, labels=y); l1.backward(); opt.step()
    opt2.zero_grad(); l2 = sm2(x, labels=y); l2.backward(); opt2.step()
    assert abs(l1.item() - l2.item()) < 1e-4


def test_speculator_stage2_gpu(tmp_path):
    """Stage-2 speculator loss (KV-cache generation) on GPU in bf16."""
    from fms_fsdp_amd.config import train_config
    from fms_fsdp_amd.models import Llama, LlamaConfig
    from fms_fsdp_amd.models.speculator import MLPSpeculator
    from speculator.train_speculator_utils import stage2_loss
    torch.manual_seed(1)
    mcfg = LlamaConfig(src_vocab_size=512, emb_dim=256, nheads=2, kvheads=2,
                       nlayers=2, max_expected_seq_len=512)
    with torch.device(dev()):
        m = Llama(mcfg)
        m.reset_parameters()
    m = m.bfloat16().eval()
    spec = MLPSpeculator(256, 128, 512, 2).to(dev()).bfloat16()
    spec.reset_parameters()
    cfg = train_config()
    cfg.batch_size = 1
    cfg.seq_length = 256
    cfg.stage2_batch_size = 4
    cfg.stage2_prompt_length = 32
    cfg.stage2_seq_length = 64
    inp = torch.randint(0, 512, (1, 256), device=dev())
    stats = torch.zeros(2 + 2, device=dev())
    loss, stats, ntok = stage2_loss(cfg, m, spec, inp, inp, stats)
    assert torch.isfinite(loss)
    loss.backward()


def test_segsum_exp():
    torch.manual_seed(9)
    from fms_fsdp_amd import _C
    from fms_fsdp_amd.ops import reference  # noqa
    N, Q = 6, 128
    cs = (-torch.rand(N, Q, device=dev()).cumsum(-1)).contiguous()
    L = _C.segsum_exp_fwd(cs)
    mask = torch.tril(torch.ones(Q, Q, dtype=torch.bool, device=dev()), 0)
    ref = torch.exp((cs[:, :, None] - cs[:, None, :]).masked_fill(~mask, -torch.inf))
    assert relerr(L, ref) < 1e-2
    # backward vs autograd
    csf = cs.clone().requires_grad_()
    Lf = torch.exp((csf[:, :, None] - csf[:, None, :]).masked_fill(~mask, -torch.inf))
    g = torch.randn_like(Lf)
    Lf.backward(g)
    dcs = _C.segsum_exp_bwd(g.bfloat16().contiguous(), cs)
    assert relerr(dcs, csf.grad) < 5e-2


@pytest.mark.parametrize("b,s,h,kvh,d", [
    (1, 256, 4, 4, 128),
    (2, 256, 8, 2, 128),
    (1, 256, 4, 4, 64),
])
def test_qkv_rope_attention_fused(b, s, h, kvh, d):
    """Fused split+RoPE+attention (strided qkv-slice kernels, fused dqkv
    backward buffer) must match the modular split -> rope -> attention
    composition, values and grads."""
    from fms_fsdp_amd import ops
    torch.manual_seed(12)
    E = (h + 2 * kvh) * d
    qkv = torch.randn(b, s, E, device=dev(), dtype=torch.bfloat16)
    pos = torch.arange(s, dtype=torch.float32)
    inv = 1.0 / (10000.0 ** (torch.arange(0, d, 2,
                                          dtype=torch.float32) / d))
    ang = torch.outer(pos, inv)
    cos, sin = ang.cos().to(dev()), ang.sin().to(dev())

    a = qkv.clone().requires_grad_()
    o_f = ops.qkv_rope_attention(a, cos, sin, h, kvh, d)
    do = torch.randn_like(o_f)
    o_f.backward(do)

    b2 = qkv.clone().requires_grad_()
    q, k, v = b2.split([h * d, kvh * d, kvh * d], dim=-1)
    q = q.view(b, s, h, d)
    k = k.view(b, s, kvh, d)
    v = v.view(b, s, kvh, d)
    qr, kr = ops.rope_apply(q, k, cos, sin)
    o_m = ops.attention_causal(qr, kr, v)
    o_m.backward(do)

    assert relerr(o_f, o_m) < 1e-2
    assert relerr(a.grad, b2.grad) < 2e-2


def test_attention_padded_seq():
    """Arbitrary (non-128-multiple) seq lens via end-padding dispatch."""
    from fms_fsdp_amd import ops
    torch.manual_seed(11)
    b, s, h, kvh, d = 2, 200, 2, 2, 128
    q = torch.randn(b, s, h, d, device=dev(), dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(b, s, kvh, d, device=dev(), dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(b, s, kvh, d, device=dev(), dtype=torch.bfloat16,
                    requires_grad=True)
    o = ops.attention_causal(q, k, v)
    ref = reference.attention_causal(q.detach(), k.detach(), v.detach())
    assert relerr(o, ref) < 3e-2
    o.sum().backward()
    assert torch.isfinite(q.grad).all() and torch.isfinite(k.grad).all()


@pytest.mark.parametrize("policy", ["fp16", "bf16_working"])
def test_model_gpu_step_mp_policies(policy):
    """fp16 (with dynamic loss scaling) and bf16_working policies on GPU:
    loss decreases, fused AdamW handles the policy's grad/publish dtypes
    (reference mixed_precision.py:5-27)."""
    from fms_fsdp_amd.config import train_config
    from fms_fsdp_amd.models import Llama, LlamaBlock, LlamaConfig
    from fms_fsdp_amd.parallel import (DynamicGradScaler, ShardedAdamW,
                                       ShardedModel)
    from fms_fsdp_amd.parallel.policies import (get_mixed_precision_dtypes,
                                                needs_loss_scaling)
    torch.manual_seed(0)
    cfg = LlamaConfig(src_vocab_size=512, emb_dim=256, nheads=2, kvheads=2,
                      nlayers=2, max_expected_seq_len=256)
    with torch.device(dev()):
        m = Llama(cfg)
        m.reset_parameters()
    tc = train_config()
    tc.mp_policy = policy
    pd, rd = get_mixed_precision_dtypes(tc)
    sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                      param_dtype=pd, reduce_dtype=rd)
    opt = ShardedAdamW(sm, lr=1e-3)
    scaler = DynamicGradScaler(enabled=needs_loss_scaling(tc))
    x = torch.randint(0, 512, (2, 256), device=dev())
    y = torch.randint(0, 512, (2, 256), device=dev())
    losses = []
    for _ in range(5):
        opt.zero_grad()
        loss = sm(x, labels=y)
        scaler.scale_loss(loss).backward()
        scaler.clip_and_step(sm, opt, 1.0)
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses
    assert all(l == l for l in losses), losses


def test_ssd_fused_scan_matches_torch():
    """Fused SSD path (ssd_prep/ssd_xdt/ssd_scores_decay/ssd_ygate HIP
    kernels) vs the plain-torch fp32 ssd_chunked on the same mixer:
    outputs and every gradient must agree."""
    from fms_fsdp_amd.models.mamba import Mamba2Mixer, MambaConfig
    torch.manual_seed(0)
    cfg = MambaConfig(d_model=256, n_layer=1, vocab_size=512, d_state=64,
                      headdim=64, expand=2, ngroups=2, chunk_size=128)
    mixer = Mamba2Mixer(cfg, 0)
    mixer.reset_parameters()
    mixer = mixer.to(dev()).bfloat16()
    u = (torch.randn(2, 256, 256, device=dev(), dtype=torch.bfloat16) * 0.5)

    def run(force_torch):
        mixer._force_torch_scan = force_torch
        for p in mixer.parameters():
            p.grad = None
        ui = u.clone().requires_grad_()
        out = mixer(ui)
        out.float().pow(2).mean().backward()
        return (out.detach().clone(), ui.grad.clone(),
                {n: p.grad.clone() for n, p in mixer.named_parameters()
                 if p.grad is not None})

    o_f, gu_f, gp_f = run(False)
    o_t, gu_t, gp_t = run(True)
    assert relerr(o_f, o_t) < 3e-2, relerr(o_f, o_t)
    assert relerr(gu_f, gu_t) < 6e-2, relerr(gu_f, gu_t)
    for n in gp_t:
        assert relerr(gp_f[n], gp_t[n]) < 8e-2, (n, relerr(gp_f[n], gp_t[n]))


def test_ssd_fused_scan_gqa_heads():
    """ngroups=1 (the registry configs): 4 heads share one group's
    scores — exercises the group-broadcast + d_scores group reduction."""
    from fms_fsdp_amd.models.mamba import Mamba2Mixer, MambaConfig
    torch.manual_seed(1)
    cfg = MambaConfig(d_model=128, n_layer=1, vocab_size=512, d_state=32,
                      headdim=64, expand=2, ngroups=1, chunk_size=128)
    mixer = Mamba2Mixer(cfg, 0).to(dev())
    mixer.reset_parameters()
    mixer = mixer.bfloat16()
    u = (torch.randn(1, 384, 128, device=dev(), dtype=torch.bfloat16) * 0.5)

    def run(force_torch):
        mixer._force_torch_scan = force_torch
        for p in mixer.parameters():
            p.grad = None
        ui = u.clone().requires_grad_()
        out = mixer(ui)
        out.float().pow(2).mean().backward()
        return out.detach().clone(), ui.grad.clone()

    o_f, gu_f = run(False)
    o_t, gu_t = run(True)
    assert relerr(o_f, o_t) < 3e-2
    assert relerr(gu_f, gu_t) < 6e-2


def test_gemm_nt_exact():
    """Hand-written dgrad-layout GEMM (documented experiment): must be
    numerically exact vs hipBLASLt (same fp32 accumulation order class)."""
    from fms_fsdp_amd import _C
    torch.manual_seed(0)
    a = torch.randn(512, 256, device=dev(), dtype=torch.bfloat16) * 0.1
    b = torch.randn(256, 512, device=dev(), dtype=torch.bfloat16) * 0.1
    got = _C.gemm_nt(a, b)
    ref = torch.mm(a, b)
    assert relerr(got, ref) < 1e-2


def test_speculator_gpu_smoke(tmp_path):
    """Speculator trainer end-to-end on GPU: stage-1 AND stage-2 (the
    KV-cached generate path) on the HIP kernels."""
    from speculator import train_speculator as ts
    ts.main(model_variant="llama2_125m", use_dummy_dataset=True,
            batch_size=1, seq_length=256, num_steps=2, report_interval=1,
            checkpoint_interval=100, n_speculator_heads=2,
            speculator_width=64, stage2_start_step=1, stage2_batch_size=4,
            stage2_prompt_length=16, stage2_seq_length=32,
            model_path="/nonexistent", ckpt_save_path=str(tmp_path),
            ckpt_load_path=str(tmp_path), vocab_size=256,
            learning_rate=1e-4, sharding_strategy="fsdp")


def test_main_llama_entry_gpu(tmp_path):
    """The llama ENTRY POINT end-to-end on GPU (meta init path, selective
    AC, checkpoint save) — entry-only wiring bugs don't hide behind the
    runtime tests."""
    import main_training_llama
    main_training_llama.main(
        model_variant="llama2_125m", use_dummy_dataset=True, batch_size=1,
        seq_length=256, num_steps=3, report_interval=1,
        checkpoint_interval=2, low_cpu_fsdp=True,
        fsdp_activation_checkpointing=True, selective_checkpointing="1/2",
        ckpt_save_path=str(tmp_path), ckpt_load_path=str(tmp_path),
        vocab_size=512, learning_rate=1e-4, sharding_strategy="fsdp")
    import os
    assert os.path.exists(tmp_path / "checkpoints" / "step_2_ckp"
                          / "metadata.pth")


def test_main_mamba_entry_gpu(tmp_path):
    """Mamba entry on GPU: fused SSD path through the real entry."""
    import main_training_mamba
    main_training_mamba.main(
        model_variant="mamba_test", use_dummy_dataset=True, batch_size=1,
        seq_length=256, num_steps=2, report_interval=1,
        checkpoint_interval=100, ckpt_save_path=str(tmp_path),
        ckpt_load_path=str(tmp_path), vocab_size=512, learning_rate=1e-4,
        sharding_strategy="fsdp")
