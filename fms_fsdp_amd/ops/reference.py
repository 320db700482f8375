"""Plain-PyTorch fp32 reference implementations of every hot op.

These are (a) the CPU execution path for the gloo-backed test suite and
(b) the numerics oracle the HIP kernels are validated against
("numerics tests compare against a plain PyTorch fp32
reference of the same op").
"""

import torch
import torch.nn.functional as F


def rmsnorm(x, weight, eps):
    xf = x.float()
    y = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (y * weight.float()).to(x.dtype)


def rope_apply(q, k, cos, sin):
    """q (b,s,h,d), k (b,s,kvh,d); cos/sin (s, d/2) fp32. Half-rotation
    (HF) convention: (x1, x2) -> (x1*cos - x2*sin, x2*cos + x1*sin) with
    x1 = x[..., :d/2], x2 = x[..., d/2:]."""
    def rot(t):
        tf = t.float()
        d2 = tf.shape[-1] // 2
        t1, t2 = tf[..., :d2], tf[..., d2:]
        c = cos.view(1, -1, 1, d2)
        s = sin.view(1, -1, 1, d2)
        return torch.cat([t1 * c - t2 * s, t2 * c + t1 * s], dim=-1).to(t.dtype)
    return rot(q), rot(k)


def attention_causal(q, k, v):
    """q (b,s,h,d), k/v (b,s,kvh,d) -> (b,s,h,d). Causal, GQA broadcast."""
    b, s, h, d = q.shape
    kvh = k.shape[2]
    qt = q.transpose(1, 2).float()
    kt = k.transpose(1, 2).float()
    vt = v.transpose(1, 2).float()
    o = F.scaled_dot_product_attention(qt, kt, vt, is_causal=True,
                                       enable_gqa=(kvh != h))
    return o.transpose(1, 2).to(q.dtype)


def swiglu(gu):
    """gu (..., 2H) fused gate|up -> silu(g) * u, (..., H)."""
    g, u = gu.float().chunk(2, dim=-1)
    return (F.silu(g) * u).to(gu.dtype)


def linear_cross_entropy(x, weight, labels, ignore_index=-100):
    """x (b,s,e), weight (V,e), labels (b,s) -> mean CE over non-ignored."""
    logits = F.linear(x.float(), weight.float())
    return F.cross_entropy(logits.view(-1, logits.shape[-1]), labels.view(-1),
                           ignore_index=ignore_index)


def causal_conv1d(x, weight, bias):
    """Depthwise causal conv + silu (mamba xBC conv). x (b, l, C),
    weight (C, W), bias (C)."""
    b, l, C = x.shape
    W = weight.shape[1]
    xt = x.transpose(1, 2).float()
    y = F.conv1d(F.pad(xt, (W - 1, 0)), weight.float().unsqueeze(1),
                 bias.float(), groups=C)
    return F.silu(y).transpose(1, 2).to(x.dtype)


def adamw_step(p, g, m, v, step, lr, beta1, beta2, eps, weight_decay):
    """In-place fp32 AdamW on flat tensors (the shard update)."""
    p.mul_(1 - lr * weight_decay)
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    denom = (v / bc2).sqrt_().add_(eps)
    p.addcdiv_(m, denom, value=-lr / bc1)
