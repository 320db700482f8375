"""Op dispatch: hand-written CDNA4 HIP kernels on GPU, fp32 PyTorch
reference on CPU.

On a ROCm GPU the in-tree extension `fms_fsdp_amd._C` (built by
`__graft_entry__.build()` / `setup.py build_ext --inplace`) is REQUIRED:
ops raise RuntimeError rather than silently falling back to eager
(per-project rule: no silent PyTorch fallback on the GPU path).
"""

import os

import torch

from . import reference

# fp16 policy note (reference fpSixteen, mixed_precision.py:5-9): fp16 is
# supported as the STORAGE/COMM dtype; op-internal compute bridges through
# bf16 (identical MFMA rate on CDNA4, wider exponent, fp32 accumulation
# inside every kernel) so the HIP kernels stay single-dtype. The casts are
# autograd-tracked, so gradients flow back to the fp16 tensors.
#
# fp32 activations on GPU (bf16_working / fp32 policies) take the fp32
# reference implementations by DESIGN: "working precision fp32" means the
# math genuinely runs in fp32 (the bf16 MFMA kernels would silently drop
# precision), and at fp32 rates the eager ops are not the bottleneck. The
# default (bf16) path never touches this branch.

_C = None
_C_err = None
try:
    from fms_fsdp_amd import _C  # built in-tree, travels with the repo snapshot
except ImportError as e:  # pragma: no cover - exercised only when unbuilt
    _C_err = e


def _require_ext(op):
    if _C is None:
        raise RuntimeError(
            f"fms_fsdp_amd._C HIP extension is required for {op} on GPU but "
            f"could not be imported ({_C_err}). Build it with "
            f"`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950)."
        )
    return _C


# --------------------------------------------------------------------------
# RMSNorm
# --------------------------------------------------------------------------
class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        ext = _require_ext("rmsnorm")
        x2d = x.contiguous().view(-1, x.shape[-1])
        y, rinv = ext.rmsnorm_fwd(x2d, weight, eps)
        ctx.save_for_backward(x2d, weight, rinv)
        ctx.shape = x.shape
        return y.view(x.shape)

    @staticmethod
    def backward(ctx, dy):
        x2d, weight, rinv = ctx.saved_tensors
        dx, dw = _C.rmsnorm_bwd(dy.contiguous().view_as(x2d), x2d, weight,
                                rinv, None)
        return dx.view(ctx.shape), dw.to(weight.dtype), None


def rmsnorm(x, weight, eps=1e-6):
    if x.is_cuda and x.dtype != torch.float32:
        if x.dtype == torch.float16:
            return _RMSNormFn.apply(x.bfloat16(), weight.bfloat16(),
                                    eps).half()
        return _RMSNormFn.apply(x, weight, eps)
    return reference.rmsnorm(x, weight, eps)


class _AddRMSNormFn(torch.autograd.Function):
    """Fused s = x + res; y = rmsnorm(s): one HBM pass instead of two.
    Backward fuses the residual grad into the rmsnorm dx kernel."""

    @staticmethod
    def forward(ctx, x, res, weight, eps):
        ext = _require_ext("add_rmsnorm")
        x2d = x.contiguous().view(-1, x.shape[-1])
        r2d = res.contiguous().view_as(x2d)
        y, s, rinv = ext.add_rmsnorm_fwd(x2d, r2d, weight, eps)
        ctx.save_for_backward(s, weight, rinv)
        ctx.shape = x.shape
        return y.view(x.shape), s.view(x.shape)

    @staticmethod
    def backward(ctx, dy, ds):
        s, weight, rinv = ctx.saved_tensors
        dx, dw = _C.rmsnorm_bwd(dy.contiguous().view_as(s), s, weight, rinv,
                                ds.contiguous().view_as(s))
        dx = dx.view(ctx.shape)
        return dx, dx, dw.to(weight.dtype), None


def add_rmsnorm(x, res, weight, eps=1e-6):
    """(rmsnorm(x + res), x + res)"""
    if x.is_cuda and x.dtype != torch.float32:
        if x.dtype == torch.float16:
            y, s = _AddRMSNormFn.apply(x.bfloat16(), res.bfloat16(),
                                       weight.bfloat16(), eps)
            return y.half(), s.half()
        return _AddRMSNormFn.apply(x, res, weight, eps)
    s = (x.float() + res.float()).to(x.dtype)
    return reference.rmsnorm(s, weight, eps), s


# --------------------------------------------------------------------------
# RoPE (half-rotation convention; cos/sin tables (s, d/2) fp32)
# --------------------------------------------------------------------------
def _row_view_ok(t):
    """(b, s, h, d) bf16 view the strided kernels accept directly:
    contiguous within each (b, s) row, uniform row stride (a last-dim
    slice of a fused qkv projection qualifies — no copy needed)."""
    return (t.dim() == 4 and t.stride(3) == 1 and t.stride(2) == t.shape[3]
            and t.stride(1) >= t.shape[2] * t.shape[3]
            and t.stride(0) == t.shape[1] * t.stride(1))


def _as_row_view(t):
    return t if _row_view_ok(t) else t.contiguous()


class _RoPEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, cos, sin):
        ext = _require_ext("rope")
        cos = cos.float().contiguous()   # tables may arrive bf16 after a
        sin = sin.float().contiguous()   # module-wide .bfloat16() cast
        # strided reads handle fused-qkv slices without a copy
        qo, ko = ext.rope_fwd(_as_row_view(q), _as_row_view(k), cos, sin,
                              False)
        ctx.save_for_backward(cos, sin)
        return qo, ko

    @staticmethod
    def backward(ctx, dq, dk):
        cos, sin = ctx.saved_tensors
        dqo, dko = _C.rope_fwd(_as_row_view(dq), _as_row_view(dk), cos, sin,
                               True)
        return dqo, dko, None, None


def rope_apply(q, k, cos, sin):
    if q.is_cuda and q.dtype != torch.float32:
        if q.dtype == torch.float16:
            qo, ko = _RoPEFn.apply(q.bfloat16(), k.bfloat16(), cos, sin)
            return qo.half(), ko.half()
        return _RoPEFn.apply(q, k, cos, sin)
    return reference.rope_apply(q, k, cos, sin)


# --------------------------------------------------------------------------
# Linear with wgrad written DIRECTLY into the sharded runtime's flat grad
# buffer (hipBLASLt beta=1 accumulate) instead of autograd's
# temp-then-add accumulation. The FSDP FlatUnit attaches
# `weight._flat_grad_view` (a view of the unit's flat bf16 grad buffer)
# and `weight._wgrad_done` (the unit's grad-countdown callback); models
# opt weights in by setting `weight._direct_wgrad = True`. Saves one
# full read+write pass over every projection grad per step (the
# CUDAFunctor_add kernels: ~8.5 ms/step on Llama2-7B, profiles/).
# --------------------------------------------------------------------------
class _DirectWgradLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, residual):
        ctx.save_for_backward(x, weight)
        # attribute holders stashed at forward time: saved_tensors can
        # return attr-less re-wrapped tensors (e.g. under AC recompute)
        ctx.gview = weight._flat_grad_view
        ctx.wobj = weight
        ctx.cb = getattr(weight, "_wgrad_done", None)
        ctx.res_shape = residual.shape if residual is not None else None
        x2 = x.reshape(-1, x.shape[-1])
        if residual is not None:
            y = torch.addmm(residual.reshape(x2.shape[0], -1), x2,
                            weight.t())
        else:
            y = x2 @ weight.t()
        return y.view(*x.shape[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        # NOTE: rerouting this dgrad through a transposed weight copy
        # (mm(dy, wT.t()) = hipBLASLt's faster "NN" class instead of the
        # "NT" class this layout hits) measured 29 ms/step SLOWER on the
        # 7B bench — the .t().contiguous() sweeps cost more than the
        # GEMM class difference saves. Measured and rejected.
        dx = (dy2 @ w).view(x.shape)
        # wgrad straight into the flat grad buffer. First touch since
        # zero_grad overwrites (beta=0) so the runtime never has to
        # zero-fill these slices; later touches accumulate (beta=1).
        if getattr(ctx.wobj, "_wgrad_fresh", False):
            torch.addmm(ctx.gview, dy2.t(), x2, beta=0, out=ctx.gview)
            ctx.wobj._wgrad_fresh = False
        else:
            ctx.gview.addmm_(dy2.t(), x2)
        if ctx.cb is not None:
            ctx.cb()
        dres = dy.reshape(ctx.res_shape) if ctx.res_shape is not None else None
        return dx, None, dres


def linear_flat(x, weight, residual=None):
    """F.linear (optionally fused with a residual add in the GEMM
    epilogue) that lands the weight grad directly in the sharded
    runtime's flat grad buffer when the weight is attached to one.
    Falls back to plain autograd ops otherwise (plain/unsharded models,
    frozen weights, no-grad inference)."""
    if (getattr(weight, "_flat_grad_view", None) is not None
            and weight.requires_grad and torch.is_grad_enabled()):
        return _DirectWgradLinearFn.apply(x, weight, residual)
    x2 = x.reshape(-1, x.shape[-1])
    if residual is not None:
        y = torch.addmm(residual.reshape(x2.shape[0], -1), x2, weight.t())
    else:
        y = x2 @ weight.t()
    return y.view(*x.shape[:-1], weight.shape[0])


# --------------------------------------------------------------------------
# Causal flash attention (MFMA tiled, GQA)
# --------------------------------------------------------------------------
class _FlashAttnFn(torch.autograd.Function):
    """Kernel operates on seq multiples of 128; arbitrary lengths are
    zero-padded at the END (causality keeps pad keys invisible to real
    queries) and sliced back."""

    @staticmethod
    def _pad(t, s_pad):
        b, s, h, d = t.shape
        if s == s_pad:
            return t.contiguous()
        out = torch.zeros(b, s_pad, h, d, dtype=t.dtype, device=t.device)
        out[:, :s] = t
        return out

    @staticmethod
    def forward(ctx, q, k, v):
        ext = _require_ext("attention")
        s = q.shape[1]
        s_pad = (s + 127) // 128 * 128
        qp = _FlashAttnFn._pad(q, s_pad)
        kp = _FlashAttnFn._pad(k, s_pad)
        vp = _FlashAttnFn._pad(v, s_pad)
        o, lse = ext.attn_fwd(qp, kp, vp)
        ctx.save_for_backward(qp, kp, vp, o, lse)
        ctx.s = s
        return o[:, :s] if s != s_pad else o

    @staticmethod
    def backward(ctx, do):
        qp, kp, vp, o, lse = ctx.saved_tensors
        s, s_pad = ctx.s, qp.shape[1]
        dop = _FlashAttnFn._pad(do, s_pad)
        dq, dk, dv = _C.attn_bwd(dop, qp, kp, vp, o, lse, None)
        if s != s_pad:
            dq, dk, dv = dq[:, :s], dk[:, :s], dv[:, :s]
        return dq, dk, dv


class _QKVRopeAttnFn(torch.autograd.Function):
    """Fused qkv-split + RoPE + causal flash attention over the FUSED qkv
    projection (b, s, (h+2kv)*d).

    Removes the per-layer copy traffic the modular path pays: q/k are
    rotated straight out of their strided qkv slices (strided rope
    reads), v feeds the attention kernels as a strided view (vstride
    plumbed into the HIP kernels), and the backward writes dq/dk/dv into
    ONE preallocated fused dqkv buffer — no autograd `cat` of the split,
    no zero-fill, no `.contiguous()` copies. Replaces the reference's
    separate rope+SDPA calls (fms MultiHeadAttention).
    """

    @staticmethod
    def forward(ctx, qkv, cos, sin, nheads, kvheads, head_dim):
        ext = _require_ext("attention")
        b, s, _ = qkv.shape
        d = head_dim
        hq, hk = nheads * d, kvheads * d
        q = qkv[..., :hq].view(b, s, nheads, d)
        k = qkv[..., hq:hq + hk].view(b, s, kvheads, d)
        v = qkv[..., hq + hk:].view(b, s, kvheads, d)
        cos = cos.float().contiguous()
        sin = sin.float().contiguous()
        qr, kr = ext.rope_fwd(q, k, cos, sin, False)   # strided reads
        o, lse = ext.attn_fwd(qr, kr, v)               # strided v
        ctx.save_for_backward(qkv, qr, kr, o, lse, cos, sin)
        ctx.dims = (nheads, kvheads, d)
        return o

    @staticmethod
    def backward(ctx, do):
        qkv, qr, kr, o, lse, cos, sin = ctx.saved_tensors
        nheads, kvheads, d = ctx.dims
        b, s, _ = qkv.shape
        hq, hk = nheads * d, kvheads * d
        v = qkv[..., hq + hk:].view(b, s, kvheads, d)
        dqkv = torch.empty_like(qkv)
        dq_sl = dqkv[..., :hq].view(b, s, nheads, d)
        dk_sl = dqkv[..., hq:hq + hk].view(b, s, kvheads, d)
        dv_sl = dqkv[..., hq + hk:].view(b, s, kvheads, d)
        if nheads == kvheads:
            # dv written straight into the fused buffer by the kernel
            dq, dk, _ = _C.attn_bwd(do.contiguous(), qr, kr, v, o, lse,
                                    dv_sl)
        else:
            dq, dk, dv = _C.attn_bwd(do.contiguous(), qr, kr, v, o, lse,
                                     None)
            dv_sl.copy_(dv)
        # inverse rotation scattered into the fused buffer slices
        _C.rope_into(dq, dq_sl, cos, sin, True)
        _C.rope_into(dk, dk_sl, cos, sin, True)
        return dqkv, None, None, None, None, None


def qkv_rope_attention(qkv, cos, sin, nheads, kvheads, head_dim):
    """Fused GPU path; callers gate on seq%128==0 and head_dim in
    (64, 128) and fall back to the modular split+rope+attention path
    otherwise (decode/KV-cache, odd lengths, CPU)."""
    if qkv.dtype == torch.float16:
        return _QKVRopeAttnFn.apply(qkv.bfloat16(), cos, sin, nheads,
                                    kvheads, head_dim).half()
    return _QKVRopeAttnFn.apply(qkv, cos, sin, nheads, kvheads, head_dim)


def attention_causal(q, k, v):
    if q.is_cuda:
        if os.environ.get("FMS_AMD_ALLOW_TORCH_SDPA") == "1":
            import torch.nn.functional as F
            o = F.scaled_dot_product_attention(
                q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
                is_causal=True, enable_gqa=(k.shape[2] != q.shape[2]))
            return o.transpose(1, 2)
        if q.dtype == torch.float16:
            return _FlashAttnFn.apply(q.bfloat16(), k.bfloat16(),
                                      v.bfloat16()).half()
        if q.dtype == torch.float32:
            return reference.attention_causal(q, k, v)
        return _FlashAttnFn.apply(q, k, v)
    return reference.attention_causal(q, k, v)


# --------------------------------------------------------------------------
# SwiGLU epilogue: silu(g) * u over fused (..., 2H) gate|up
# --------------------------------------------------------------------------
class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gu):
        ext = _require_ext("swiglu")
        gu = gu.contiguous()
        h = ext.swiglu_fwd(gu.view(-1, gu.shape[-1]))
        ctx.save_for_backward(gu)
        return h.view(*gu.shape[:-1], gu.shape[-1] // 2)

    @staticmethod
    def backward(ctx, dy):
        (gu,) = ctx.saved_tensors
        dgu = _C.swiglu_bwd(dy.contiguous().view(-1, dy.shape[-1]),
                            gu.view(-1, gu.shape[-1]))
        return dgu.view(gu.shape)


def swiglu(gu):
    if gu.is_cuda and gu.dtype != torch.float32:
        if gu.dtype == torch.float16:
            return _SwiGLUFn.apply(gu.bfloat16()).half()
        return _SwiGLUFn.apply(gu)
    return reference.swiglu(gu)


# --------------------------------------------------------------------------
# Fused linear + chunked cross-entropy (never materializes full fp32 logits)
# --------------------------------------------------------------------------
class _LinearCEFn(torch.autograd.Function):
    """Computes mean CE of linear(x, W) vs labels, chunked over rows.

    Forward also produces dx and dW (scaled for grad_output=1); backward
    rescales. This trades one saved dx/dW pair for never holding the
    (b*s, V) fp32 softmax (SURVEY.md hard-part 6: llama3 V=128256 logits
    would be ~8 GB fp32 at b2 s8192).
    """

    # rows per chunk: 8192 x 128256 bf16 logits = 2.1 GB transient — easily
    # afforded by 288 GB HBM, and the lm-head GEMMs run at full M
    CHUNK = 8192

    @staticmethod
    def forward(ctx, x, weight, labels, ignore_index):
        ext = _require_ext("cross_entropy")
        e = x.shape[-1]
        x2d = x.contiguous().view(-1, e)
        lab = labels.contiguous().view(-1)
        n = x2d.shape[0]
        dx = torch.empty_like(x2d)
        dw = torch.zeros_like(weight, dtype=torch.float32)
        loss_sum = torch.zeros((), device=x.device, dtype=torch.float32)
        count = (lab != ignore_index).sum()
        denom = count.clamp(min=1).float()

        for i in range(0, n, _LinearCEFn.CHUNK):
            xc = x2d[i:i + _LinearCEFn.CHUNK]
            lc = lab[i:i + _LinearCEFn.CHUNK]
            logits = torch.mm(xc, weight.t())          # hipBLASLt GEMM, bf16
            # in-place: logits -> dlogits (softmax - onehot)/denom, adds loss
            ext.ce_fwd_bwd(logits, lc, loss_sum, denom, ignore_index)
            torch.mm(logits, weight, out=dx[i:i + _LinearCEFn.CHUNK])
            dw.add_(torch.mm(logits.t(), xc).float())
        ctx.save_for_backward(dx, dw)
        ctx.xshape = x.shape
        ctx.wdtype = weight.dtype
        return loss_sum / denom

    @staticmethod
    def backward(ctx, gout):
        dx, dw = ctx.saved_tensors
        return (dx.view(ctx.xshape) * gout, (dw * gout).to(ctx.wdtype),
                None, None)


def linear_cross_entropy(x, weight, labels, ignore_index=-100):
    if x.is_cuda and x.dtype != torch.float32:
        if x.dtype == torch.float16:
            return _LinearCEFn.apply(x.bfloat16(), weight.bfloat16(),
                                     labels, ignore_index)
        return _LinearCEFn.apply(x, weight, labels, ignore_index)
    return reference.linear_cross_entropy(x, weight, labels, ignore_index)


# --------------------------------------------------------------------------
# Fused causal depthwise conv1d + silu (mamba xBC conv)
# --------------------------------------------------------------------------
class _CausalConv1dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ext = _require_ext("causal_conv1d")
        x = x.contiguous()
        wb = weight.contiguous().bfloat16()
        bf = bias.float().contiguous()
        y = ext.cconv_fwd(x, wb, bf)
        ctx.save_for_backward(x, wb, bf)
        ctx.wdtype = weight.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        x, wb, bf = ctx.saved_tensors
        dx, dw, db = _C.cconv_bwd(dy.contiguous(), x, wb, bf)
        return dx, dw.to(ctx.wdtype), db.to(ctx.wdtype)


def causal_conv1d(x, weight, bias):
    if x.is_cuda and x.dtype == torch.bfloat16:
        return _CausalConv1dFn.apply(x, weight, bias)
    return reference.causal_conv1d(x, weight, bias)


# --------------------------------------------------------------------------
# Fused exp(segsum) for the Mamba2 SSD scan: L[n,i,j] = exp(cs_i - cs_j)
# lower-triangular, emitted bf16 in one pass.
# --------------------------------------------------------------------------
class _SegsumExpFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, cs):
        ext = _require_ext("segsum_exp")
        cs = cs.contiguous()
        L = ext.segsum_exp_fwd(cs)
        ctx.save_for_backward(cs)
        return L

    @staticmethod
    def backward(ctx, gL):
        (cs,) = ctx.saved_tensors
        return _C.segsum_exp_bwd(gL.contiguous().to(torch.bfloat16), cs)


def segsum_exp(cs):
    """cs (..., Q) fp32 cumulative sums -> bf16 (..., Q, Q) decay matrix."""
    if cs.is_cuda:
        return _SegsumExpFn.apply(cs)
    Q = cs.shape[-1]
    out = cs[..., :, None] - cs[..., None, :]
    mask = torch.tril(torch.ones(Q, Q, dtype=torch.bool, device=cs.device), 0)
    return torch.exp(out.masked_fill(~mask, -torch.inf))


# --------------------------------------------------------------------------
# Fused SSD scan pieces (Mamba2): every fp32 elementwise chain around the
# batched GEMMs is a HIP kernel with a custom backward (north-star item:
# hand-written selective-scan; reference reaches mamba_ssm's kernels from
# main_training_mamba.py:8-9,67). GPU-only: the CPU path keeps the plain
# torch ssd_chunked as the numerics oracle.
# --------------------------------------------------------------------------
class _SSDPrepFn(torch.autograd.Function):
    """(dt2d strided bf16 (b*l, H), dt_bias, A_log) ->
    dtf (N,Q) fp32, dacs (N,Q) fp32 with N=(b*l/Q)*H, n = bc*H + h:
    dtf = softplus(dt + bias); dacs = chunk-cumsum(dtf * -exp(A_log))."""

    @staticmethod
    def forward(ctx, dt2d, dt_bias, A_log, chunk):
        ext = _require_ext("ssd_prep")
        bias_f = dt_bias.detach().float().contiguous()
        alog_f = A_log.detach().float().contiguous()
        dtf, dacs = ext.ssd_prep_fwd(dt2d, bias_f, alog_f, chunk)
        ctx.save_for_backward(dt2d, bias_f, alog_f)
        ctx.meta = (chunk, dt_bias.dtype, A_log.dtype)
        return dtf, dacs

    @staticmethod
    def backward(ctx, ddtf, ddacs):
        dt2d, bias_f, alog_f = ctx.saved_tensors
        chunk, bdt, adt = ctx.meta
        ddt, dbias, dalog = _C.ssd_prep_bwd(
            ddtf.contiguous(), ddacs.contiguous(), dt2d, bias_f, alog_f,
            chunk)
        return ddt, dbias.to(bdt), dalog.to(adt), None


def ssd_prep(dt2d, dt_bias, A_log, chunk):
    return _SSDPrepFn.apply(dt2d, dt_bias, A_log, chunk)


class _SSDXdtFn(torch.autograd.Function):
    """x2d (b*l, H*P) strided bf16 -> (xdt, xdt_decayed) bf16 contiguous
    in H-MAJOR layout (b,nc,H,Q,P) — bmm-native for y_diag (no einsum
    permute copies); xdt = x*dtf, decayed by exp(dacs_end-dacs)."""

    @staticmethod
    def forward(ctx, x2d, dtf, dacs, H, P, Q):
        ext = _require_ext("ssd_xdt")
        xdt, xdtd = ext.ssd_xdt_fwd(x2d, dtf, dacs, H, P, Q)
        ctx.save_for_backward(x2d, dtf, dacs)
        ctx.meta = (H, P, Q)
        return xdt, xdtd

    @staticmethod
    def backward(ctx, dxdt, dxdtd):
        x2d, dtf, dacs = ctx.saved_tensors
        H, P, Q = ctx.meta
        dx, ddtf, sdec = _C.ssd_xdt_bwd(
            dxdt.contiguous(), dxdtd.contiguous(), x2d, dtf, dacs, H, P, Q)
        # fold the decay-exponent grads: d/d dacs[q] = -sdec[q], and the
        # end column accumulates all q of its row
        ddacs = sdec.neg()
        ddacs[:, -1] += sdec.sum(-1)
        return dx, ddtf, ddacs, None, None, None


def ssd_xdt(x2d, dtf, dacs, H, P, Q):
    return _SSDXdtFn.apply(x2d, dtf, dacs, H, P, Q)


class _SSDScoresDecayFn(torch.autograd.Function):
    """sL[n,i,j] = scores[m,i,j] * exp(dacs[n,i]-dacs[n,j]) (j<=i), the
    decay matrix L never materialized. scores per GROUP (M=b*nc*G rows);
    backward returns per-head d_scores summed over each group's heads."""

    @staticmethod
    def forward(ctx, dacs, scores3d, H, G):
        ext = _require_ext("ssd_sl")
        sL = ext.ssd_sl_fwd(dacs, scores3d, H, G)
        ctx.save_for_backward(dacs, scores3d)
        ctx.meta = (H, G)
        return sL

    @staticmethod
    def backward(ctx, g):
        dacs, scores3d = ctx.saved_tensors
        H, G = ctx.meta
        dsh, dcs = _C.ssd_sl_bwd(g.contiguous(), scores3d, dacs, H, G)
        rep = H // G
        M = scores3d.shape[0]
        Q = scores3d.shape[1]
        dscores = dsh.view(M // G, G, rep, Q, Q) \
            .sum(dim=2, dtype=torch.float32).to(scores3d.dtype).view(M, Q, Q)
        return dcs, dscores, None, None


def ssd_scores_decay(dacs, scores3d, H, G):
    return _SSDScoresDecayFn.apply(dacs, scores3d, H, G)


class _SSDYGateFn(torch.autograd.Function):
    """out = (ydiag + yoff*exp(dacs) + x*D) * silu(z), bf16 in one pass
    (y assembly + D residual + the gated epilogue)."""

    @staticmethod
    def forward(ctx, ydiag, yoff, dacs, x2d, D, z2d, H, G, P, Q):
        ext = _require_ext("ssd_ygate")
        D_f = D.detach().float().contiguous()
        out = ext.ssd_ygate_fwd(ydiag, yoff, dacs, x2d, D_f, z2d, H, G, P, Q)
        ctx.save_for_backward(ydiag, yoff, dacs, x2d, D_f, z2d)
        ctx.meta = (H, G, P, Q, D.dtype)
        return out

    @staticmethod
    def backward(ctx, dout):
        ydiag, yoff, dacs, x2d, D_f, z2d = ctx.saved_tensors
        H, G, P, Q, ddt = ctx.meta
        dydiag, dyoff, ddacs, dx, dD_rows, dz = _C.ssd_ygate_bwd(
            dout.contiguous(), ydiag, yoff, dacs, x2d, D_f, z2d, H, G, P, Q)
        dD = dD_rows.sum(0)
        return (dydiag, dyoff, ddacs, dx, dD.to(ddt), dz, None, None, None,
                None)


def ssd_ygate(ydiag, yoff, dacs, x2d, D, z2d, H, G, P, Q):
    """ydiag h-major (b,nc,h,Q,p) flat; yoff in the y_off einsum's
    natural (b,nc,g,Q,rep,p) layout; out (b, l, H*P) row-major."""
    return _SSDYGateFn.apply(ydiag, yoff, dacs, x2d, D, z2d, H, G, P, Q)


# --------------------------------------------------------------------------
# Fused AdamW on flat fp32 shards + multi-tensor sq-norm
# --------------------------------------------------------------------------
def fused_adamw(p, g, m, v, step, lr, beta1, beta2, eps, weight_decay,
                grad_scale=None, p_bf16_out=None):
    """In-place AdamW on 1-D fp32 tensors. g may be fp32, bf16 or fp16;
    grad_scale (0-dim fp32 tensor) folds grad clipping (and fp16 loss
    unscaling) into the update; p_bf16_out receives the updated bf16 OR
    fp16 shard in the same pass."""
    if p.is_cuda:
        ext = _require_ext("adamw")
        ext.adamw(p, g.view(-1), m, v, float(step), lr, beta1, beta2, eps,
                  weight_decay, grad_scale,
                  p_bf16_out.view(-1) if p_bf16_out is not None else None)
        return True
    gf = g.float().view(-1)
    if grad_scale is not None:
        gf = gf * grad_scale
    reference.adamw_step(p, gf, m, v, step, lr, beta1, beta2, eps,
                         weight_decay)
    return False


def sq_norm(tensors):
    """Sum of squares over a list of tensors -> fp32 scalar tensor."""
    tensors = [t for t in tensors if t.numel() > 0]
    if tensors and tensors[0].is_cuda and _C is not None:
        out = torch.zeros((), device=tensors[0].device, dtype=torch.float32)
        for t in tensors:
            _C.sq_norm_accum(t.view(-1), out)
        return out
    return sum((t.float().pow(2).sum() for t in tensors),
               torch.zeros((), dtype=torch.float32,
                           device=tensors[0].device if tensors else "cpu"))
