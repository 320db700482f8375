// Python bindings for the CDNA4 kernel library (fms_fsdp_amd._C).
// Pure dispatch: shape/dtype checks + launch on the current HIP stream.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime_api.h>

#include <tuple>
#include <vector>

using torch::Tensor;

extern "C" {
void launch_rmsnorm_fwd(const void*, const void*, void*, float*, int, int,
                        float, hipStream_t);
void launch_rmsnorm_bwd(const void*, const void*, const void*, const float*,
                        const void*, void*, float*, int, int, hipStream_t);
void launch_add_rmsnorm_fwd(const void*, const void*, const void*, void*,
                            void*, float*, int, int, float, hipStream_t);
void launch_rope(const void*, void*, const float*, const float*, int, int,
                 int, int, int, long long, long long, hipStream_t);
void launch_swiglu_fwd(const void*, void*, long long, int, hipStream_t);
void launch_swiglu_bwd(const void*, const void*, void*, long long, int,
                       hipStream_t);
void launch_ce_fwd_bwd(void*, const long long*, float*, const float*,
                       long long, int, int, hipStream_t);
void launch_adamw(float*, const void*, int, float*, float*, void*, int,
                  long long, float, float, float, float, float, float, float,
                  const float*, hipStream_t);
void launch_sqnorm(const void*, int, float*, long long, hipStream_t);
void launch_attn_fwd(const void*, const void*, const void*, void*, float*,
                     int, int, int, int, int, float, long long, hipStream_t);
void launch_attn_bwd(const void*, const void*, const void*, const void*,
                     const void*, const float*, void*, void*, void*, float*,
                     void*, int, int, int, int, int, float, int,
                     long long, long long, hipStream_t);
void launch_cconv_fwd(const void*, const void*, const float*, void*, int,
                      int, int, int, hipStream_t);
void launch_cconv_bwd(const void*, const void*, const void*, const float*,
                      void*, void*, float*, float*, int, int, int, int,
                      hipStream_t);
int launch_gemm_nt(const void*, const void*, void*, int, int, int,
                   hipStream_t);
void launch_segsum_exp_fwd(const float*, void*, long long, int, hipStream_t);
void launch_segsum_exp_bwd(const void*, const float*, float*, long long, int,
                           hipStream_t);
void launch_ssd_prep_fwd(const void*, const float*, const float*, float*,
                         float*, long long, int, int, long long, hipStream_t);
void launch_ssd_prep_bwd(const float*, const float*, const void*,
                         const float*, const float*, void*, float*, float*,
                         long long, int, int, long long, long long,
                         hipStream_t);
void launch_ssd_xdt_fwd(const void*, const float*, const float*, void*, void*,
                        long long, int, int, int, long long, hipStream_t);
void launch_ssd_xdt_bwd(const void*, const void*, const void*, const float*,
                        const float*, void*, float*, float*, long long, int,
                        int, int, long long, long long, hipStream_t);
void launch_ssd_sl_fwd(const float*, const void*, void*, long long, int, int,
                       int, hipStream_t);
void launch_ssd_sl_bwd(const void*, const void*, const float*, void*, float*,
                       long long, int, int, int, hipStream_t);
void launch_ssd_ygate_fwd(const void*, const void*, const float*, const void*,
                          const float*, const void*, void*, long long, int,
                          int, int, int, long long, long long, hipStream_t);
void launch_ssd_ygate_bwd(const void*, const void*, const void*, const float*,
                          const void*, const float*, const void*, void*,
                          void*, float*, void*, float*, void*, long long, int,
                          int, int, int, long long, long long, long long,
                          long long, hipStream_t);
}

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

#define CHECK_BF16_CONTIG(t)                                        \
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, #t " must be bf16"); \
  TORCH_CHECK(t.is_contiguous(), #t " must be contiguous");

std::tuple<Tensor, Tensor> rmsnorm_fwd(Tensor x, Tensor w, double eps) {
  CHECK_BF16_CONTIG(x);
  CHECK_BF16_CONTIG(w);
  const int rows = x.size(0), H = x.size(1);
  TORCH_CHECK(H % 8 == 0, "H must be divisible by 8");
  auto y = torch::empty_like(x);
  auto rinv = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  launch_rmsnorm_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                     rinv.data_ptr<float>(), rows, H, (float)eps,
                     cur_stream());
  return {y, rinv};
}

std::tuple<Tensor, Tensor> rmsnorm_bwd(Tensor dy, Tensor x, Tensor w,
                                       Tensor rinv,
                                       c10::optional<Tensor> dextra) {
  CHECK_BF16_CONTIG(dy);
  CHECK_BF16_CONTIG(x);
  const int rows = x.size(0), H = x.size(1);
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({H}, x.options().dtype(torch::kFloat32));
  launch_rmsnorm_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                     rinv.data_ptr<float>(),
                     dextra.has_value() ? dextra->data_ptr() : nullptr,
                     dx.data_ptr(), dw.data_ptr<float>(), rows, H,
                     cur_stream());
  return {dx, dw};
}

std::tuple<Tensor, Tensor, Tensor> add_rmsnorm_fwd(Tensor x, Tensor res,
                                                   Tensor w, double eps) {
  CHECK_BF16_CONTIG(x);
  CHECK_BF16_CONTIG(res);
  const int rows = x.size(0), H = x.size(1);
  auto y = torch::empty_like(x);
  auto s_out = torch::empty_like(x);
  auto rinv = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  launch_add_rmsnorm_fwd(x.data_ptr(), res.data_ptr(), w.data_ptr(),
                         y.data_ptr(), s_out.data_ptr(),
                         rinv.data_ptr<float>(), rows, H, (float)eps,
                         cur_stream());
  return {y, s_out, rinv};
}

// (b, s, h, d) bf16 view, contiguous within a (b, s) row, uniform row
// stride — a last-dim slice of a fused qkv projection qualifies.
static long long row_stride4(const Tensor& t) {
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, "expected bf16");
  TORCH_CHECK(t.dim() == 4 && t.stride(3) == 1 &&
                  t.stride(2) == t.size(3) &&
                  t.stride(1) >= t.size(2) * t.size(3) &&
                  t.stride(0) == t.size(1) * t.stride(1),
              "expected row-contiguous (b,s,h,d) view");
  return (long long)t.stride(1);
}

std::tuple<Tensor, Tensor> rope_fwd(Tensor q, Tensor k, Tensor cos,
                                    Tensor sin, bool conj) {
  TORCH_CHECK(cos.scalar_type() == torch::kFloat32 && cos.is_contiguous());
  const int b = q.size(0), s = q.size(1), h = q.size(2), d = q.size(3);
  const int kvh = k.size(2);
  const long long qs = row_stride4(q), ks = row_stride4(k);
  TORCH_CHECK((d / 2) % 4 == 0, "head_dim/2 must be divisible by 4");
  auto qo = torch::empty({b, s, h, d}, q.options());
  auto ko = torch::empty({b, s, kvh, d}, k.options());
  launch_rope(q.data_ptr(), qo.data_ptr(), cos.data_ptr<float>(),
              sin.data_ptr<float>(), b, s, h, d, conj, qs,
              (long long)h * d, cur_stream());
  launch_rope(k.data_ptr(), ko.data_ptr(), cos.data_ptr<float>(),
              sin.data_ptr<float>(), b, s, kvh, d, conj, ks,
              (long long)kvh * d, cur_stream());
  return {qo, ko};
}

// Rotate `src` (contiguous (b,s,h,d)) and scatter into `dst`, a
// row-strided view (slice of a fused dqkv grad buffer). Used by the
// fused qkv+rope+attention backward.
void rope_into(Tensor src, Tensor dst, Tensor cos, Tensor sin, bool conj) {
  const int b = src.size(0), s = src.size(1), h = src.size(2),
            d = src.size(3);
  const long long is = row_stride4(src), os = row_stride4(dst);
  TORCH_CHECK(dst.size(2) == h && dst.size(3) == d);
  launch_rope(src.data_ptr(), dst.data_ptr(), cos.data_ptr<float>(),
              sin.data_ptr<float>(), b, s, h, d, conj, is, os,
              cur_stream());
}

Tensor swiglu_fwd(Tensor gu) {
  CHECK_BF16_CONTIG(gu);
  const long long rows = gu.size(0);
  const int H2 = gu.size(1);
  TORCH_CHECK(H2 % 16 == 0);
  auto h = torch::empty({rows, H2 / 2}, gu.options());
  launch_swiglu_fwd(gu.data_ptr(), h.data_ptr(), rows, H2 / 2, cur_stream());
  return h;
}

Tensor swiglu_bwd(Tensor dy, Tensor gu) {
  CHECK_BF16_CONTIG(dy);
  CHECK_BF16_CONTIG(gu);
  const long long rows = gu.size(0);
  const int H2 = gu.size(1);
  auto dgu = torch::empty_like(gu);
  launch_swiglu_bwd(dy.data_ptr(), gu.data_ptr(), dgu.data_ptr(), rows,
                    H2 / 2, cur_stream());
  return dgu;
}

void ce_fwd_bwd(Tensor logits, Tensor labels, Tensor loss_sum, Tensor denom,
                int64_t ignore_index) {
  CHECK_BF16_CONTIG(logits);
  TORCH_CHECK(labels.scalar_type() == torch::kInt64 && labels.is_contiguous());
  const int rows = logits.size(0), V = logits.size(1);
  launch_ce_fwd_bwd(logits.data_ptr(),
                    reinterpret_cast<const long long*>(labels.data_ptr<int64_t>()),
                    loss_sum.data_ptr<float>(), denom.data_ptr<float>(),
                    (long long)ignore_index, rows, V, cur_stream());
}

void adamw(Tensor p, Tensor g, Tensor m, Tensor v, double step, double lr,
           double b1, double b2, double eps, double wd,
           c10::optional<Tensor> grad_scale,
           c10::optional<Tensor> p_bf16_out) {
  TORCH_CHECK(p.scalar_type() == torch::kFloat32 && p.is_contiguous());
  const long long n = p.numel();
  TORCH_CHECK(n % 4 == 0, "shard size must be divisible by 4");
  auto dt_code = [](torch::ScalarType t) {
    if (t == torch::kBFloat16) return 1;
    if (t == torch::kFloat16) return 2;
    TORCH_CHECK(t == torch::kFloat32, "dtype must be fp32/bf16/fp16");
    return 0;
  };
  const int gdt = dt_code(g.scalar_type());
  int odt = 0;
  if (p_bf16_out.has_value()) {
    odt = dt_code(p_bf16_out->scalar_type());
    TORCH_CHECK(odt != 0, "low-precision publish target must be bf16/fp16");
  }
  const float bc1 = 1.f - powf((float)b1, (float)step);
  const float bc2 = 1.f - powf((float)b2, (float)step);
  launch_adamw(p.data_ptr<float>(), g.data_ptr(), gdt, m.data_ptr<float>(),
               v.data_ptr<float>(),
               p_bf16_out.has_value() ? p_bf16_out->data_ptr() : nullptr, odt,
               n,
               (float)lr, (float)b1, (float)b2, (float)eps, (float)wd, bc1,
               bc2,
               grad_scale.has_value() ? grad_scale->data_ptr<float>() : nullptr,
               cur_stream());
}

void sq_norm_accum(Tensor t, Tensor out) {
  TORCH_CHECK(t.is_contiguous());
  const long long n = t.numel();
  TORCH_CHECK(n % 4 == 0);
  const int dt = t.scalar_type() == torch::kBFloat16 ? 1
                 : t.scalar_type() == torch::kFloat16 ? 2 : 0;
  launch_sqnorm(t.data_ptr(), dt, out.data_ptr<float>(), n, cur_stream());
}

std::tuple<Tensor, Tensor> attn_fwd(Tensor q, Tensor k, Tensor v) {
  CHECK_BF16_CONTIG(q);
  CHECK_BF16_CONTIG(k);
  const long long vs = row_stride4(v);  // strided (fused-qkv slice) ok
  const int b = q.size(0), s = q.size(1), h = q.size(2), d = q.size(3);
  const int kvh = k.size(2);
  TORCH_CHECK(d == 64 || d == 128, "head_dim must be 64 or 128");
  TORCH_CHECK(s % 128 == 0, "seq len must be a multiple of 128");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({b, h, s}, q.options().dtype(torch::kFloat32));
  const float scale = 1.f / sqrtf((float)d);
  launch_attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                  lse.data_ptr<float>(), b, s, h, kvh, d, scale, vs,
                  cur_stream());
  return {o, lse};
}

std::tuple<Tensor, Tensor, Tensor> attn_bwd(Tensor do_, Tensor q, Tensor k,
                                            Tensor v, Tensor o, Tensor lse,
                                            c10::optional<Tensor> dv_out) {
  CHECK_BF16_CONTIG(do_);
  const long long vs = row_stride4(v);
  const int b = q.size(0), s = q.size(1), h = q.size(2), d = q.size(3);
  const int kvh = k.size(2);
  auto dq = torch::empty_like(q);
  const bool out_bf16 = (kvh == h);  // no GQA collisions: direct bf16
  auto opts = q.options().dtype(out_bf16 ? torch::kBFloat16 : torch::kFloat32);
  auto dk = out_bf16 ? torch::empty({b, s, kvh, d}, opts)
                     : torch::zeros({b, s, kvh, d}, opts);
  // dv may land directly in a strided slice of a fused dqkv buffer
  // (bf16 path only — the GQA path accumulates in fp32 scratch)
  Tensor dv;
  long long dvs = (long long)kvh * d;
  if (dv_out.has_value() && out_bf16) {
    dv = *dv_out;
    dvs = row_stride4(dv);
  } else {
    dv = out_bf16 ? torch::empty({b, s, kvh, d}, opts)
                  : torch::zeros({b, s, kvh, d}, opts);
  }
  auto delta = torch::empty({b, h, s}, q.options().dtype(torch::kFloat32));
  // dS workspace (b, h, s, s) bf16: written by dkv, consumed by dq —
  // the backward's P/dP recompute runs once instead of twice. ~2.1 GB
  // at the 7B shape; the caching allocator reuses it across layers.
  auto ds_ws = torch::empty({(long long)b * h, (long long)s, (long long)s},
                            q.options());
  const float scale = 1.f / sqrtf((float)d);
  launch_attn_bwd(do_.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
                  o.data_ptr(), lse.data_ptr<float>(), dq.data_ptr(),
                  dk.data_ptr(), dv.data_ptr(), delta.data_ptr<float>(),
                  ds_ws.data_ptr(), b, s,
                  h, kvh, d, scale, out_bf16 ? 1 : 0, vs, dvs, cur_stream());
  if (out_bf16) return {dq, dk, dv};
  return {dq, dk.to(torch::kBFloat16), dv.to(torch::kBFloat16)};
}

Tensor cconv_fwd(Tensor x, Tensor w, Tensor bias) {
  CHECK_BF16_CONTIG(x);
  CHECK_BF16_CONTIG(w);
  const int b = x.size(0), l = x.size(1), c = x.size(2);
  const int W = w.size(1);
  TORCH_CHECK(c % 8 == 0 && W >= 2 && W <= 4);
  TORCH_CHECK(bias.scalar_type() == torch::kFloat32);
  auto y = torch::empty_like(x);
  launch_cconv_fwd(x.data_ptr(), w.data_ptr(), bias.data_ptr<float>(),
                   y.data_ptr(), b * l, l, c, W, cur_stream());
  return y;
}

std::tuple<Tensor, Tensor, Tensor> cconv_bwd(Tensor dy, Tensor x, Tensor w,
                                             Tensor bias) {
  CHECK_BF16_CONTIG(dy);
  CHECK_BF16_CONTIG(x);
  const int b = x.size(0), l = x.size(1), c = x.size(2);
  const int W = w.size(1);
  TORCH_CHECK(c % 8 == 0 && W >= 2 && W <= 4);
  auto g = torch::empty_like(x);
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({c, W}, x.options().dtype(torch::kFloat32));
  auto db = torch::zeros({c}, x.options().dtype(torch::kFloat32));
  launch_cconv_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                   bias.data_ptr<float>(), g.data_ptr(), dx.data_ptr(),
                   dw.data_ptr<float>(), db.data_ptr<float>(), b * l, l, c, W,
                   cur_stream());
  return {dx, dw, db};
}

Tensor segsum_exp_fwd(Tensor cs) {
  TORCH_CHECK(cs.scalar_type() == torch::kFloat32 && cs.is_contiguous());
  const long long N = cs.numel() / cs.size(-1);
  const int Q = cs.size(-1);
  TORCH_CHECK(Q % 8 == 0);
  std::vector<int64_t> shape(cs.sizes().begin(), cs.sizes().end());
  shape.push_back(Q);
  auto out = torch::empty(shape, cs.options().dtype(torch::kBFloat16));
  launch_segsum_exp_fwd(cs.data_ptr<float>(), out.data_ptr(), N, Q,
                        cur_stream());
  return out;
}

Tensor segsum_exp_bwd(Tensor g, Tensor cs) {
  CHECK_BF16_CONTIG(g);
  const long long N = cs.numel() / cs.size(-1);
  const int Q = cs.size(-1);
  auto dcs = torch::empty_like(cs);
  launch_segsum_exp_bwd(g.data_ptr(), cs.data_ptr<float>(),
                        dcs.data_ptr<float>(), N, Q, cur_stream());
  return dcs;
}


Tensor gemm_nt(Tensor A, Tensor B) {
  CHECK_BF16_CONTIG(A);
  CHECK_BF16_CONTIG(B);
  const int M = A.size(0), K = A.size(1), N = B.size(1);
  TORCH_CHECK(B.size(0) == K);
  auto C = torch::empty({M, N}, A.options());
  const int rc = launch_gemm_nt(A.data_ptr(), B.data_ptr(), C.data_ptr(), M,
                                N, K, cur_stream());
  TORCH_CHECK(rc == 0, "gemm_nt: shape not supported (M%256/N%256/K%64)");
  return C;
}

// ---- fused SSD scan pieces (see ops/hip/ssd.hip) ----
// strided 2-D bf16 slice (rows, cols) with unit inner stride
static long long slice_stride(const Tensor& t, const char* nm) {
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, nm, " must be bf16");
  TORCH_CHECK(t.dim() == 2 && t.stride(1) == 1, nm, " must be 2-D row-major");
  return t.stride(0);
}

std::tuple<Tensor, Tensor> ssd_prep_fwd(Tensor dt, Tensor bias, Tensor alog,
                                        int64_t Q) {
  const long long sdt = slice_stride(dt, "dt");
  const long long rows = dt.size(0);
  const int H = dt.size(1);
  TORCH_CHECK(rows % Q == 0 && Q <= 256 && Q % 64 == 0);
  const long long N = rows / Q * H;
  auto opt = dt.options().dtype(torch::kFloat32);
  auto dtf = torch::empty({N, Q}, opt);
  auto dacs = torch::empty({N, Q}, opt);
  launch_ssd_prep_fwd(dt.data_ptr(), bias.data_ptr<float>(),
                      alog.data_ptr<float>(), dtf.data_ptr<float>(),
                      dacs.data_ptr<float>(), N, H, (int)Q, sdt,
                      cur_stream());
  return {dtf, dacs};
}

std::tuple<Tensor, Tensor, Tensor> ssd_prep_bwd(Tensor ddtf, Tensor ddacs,
                                                Tensor dt, Tensor bias,
                                                Tensor alog, int64_t Q) {
  const long long sdt = slice_stride(dt, "dt");
  const long long rows = dt.size(0);
  const int H = dt.size(1);
  const long long N = rows / Q * H;
  auto ddt = torch::empty({rows, (long long)H},
                          dt.options().dtype(torch::kBFloat16));
  auto dbias = torch::zeros({H}, dt.options().dtype(torch::kFloat32));
  auto dalog = torch::zeros({H}, dt.options().dtype(torch::kFloat32));
  launch_ssd_prep_bwd(ddtf.data_ptr<float>(), ddacs.data_ptr<float>(),
                      dt.data_ptr(), bias.data_ptr<float>(),
                      alog.data_ptr<float>(), ddt.data_ptr(),
                      dbias.data_ptr<float>(), dalog.data_ptr<float>(), N, H,
                      (int)Q, sdt, H, cur_stream());
  return {ddt, dbias, dalog};
}

std::tuple<Tensor, Tensor> ssd_xdt_fwd(Tensor x, Tensor dtf, Tensor dacs,
                                       int64_t H, int64_t P, int64_t Q) {
  const long long sx = slice_stride(x, "x");
  const long long rows = x.size(0);
  auto xdt = torch::empty({rows, H * P},
                          x.options().dtype(torch::kBFloat16));
  auto xdtd = torch::empty_like(xdt);
  const long long total8 = rows * H * P / 8;
  launch_ssd_xdt_fwd(x.data_ptr(), dtf.data_ptr<float>(),
                     dacs.data_ptr<float>(), xdt.data_ptr(), xdtd.data_ptr(),
                     total8, (int)H, (int)Q, (int)P, sx, cur_stream());
  return {xdt, xdtd};
}

std::tuple<Tensor, Tensor, Tensor> ssd_xdt_bwd(Tensor dxdt, Tensor dxdtd,
                                               Tensor x, Tensor dtf,
                                               Tensor dacs, int64_t H,
                                               int64_t P, int64_t Q) {
  const long long sx = slice_stride(x, "x");
  const long long rows = x.size(0);
  auto dx = torch::empty({rows, H * P}, x.options().dtype(torch::kBFloat16));
  auto ddtf = torch::empty_like(dtf);
  auto sdec = torch::empty_like(dacs);
  launch_ssd_xdt_bwd(dxdt.data_ptr(), dxdtd.data_ptr(), x.data_ptr(),
                     dtf.data_ptr<float>(), dacs.data_ptr<float>(),
                     dx.data_ptr(), ddtf.data_ptr<float>(),
                     sdec.data_ptr<float>(), rows * H, (int)H, (int)Q,
                     (int)P, sx, H * P, cur_stream());
  return {dx, ddtf, sdec};
}

Tensor ssd_sl_fwd(Tensor dacs, Tensor scores, int64_t H, int64_t G) {
  TORCH_CHECK(scores.scalar_type() == torch::kBFloat16 &&
              scores.is_contiguous());
  const long long N = dacs.size(0);
  const int Q = dacs.size(1);
  auto out = torch::empty({N, (long long)Q, (long long)Q},
                          scores.options());
  launch_ssd_sl_fwd(dacs.data_ptr<float>(), scores.data_ptr(),
                    out.data_ptr(), N * Q * (Q / 8), (int)H, (int)G, Q,
                    cur_stream());
  return out;
}

std::tuple<Tensor, Tensor> ssd_sl_bwd(Tensor g, Tensor scores, Tensor dacs,
                                      int64_t H, int64_t G) {
  const long long N = dacs.size(0);
  const int Q = dacs.size(1);
  auto dsh = torch::empty({N, (long long)Q, (long long)Q}, g.options());
  auto dcs = torch::zeros_like(dacs);
  launch_ssd_sl_bwd(g.data_ptr(), scores.data_ptr(), dacs.data_ptr<float>(),
                    dsh.data_ptr(), dcs.data_ptr<float>(), N, (int)H, (int)G,
                    Q, cur_stream());
  return {dsh, dcs};
}

Tensor ssd_ygate_fwd(Tensor ydiag, Tensor yoff, Tensor dacs, Tensor x,
                     Tensor Dp, Tensor z, int64_t H, int64_t G, int64_t P,
                     int64_t Q) {
  const long long sx = slice_stride(x, "x");
  const long long sz = slice_stride(z, "z");
  TORCH_CHECK(ydiag.is_contiguous() && yoff.is_contiguous());
  const long long rows = x.size(0);
  auto out = torch::empty({rows, H * P}, x.options().dtype(torch::kBFloat16));
  launch_ssd_ygate_fwd(ydiag.data_ptr(), yoff.data_ptr(),
                       dacs.data_ptr<float>(), x.data_ptr(),
                       Dp.data_ptr<float>(), z.data_ptr(), out.data_ptr(),
                       rows * H * P / 8, (int)H, (int)G, (int)Q, (int)P, sx,
                       sz, cur_stream());
  return out;
}

std::vector<Tensor> ssd_ygate_bwd(Tensor dout, Tensor ydiag, Tensor yoff,
                                  Tensor dacs, Tensor x, Tensor Dp, Tensor z,
                                  int64_t H, int64_t G, int64_t P,
                                  int64_t Q) {
  const long long sx = slice_stride(x, "x");
  const long long sz = slice_stride(z, "z");
  const long long rows = x.size(0);
  auto o = x.options().dtype(torch::kBFloat16);
  auto dydiag = torch::empty({rows, H * P}, o);
  auto dyoff = torch::empty({rows, H * P}, o);
  auto ddacs = torch::empty_like(dacs);
  auto dx = torch::empty({rows, H * P}, o);
  auto dD_rows = torch::empty({rows, H}, x.options().dtype(torch::kFloat32));
  auto dz = torch::empty({rows, H * P}, o);
  launch_ssd_ygate_bwd(dout.data_ptr(), ydiag.data_ptr(), yoff.data_ptr(),
                       dacs.data_ptr<float>(), x.data_ptr(),
                       Dp.data_ptr<float>(), z.data_ptr(), dydiag.data_ptr(),
                       dyoff.data_ptr(), ddacs.data_ptr<float>(),
                       dx.data_ptr(), dD_rows.data_ptr<float>(),
                       dz.data_ptr(), rows * H, (int)H, (int)G, (int)Q,
                       (int)P, sx, sz, H * P, H * P, cur_stream());
  return {dydiag, dyoff, ddacs, dx, dD_rows, dz};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("segsum_exp_fwd", &segsum_exp_fwd);
  mod.def("segsum_exp_bwd", &segsum_exp_bwd);
  mod.def("cconv_fwd", &cconv_fwd);
  mod.def("cconv_bwd", &cconv_bwd);
  mod.def("rmsnorm_fwd", &rmsnorm_fwd);
  mod.def("rmsnorm_bwd", &rmsnorm_bwd);
  mod.def("add_rmsnorm_fwd", &add_rmsnorm_fwd);
  mod.def("rope_fwd", &rope_fwd);
  mod.def("rope_into", &rope_into);
  mod.def("swiglu_fwd", &swiglu_fwd);
  mod.def("swiglu_bwd", &swiglu_bwd);
  mod.def("ce_fwd_bwd", &ce_fwd_bwd);
  mod.def("adamw", &adamw);
  mod.def("sq_norm_accum", &sq_norm_accum);
  mod.def("gemm_nt", &gemm_nt);
  mod.def("ssd_prep_fwd", &ssd_prep_fwd);
  mod.def("ssd_prep_bwd", &ssd_prep_bwd);
  mod.def("ssd_xdt_fwd", &ssd_xdt_fwd);
  mod.def("ssd_xdt_bwd", &ssd_xdt_bwd);
  mod.def("ssd_sl_fwd", &ssd_sl_fwd);
  mod.def("ssd_sl_bwd", &ssd_sl_bwd);
  mod.def("ssd_ygate_fwd", &ssd_ygate_fwd);
  mod.def("ssd_ygate_bwd", &ssd_ygate_bwd);
  mod.def("attn_fwd", &attn_fwd);
  mod.def("attn_bwd", &attn_bwd);
}
