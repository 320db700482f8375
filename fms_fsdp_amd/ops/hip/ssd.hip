// Fused pieces of the Mamba2 SSD scan (SURVEY.md §2.3 selective-scan row;
// reference reaches mamba_ssm's scan kernels from main_training_mamba.py).
// Round 2 makes the selective-scan hand-written in fact: every fp32
// elementwise chain around the batched GEMMs is a HIP kernel with a
// custom backward —
//   ssd_prep:  softplus(dt+bias), dA = dtf*A, chunk-local cumsum
//   ssd_xdt:   xdt = x*dtf and its state-decayed copy, emitted bf16
//   sL fused:  sL[i,j] = scores[g] * exp(cs_i - cs_j)  (decay matrix
//              never materialized separately; d_scores in backward)
//   ssd_ygate: y = y_diag + y_off*exp(cs) + x*D, out = y * silu(z)
// cs layouts are (N, Q) fp32 with N = b*nc*h; x/z/dt are strided slices
// of the fused in_proj/conv outputs (no .contiguous() copies).
#include "common.h"

__device__ __forceinline__ float sigmoidf(float x) {
  return 1.f / (1.f + __expf(-x));
}

// ---------------------------------------------------------------------
// L = exp(segsum(cs)) (kept for the plain segsum_exp op / tests)
// ---------------------------------------------------------------------
__global__ void segsum_exp_fwd_kernel(const float* __restrict__ cs,
                                      short* __restrict__ out,
                                      int Q, long long total8) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= total8) return;
  const int Q8 = Q / 8;
  const long long row = idx / Q8;          // (n, i) flattened
  const int j0 = (int)(idx % Q8) * 8;
  const long long n = row / Q;
  const int i = (int)(row % Q);
  const float ci = cs[n * Q + i];
  const f32x4 cj0 = *(const f32x4*)(cs + n * Q + j0);
  const f32x4 cj1 = *(const f32x4*)(cs + n * Q + j0 + 4);
  bf16x8 o;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    const int j = j0 + e;
    const float cj = e < 4 ? cj0.v[e] : cj1.v[e - 4];
    o.v[e] = (j <= i) ? f2bf(__expf(ci - cj)) : (short)0;
  }
  *(bf16x8*)(out + row * Q + j0) = o;
}

// d_cs[n,k] = sum_j g[n,k,j]*L[n,k,j]  -  sum_i g[n,i,k]*L[n,i,k]
// pass 1 (rows): one wave per (n,i): coalesced over j
__global__ void segsum_exp_bwd_row_kernel(const short* __restrict__ g,
                                          const float* __restrict__ cs,
                                          float* __restrict__ dcs,
                                          int Q, long long rows) {
  const long long row = ((long long)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const long long n = row / Q;
  const int i = (int)(row % Q);
  const float ci = cs[n * Q + i];
  float acc = 0.f;
  for (int j = lane; j <= i; j += 64)
    acc += bf2f(g[row * Q + j]) * __expf(ci - cs[n * Q + j]);
  acc = wave_reduce_sum(acc);
  if (lane == 0) dcs[row] = acc;
}

// pass 2 (columns): one wave per (n,j): strided over i (L2-assisted)
__global__ void segsum_exp_bwd_col_kernel(const short* __restrict__ g,
                                          const float* __restrict__ cs,
                                          float* __restrict__ dcs,
                                          int Q, long long cols) {
  const long long cidx = ((long long)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  if (cidx >= cols) return;
  const int lane = threadIdx.x & 63;
  const long long n = cidx / Q;
  const int j = (int)(cidx % Q);
  const float cj = cs[n * Q + j];
  float acc = 0.f;
  for (int i = j + lane; i < Q; i += 64)
    acc += bf2f(g[(n * Q + i) * Q + j]) * __expf(cs[n * Q + i] - cj);
  acc = wave_reduce_sum(acc);
  if (lane == 0) dcs[cidx] -= acc;
}

// ---------------------------------------------------------------------
// ssd_prep: one wave per (b, c, h) row.
//   dtf[n,q]  = softplus(dt[b, c*Q+q, h] + bias[h])
//   dacs[n,q] = cumsum_q(dtf * A[h]),  A = -exp(A_log)
// dt is a strided bf16 slice: element (b, l, h) at dt[(b*L+l)*sdt + h].
// ---------------------------------------------------------------------
__global__ void ssd_prep_fwd_kernel(const short* __restrict__ dt,
                                    const float* __restrict__ bias,
                                    const float* __restrict__ alog,
                                    float* __restrict__ dtf,
                                    float* __restrict__ dacs,
                                    int H, int Q, long long sdt,
                                    long long nrows) {
  const long long n = ((long long)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  if (n >= nrows) return;
  const int lane = threadIdx.x & 63;
  const int h = (int)(n % H);
  const long long bc = n / H;              // b*nc + c
  const float bh = bias[h];
  const float A = -__expf(alog[h]);
  const int per = Q / 64;                  // elems per lane (Q=128 -> 2)
  float v[4], s = 0.f;
#pragma unroll
  for (int e = 0; e < 4; ++e) {
    if (e < per) {
      const int q = lane * per + e;
      const float x = bf2f(dt[(bc * Q + q) * sdt + h]) + bh;
      // softplus with the standard overflow guard
      v[e] = x > 20.f ? x : log1pf(__expf(x));
      s += v[e];
    }
  }
  // inclusive wave scan of per-lane sums
  float sc = s;
#pragma unroll
  for (int d = 1; d < 64; d <<= 1) {
    const float o = __shfl_up(sc, d, 64);
    if (lane >= d) sc += o;
  }
  float run = (sc - s) * A;                // exclusive prefix * A
#pragma unroll
  for (int e = 0; e < 4; ++e) {
    if (e < per) {
      const int q = lane * per + e;
      dtf[n * Q + q] = v[e];
      run += v[e] * A;
      dacs[n * Q + q] = run;
    }
  }
}

// backward: d_dacs reverse-cumsum -> d_dA; d_dt, and atomics for
// d_bias / d_Alog. One wave per (b, c, h).
__global__ void ssd_prep_bwd_kernel(const float* __restrict__ ddtf,
                                    const float* __restrict__ ddacs,
                                    const short* __restrict__ dt,
                                    const float* __restrict__ bias,
                                    const float* __restrict__ alog,
                                    short* __restrict__ ddt,
                                    float* __restrict__ dbias,
                                    float* __restrict__ dalog,
                                    int H, int Q, long long sdt,
                                    long long sddt, long long nrows) {
  const long long n = ((long long)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  if (n >= nrows) return;
  const int lane = threadIdx.x & 63;
  const int h = (int)(n % H);
  const long long bc = n / H;
  const float bh = bias[h];
  const float A = -__expf(alog[h]);
  const int per = Q / 64;
  // reverse inclusive cumsum of ddacs: rc[q] = sum_{i >= q} ddacs[i]
  float g[4], s = 0.f;
#pragma unroll
  for (int e = 0; e < 4; ++e) {
    if (e < per) {
      g[e] = ddacs[n * Q + lane * per + e];
      s += g[e];
    }
  }
  float sc = s;   // suffix-sum via reversed-lane scan
#pragma unroll
  for (int d = 1; d < 64; d <<= 1) {
    const float o = __shfl_down(sc, d, 64);
    if (lane + d < 64) sc += o;
  }
  float suffix = sc - s;                   // strict suffix over later lanes
  float db = 0.f, da = 0.f;
#pragma unroll
  for (int e = 0; e < 4; ++e) {
    if (e < per) {
      const int q = lane * per + e;
      // rcq = sum_{i >= q} ddacs[i] = lane suffix + in-lane later + own
      float later = 0.f;
#pragma unroll
      for (int e2 = e + 1; e2 < 4; ++e2)
        if (e2 < per) later += g[e2];
      const float rcq = suffix + later + g[e];
      const float x = bf2f(dt[(bc * Q + q) * sdt + h]) + bh;
      const float vf = x > 20.f ? x : log1pf(__expf(x));
      const float dv = ddtf[n * Q + q] + rcq * A;   // d wrt dtf[q]
      const float ddt_q = dv * sigmoidf(x);
      ddt[(bc * Q + q) * sddt + h] = f2bf(ddt_q);
      db += ddt_q;
      da += rcq * vf * A;     // d_Alog: dA/dAlog = A (A = -exp(alog))
    }
  }
  db = wave_reduce_sum(db);
  da = wave_reduce_sum(da);
  if (lane == 0) {
    atomicAdd(dbias + h, db);
    atomicAdd(dalog + h, da);
  }
}

// ---------------------------------------------------------------------
// ssd_xdt: xdt[b,c,h,q,p]     = x[b, cQ+q, h*P+p] * dtf[n,q]   (bf16)
//          xdtd[b,c,h,q,p]    = xdt * exp(dacs[n,Q-1] - dacs[n,q])
// H-MAJOR layout: (b,nc,h,Q,p) is exactly torch.bmm's batch layout for
// y_diag = sL (N,Q,Q) @ xdt (N,Q,p) — the previous q-major layout made
// every downstream einsum materialize a permuted copy (~64 MB each).
// vectorized 8-wide over p (P % 8 == 0).
// ---------------------------------------------------------------------
__global__ void ssd_xdt_fwd_kernel(const short* __restrict__ x,
                                   const float* __restrict__ dtf,
                                   const float* __restrict__ dacs,
                                   short* __restrict__ xdt,
                                   short* __restrict__ xdtd,
                                   int H, int Q, int P, long long sx,
                                   long long total8) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= total8) return;
  const int P8 = P / 8;
  const int p0 = (int)(idx % P8) * 8;
  long long r = idx / P8;                  // (bc, h, q)
  const int q = (int)(r % Q);
  r /= Q;
  const int h = (int)(r % H);
  const long long bc = r / H;
  const long long n = bc * H + h;
  const float f = dtf[n * Q + q];
  const float dec = __expf(dacs[n * Q + Q - 1] - dacs[n * Q + q]);
  const bf16x8 xv = *(const bf16x8*)(x + (bc * Q + q) * sx + h * P + p0);
  bf16x8 o1, o2;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    const float xf = bf2f(xv.v[e]) * f;
    o1.v[e] = f2bf(xf);
    o2.v[e] = f2bf(xf * dec);
  }
  const long long out = (n * Q + q) * (long long)P + p0;
  *(bf16x8*)(xdt + out) = o1;
  *(bf16x8*)(xdtd + out) = o2;
}

// backward: one wave per (b,c,q,h): reduces over p.
//   dx   = (dxdt + dxdtd*dec) * dtf                       (bf16, strided)
//   ddtf = sum_p (dxdt + dxdtd*dec) * x
//   sdec[q]    = sum_p dxdtd * x * dtf * dec   (scratch; python folds
//                it into ddacs[q] and the ddacs[Q-1] end column — no
//                contended atomics)
__global__ void ssd_xdt_bwd_kernel(const short* __restrict__ dxdt,
                                   const short* __restrict__ dxdtd,
                                   const short* __restrict__ x,
                                   const float* __restrict__ dtf,
                                   const float* __restrict__ dacs,
                                   short* __restrict__ dx,
                                   float* __restrict__ ddtf,
                                   float* __restrict__ sdec_out,
                                   int H, int Q, int P, long long sx,
                                   long long sdx, long long nrows) {
  const long long r = ((long long)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  if (r >= nrows) return;
  const int lane = threadIdx.x & 63;
  const int q = (int)(r % Q);
  const long long bch = r / Q;
  const int h = (int)(bch % H);
  const long long bc = bch / H;
  const long long n = bch;
  const float f = dtf[n * Q + q];
  const float dec = __expf(dacs[n * Q + Q - 1] - dacs[n * Q + q]);
  const long long go = (n * Q + q) * (long long)P;   // h-major grads
  const long long xo = (bc * Q + q) * sx + h * P;
  const long long dxo = (bc * Q + q) * sdx + h * P;
  float sdtf = 0.f, sdec = 0.f;
  for (int p = lane; p < P; p += 64) {
    const float g1 = bf2f(dxdt[go + p]);
    const float g2 = bf2f(dxdtd[go + p]);
    const float xf = bf2f(x[xo + p]);
    const float gsum = g1 + g2 * dec;
    dx[dxo + p] = f2bf(gsum * f);
    sdtf += gsum * xf;
    sdec += g2 * xf * f * dec;   // d wrt (dacs_end - dacs_q)
  }
  sdtf = wave_reduce_sum(sdtf);
  sdec = wave_reduce_sum(sdec);
  if (lane == 0) {
    ddtf[n * Q + q] = sdtf;
    sdec_out[n * Q + q] = sdec;
  }
}

// ---------------------------------------------------------------------
// sL fused: sL[n,i,j] = scores[m,i,j] * exp(cs[n,i]-cs[n,j]) (j<=i)
// where n = bc*H + h and m = bc*G + h/(H/G) (per-GROUP scores).
// ---------------------------------------------------------------------
__global__ void ssd_sl_fwd_kernel(const float* __restrict__ cs,
                                  const short* __restrict__ scores,
                                  short* __restrict__ out,
                                  int H, int G, int Q, long long total8) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= total8) return;
  const int Q8 = Q / 8;
  const long long row = idx / Q8;          // (n, i)
  const int j0 = (int)(idx % Q8) * 8;
  const long long n = row / Q;
  const int i = (int)(row % Q);
  const long long m = (n / H) * G + (int)(n % H) / (H / G);
  const float ci = cs[n * Q + i];
  const f32x4 cj0 = *(const f32x4*)(cs + n * Q + j0);
  const f32x4 cj1 = *(const f32x4*)(cs + n * Q + j0 + 4);
  const bf16x8 sv = *(const bf16x8*)(scores + (m * Q + i) * Q + j0);
  bf16x8 o;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    const int j = j0 + e;
    const float cj = e < 4 ? cj0.v[e] : cj1.v[e - 4];
    o.v[e] = (j <= i) ? f2bf(bf2f(sv.v[e]) * __expf(ci - cj)) : (short)0;
  }
  *(bf16x8*)(out + row * Q + j0) = o;
}

// backward part 1: d_scores per HEAD (bf16; python sums over the group's
// heads) = g * L, L recomputed on the fly.
__global__ void ssd_sl_bwd_dscores_kernel(const short* __restrict__ g,
                                          const float* __restrict__ cs,
                                          short* __restrict__ dsh,
                                          int Q, long long total8) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= total8) return;
  const int Q8 = Q / 8;
  const long long row = idx / Q8;
  const int j0 = (int)(idx % Q8) * 8;
  const long long n = row / Q;
  const int i = (int)(row % Q);
  const float ci = cs[n * Q + i];
  const f32x4 cj0 = *(const f32x4*)(cs + n * Q + j0);
  const f32x4 cj1 = *(const f32x4*)(cs + n * Q + j0 + 4);
  const bf16x8 gv = *(const bf16x8*)(g + row * Q + j0);
  bf16x8 o;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    const int j = j0 + e;
    const float cj = e < 4 ? cj0.v[e] : cj1.v[e - 4];
    o.v[e] = (j <= i) ? f2bf(bf2f(gv.v[e]) * __expf(ci - cj)) : (short)0;
  }
  *(bf16x8*)(dsh + row * Q + j0) = o;
}

// backward part 2: d_cs rows/cols with the scores factor folded in:
// d_cs[n,k] = sum_j (g*s*L)[n,k,j] - sum_i (g*s*L)[n,i,k]
__global__ void ssd_sl_bwd_row_kernel(const short* __restrict__ g,
                                      const short* __restrict__ scores,
                                      const float* __restrict__ cs,
                                      float* __restrict__ dcs,
                                      int H, int G, int Q, long long rows) {
  const long long row = ((long long)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const long long n = row / Q;
  const int i = (int)(row % Q);
  const long long m = (n / H) * G + (int)(n % H) / (H / G);
  const float ci = cs[n * Q + i];
  float acc = 0.f;
  for (int j = lane; j <= i; j += 64)
    acc += bf2f(g[row * Q + j]) * bf2f(scores[(m * Q + i) * Q + j]) *
           __expf(ci - cs[n * Q + j]);
  acc = wave_reduce_sum(acc);
  if (lane == 0) dcs[row] += acc;
}

__global__ void ssd_sl_bwd_col_kernel(const short* __restrict__ g,
                                      const short* __restrict__ scores,
                                      const float* __restrict__ cs,
                                      float* __restrict__ dcs,
                                      int H, int G, int Q, long long cols) {
  const long long cidx = ((long long)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  if (cidx >= cols) return;
  const int lane = threadIdx.x & 63;
  const long long n = cidx / Q;
  const int j = (int)(cidx % Q);
  const long long m = (n / H) * G + (int)(n % H) / (H / G);
  const float cj = cs[n * Q + j];
  float acc = 0.f;
  for (int i = j + lane; i < Q; i += 64)
    acc += bf2f(g[(n * Q + i) * Q + j]) * bf2f(scores[(m * Q + i) * Q + j]) *
           __expf(cs[n * Q + i] - cj);
  acc = wave_reduce_sum(acc);
  if (lane == 0) dcs[cidx] -= acc;
}

// ---------------------------------------------------------------------
// ssd_ygate: y = ydiag + yoff*exp(dacs) + x*D;  out = y * silu(z)
// x and z are strided bf16 slices; out is (b, l, H*P) bf16 contiguous.
// ---------------------------------------------------------------------
// ydiag arrives H-MAJOR (b,nc,h,Q,p) (bmm-native), yoff arrives in the
// y_off einsum's natural output layout (b,nc,g,Q,rep,p) — reading each
// operand in its producer's layout removes the permute copies torch
// would otherwise materialize. out stays (b, l, H*P) row-major.
__global__ void ssd_ygate_fwd_kernel(const short* __restrict__ ydiag,
                                     const short* __restrict__ yoff,
                                     const float* __restrict__ dacs,
                                     const short* __restrict__ x,
                                     const float* __restrict__ Dp,
                                     const short* __restrict__ z,
                                     short* __restrict__ out,
                                     int H, int G, int Q, int P,
                                     long long sx, long long sz,
                                     long long total8) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= total8) return;
  const int P8 = P / 8;
  const int p0 = (int)(idx % P8) * 8;
  long long r = idx / P8;
  const int h = (int)(r % H);
  r /= H;
  const int q = (int)(r % Q);
  const long long bc = r / Q;
  const long long n = bc * H + h;
  const int rep = H / G;
  const float sd = __expf(dacs[n * Q + q]);
  const float Dh = Dp[h];
  const long long yd_o = (n * Q + q) * (long long)P + p0;
  const long long yo_o =
      ((((bc * G + h / rep) * Q + q) * rep + h % rep)) * (long long)P + p0;
  const bf16x8 yd = *(const bf16x8*)(ydiag + yd_o);
  const bf16x8 yo = *(const bf16x8*)(yoff + yo_o);
  const bf16x8 xv = *(const bf16x8*)(x + (bc * Q + q) * sx + h * P + p0);
  const bf16x8 zv = *(const bf16x8*)(z + (bc * Q + q) * sz + h * P + p0);
  bf16x8 o;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    const float y = bf2f(yd.v[e]) + bf2f(yo.v[e]) * sd + bf2f(xv.v[e]) * Dh;
    const float zf = bf2f(zv.v[e]);
    o.v[e] = f2bf(y * zf * sigmoidf(zf));
  }
  *(bf16x8*)(out + ((bc * Q + q) * (long long)H + h) * P + p0) = o;
}

// backward: one wave per (b,c,q,h) row, reduction over p. Each row owns
// a unique ddacs address (plain store); the per-head dD reduction goes
// through a (rows) scratch summed in python — a direct atomicAdd over
// only H addresses serialized 655k waves (measured 3.2 ms/call, 28%% of
// the mamba step).
__global__ void ssd_ygate_bwd_kernel(const short* __restrict__ dout,
                                     const short* __restrict__ ydiag,
                                     const short* __restrict__ yoff,
                                     const float* __restrict__ dacs,
                                     const short* __restrict__ x,
                                     const float* __restrict__ Dp,
                                     const short* __restrict__ z,
                                     short* __restrict__ dydiag,
                                     short* __restrict__ dyoff,
                                     float* __restrict__ ddacs,
                                     short* __restrict__ dx,
                                     float* __restrict__ dD_rows,
                                     short* __restrict__ dz,
                                     int H, int G, int Q, int P,
                                     long long sx, long long sz,
                                     long long sdx, long long sdz,
                                     long long nrows) {
  const long long r = ((long long)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  if (r >= nrows) return;
  const int lane = threadIdx.x & 63;
  const int h = (int)(r % H);
  const long long bcq = r / H;
  const int q = (int)(bcq % Q);
  const long long bc = bcq / Q;
  const long long n = bc * H + h;
  const int rep = H / G;
  const float sd = __expf(dacs[n * Q + q]);
  const float Dh = Dp[h];
  const long long go = ((bc * Q + q) * H + h) * (long long)P;   // out layout
  const long long yd_o = (n * Q + q) * (long long)P;            // h-major
  const long long yo_o =
      (((bc * G + h / rep) * Q + q) * rep + h % rep) * (long long)P;
  const long long xo = (bc * Q + q) * sx + h * P;
  const long long zo = (bc * Q + q) * sz + h * P;
  const long long dxo = (bc * Q + q) * sdx + h * P;
  const long long dzo = (bc * Q + q) * sdz + h * P;
  float sdd = 0.f, sD = 0.f;
  for (int p = lane; p < P; p += 64) {
    const float gy = bf2f(dout[go + p]);
    const float zf = bf2f(z[zo + p]);
    const float sg = sigmoidf(zf);
    const float sil = zf * sg;
    const float yofp = bf2f(yoff[yo_o + p]);
    const float xf = bf2f(x[xo + p]);
    const float y = bf2f(ydiag[yd_o + p]) + yofp * sd + xf * Dh;
    const float dy = gy * sil;
    dydiag[yd_o + p] = f2bf(dy);
    dyoff[yo_o + p] = f2bf(dy * sd);
    dx[dxo + p] = f2bf(dy * Dh);
    dz[dzo + p] = f2bf(gy * y * sg * (1.f + zf * (1.f - sg)));
    sdd += dy * yofp * sd;
    sD += dy * xf;
  }
  sdd = wave_reduce_sum(sdd);
  sD = wave_reduce_sum(sD);
  if (lane == 0) {
    ddacs[n * Q + q] = sdd;
    dD_rows[r] = sD;
  }
}

// ---------------------------------------------------------------------
// Fused sL backward for Q=128: one block per n computes, in a single
// read of g, (a) the per-head d_scores = g*L (bf16 out), and (b)
// d_cs[i] = rowsum(g*s*L)[i] - colsum(g*s*L)[i] via an LDS-staged
// product tile. Replaces the 3-pass version whose column pass read g
// with a Q-stride (412 us/call, the single largest mamba bwd kernel
// after the gate fix).
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void ssd_sl_bwd_fused_kernel(
    const short* __restrict__ g, const short* __restrict__ scores,
    const float* __restrict__ cs, short* __restrict__ dsh,
    float* __restrict__ dcs, int H, int G) {
  constexpr int Q = 128;
  __shared__ float prod[Q][Q + 5];   // g*s*L; +5 pad: row-sum lane stride 133 is coprime to the 64 banks
  __shared__ float csl[Q];
  __shared__ float rsum[Q], csum[Q];
  const long long n = blockIdx.x;
  const long long m = (n / H) * G + (int)(n % H) / (H / G);
  const int tid = threadIdx.x;
  if (tid < Q) csl[tid] = cs[n * Q + tid];
  __syncthreads();
  // stage: 256 threads x 64 elements, consecutive j per iteration
#pragma unroll
  for (int k = 0; k < Q * Q / 256; ++k) {
    const int idx = k * 256 + tid;
    const int i = idx >> 7;
    const int j = idx & (Q - 1);
    const float gv = bf2f(g[(n * Q + i) * Q + j]);
    const float L = (j <= i) ? __expf(csl[i] - csl[j]) : 0.f;
    dsh[(n * Q + i) * Q + j] = f2bf(gv * L);
    prod[i][j] = gv * L * bf2f(scores[(m * Q + i) * Q + j]);
  }
  __syncthreads();
  if (tid < Q) {           // row sums
    float a = 0.f;
    for (int j = 0; j <= tid; ++j) a += prod[tid][j];
    rsum[tid] = a;
  } else {                 // column sums (padded rows rotate banks)
    const int j = tid - Q;
    float a = 0.f;
    for (int i = j; i < Q; ++i) a += prod[i][j];
    csum[j] = a;
  }
  __syncthreads();
  if (tid < Q) dcs[n * Q + tid] += rsum[tid] - csum[tid];
}

extern "C" {

void launch_segsum_exp_fwd(const float* cs, void* out, long long N, int Q,
                           hipStream_t stream) {
  const long long total8 = N * Q * (Q / 8);
  const int block = 256;
  segsum_exp_fwd_kernel<<<(int)((total8 + block - 1) / block), block, 0,
                          stream>>>(cs, (short*)out, Q, total8);
}

void launch_segsum_exp_bwd(const void* g, const float* cs, float* dcs,
                           long long N, int Q, hipStream_t stream) {
  const long long rows = N * Q;
  const int block = 256;
  const int grid = (int)((rows * 64 + block - 1) / block);
  segsum_exp_bwd_row_kernel<<<grid, block, 0, stream>>>(
      (const short*)g, cs, dcs, Q, rows);
  segsum_exp_bwd_col_kernel<<<grid, block, 0, stream>>>(
      (const short*)g, cs, dcs, Q, rows);
}

void launch_ssd_prep_fwd(const void* dt, const float* bias, const float* alog,
                         float* dtf, float* dacs, long long nrows, int H,
                         int Q, long long sdt, hipStream_t stream) {
  const int block = 256;
  const int grid = (int)((nrows * 64 + block - 1) / block);
  ssd_prep_fwd_kernel<<<grid, block, 0, stream>>>(
      (const short*)dt, bias, alog, dtf, dacs, H, Q, sdt, nrows);
}

void launch_ssd_prep_bwd(const float* ddtf, const float* ddacs,
                         const void* dt, const float* bias, const float* alog,
                         void* ddt, float* dbias, float* dalog,
                         long long nrows, int H, int Q, long long sdt,
                         long long sddt, hipStream_t stream) {
  const int block = 256;
  const int grid = (int)((nrows * 64 + block - 1) / block);
  ssd_prep_bwd_kernel<<<grid, block, 0, stream>>>(
      ddtf, ddacs, (const short*)dt, bias, alog, (short*)ddt, dbias, dalog,
      H, Q, sdt, sddt, nrows);
}

void launch_ssd_xdt_fwd(const void* x, const float* dtf, const float* dacs,
                        void* xdt, void* xdtd, long long total8, int H,
                        int Q, int P, long long sx, hipStream_t stream) {
  const int block = 256;
  ssd_xdt_fwd_kernel<<<(int)((total8 + block - 1) / block), block, 0,
                       stream>>>((const short*)x, dtf, dacs, (short*)xdt,
                                 (short*)xdtd, H, Q, P, sx, total8);
}

void launch_ssd_xdt_bwd(const void* dxdt, const void* dxdtd, const void* x,
                        const float* dtf, const float* dacs, void* dx,
                        float* ddtf, float* ddacs, long long nrows, int H,
                        int Q, int P, long long sx, long long sdx,
                        hipStream_t stream) {
  const int block = 256;
  const int grid = (int)((nrows * 64 + block - 1) / block);
  ssd_xdt_bwd_kernel<<<grid, block, 0, stream>>>(
      (const short*)dxdt, (const short*)dxdtd, (const short*)x, dtf, dacs,
      (short*)dx, ddtf, ddacs, H, Q, P, sx, sdx, nrows);
}

void launch_ssd_sl_fwd(const float* cs, const void* scores, void* out,
                       long long total8, int H, int G, int Q,
                       hipStream_t stream) {
  const int block = 256;
  ssd_sl_fwd_kernel<<<(int)((total8 + block - 1) / block), block, 0,
                      stream>>>(cs, (const short*)scores, (short*)out, H, G,
                                Q, total8);
}

void launch_ssd_sl_bwd(const void* g, const void* scores, const float* cs,
                       void* dsh, float* dcs, long long N, int H, int G,
                       int Q, hipStream_t stream) {
  const int block = 256;
  if (Q == 128) {
    ssd_sl_bwd_fused_kernel<<<(int)N, block, 0, stream>>>(
        (const short*)g, (const short*)scores, cs, (short*)dsh, dcs, H, G);
    return;
  }
  const long long total8 = N * Q * (Q / 8);
  ssd_sl_bwd_dscores_kernel<<<(int)((total8 + block - 1) / block), block, 0,
                              stream>>>((const short*)g, cs, (short*)dsh, Q,
                                        total8);
  const long long rows = N * Q;
  const int grid = (int)((rows * 64 + block - 1) / block);
  ssd_sl_bwd_row_kernel<<<grid, block, 0, stream>>>(
      (const short*)g, (const short*)scores, cs, dcs, H, G, Q, rows);
  ssd_sl_bwd_col_kernel<<<grid, block, 0, stream>>>(
      (const short*)g, (const short*)scores, cs, dcs, H, G, Q, rows);
}

void launch_ssd_ygate_fwd(const void* ydiag, const void* yoff,
                          const float* dacs, const void* x, const float* Dp,
                          const void* z, void* out, long long total8, int H,
                          int G, int Q, int P, long long sx, long long sz,
                          hipStream_t stream) {
  const int block = 256;
  ssd_ygate_fwd_kernel<<<(int)((total8 + block - 1) / block), block, 0,
                         stream>>>((const short*)ydiag, (const short*)yoff,
                                   dacs, (const short*)x, Dp,
                                   (const short*)z, (short*)out, H, G, Q, P,
                                   sx, sz, total8);
}

void launch_ssd_ygate_bwd(const void* dout, const void* ydiag,
                          const void* yoff, const float* dacs, const void* x,
                          const float* Dp, const void* z, void* dydiag,
                          void* dyoff, float* ddacs, void* dx, float* dD,
                          void* dz, long long nrows, int H, int G, int Q,
                          int P, long long sx, long long sz, long long sdx,
                          long long sdz, hipStream_t stream) {
  const int block = 256;
  const int grid = (int)((nrows * 64 + block - 1) / block);
  ssd_ygate_bwd_kernel<<<grid, block, 0, stream>>>(
      (const short*)dout, (const short*)ydiag, (const short*)yoff, dacs,
      (const short*)x, Dp, (const short*)z, (short*)dydiag, (short*)dyoff,
      ddacs, (short*)dx, dD, (short*)dz, H, G, Q, P, sx, sz, sdx, sdz,
      nrows);
}

}  // extern "C"
