// Fused causal depthwise conv1d + silu (mamba xBC conv, width<=4) and its
// backward. Memory-bound; bf16x8 vectorized channel access (guide G13).
// Replaces mamba_ssm's causal-conv1d .cu kernel.
// x (b, l, C) bf16 row-major; weight (C, W) bf16; bias (C) fp32-or-bf16.
//
// The W=4 path (every registry config) loads each thread's 8-channel
// weight block as FOUR bf16x8 vectors and the bias as two f32x4 — the
// original scalar per-tap loads (32 x 2B per thread per row) were
// issue-bound and ran ~12x off the HBM roofline (torch.profiler:
// ~300us per kernel for ~180 MB of traffic).
#include "common.h"

// per-thread 8-channel weight block: channels c0..c0+7, W taps each,
// stored row-major (c, wi) -> flat j*W+wi over 4 bf16x8 vectors
template <int W>
struct WBlock {
  bf16x8 v[(8 * W + 7) / 8];
  __device__ __forceinline__ void load(const short* w, int c0) {
#pragma unroll
    for (int k = 0; k < (8 * W + 7) / 8; ++k)
      v[k] = ((const bf16x8*)(w + (long long)c0 * W))[k];
  }
  __device__ __forceinline__ float at(int j, int wi) const {
    const int f = j * W + wi;
    return bf2f(v[f / 8].v[f % 8]);
  }
};

template <int W>
__global__ void cconv_fwd_kernel(const short* __restrict__ x,
                                 const short* __restrict__ w,
                                 const float* __restrict__ bias,
                                 short* __restrict__ y,
                                 int L, int C, long long total8) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total8) return;
  const int C8 = C / 8;
  const long long bt = i / C8;          // (b, t) flattened
  const int c0 = (int)(i % C8) * 8;
  const int t = (int)(bt % L);
  const long long row0 = bt - t;        // start of this sequence

  WBlock<W> wb;
  wb.load(w, c0);
  const f32x4 b0 = *(const f32x4*)(bias + c0);
  const f32x4 b1 = *(const f32x4*)(bias + c0 + 4);
  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = j < 4 ? b0.v[j] : b1.v[j - 4];
#pragma unroll
  for (int wi = 0; wi < W; ++wi) {
    const int ti = t - W + 1 + wi;
    if (ti < 0) continue;
    const bf16x8 xv = *(const bf16x8*)(x + (row0 + ti) * C + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      acc[j] += bf2f(xv.v[j]) * wb.at(j, wi);
  }
  bf16x8 o;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const float s = 1.f / (1.f + __expf(-acc[j]));
    o.v[j] = f2bf(acc[j] * s);
  }
  *(bf16x8*)(y + bt * C + c0) = o;
}

// pass 1: g[t,c] = dy[t,c] * dsilu(z[t,c]) with z recomputed
template <int W>
__global__ void cconv_bwd_g_kernel(const short* __restrict__ dy,
                                   const short* __restrict__ x,
                                   const short* __restrict__ w,
                                   const float* __restrict__ bias,
                                   short* __restrict__ g,
                                   int L, int C, long long total8) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total8) return;
  const int C8 = C / 8;
  const long long bt = i / C8;
  const int c0 = (int)(i % C8) * 8;
  const int t = (int)(bt % L);
  const long long row0 = bt - t;
  WBlock<W> wb;
  wb.load(w, c0);
  const f32x4 b0 = *(const f32x4*)(bias + c0);
  const f32x4 b1 = *(const f32x4*)(bias + c0 + 4);
  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = j < 4 ? b0.v[j] : b1.v[j - 4];
#pragma unroll
  for (int wi = 0; wi < W; ++wi) {
    const int ti = t - W + 1 + wi;
    if (ti < 0) continue;
    const bf16x8 xv = *(const bf16x8*)(x + (row0 + ti) * C + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      acc[j] += bf2f(xv.v[j]) * wb.at(j, wi);
  }
  const bf16x8 d = *(const bf16x8*)(dy + bt * C + c0);
  bf16x8 o;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const float z = acc[j];
    const float s = 1.f / (1.f + __expf(-z));
    o.v[j] = f2bf(bf2f(d.v[j]) * s * (1.f + z * (1.f - s)));
  }
  *(bf16x8*)(g + bt * C + c0) = o;
}

// pass 2: dx[t,c] = sum_i g[t + (W-1) - i, c] * w[c, i]
template <int W>
__global__ void cconv_bwd_dx_kernel(const short* __restrict__ g,
                                    const short* __restrict__ w,
                                    short* __restrict__ dx,
                                    int L, int C, long long total8) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total8) return;
  const int C8 = C / 8;
  const long long bt = i / C8;
  const int c0 = (int)(i % C8) * 8;
  const int t = (int)(bt % L);
  const long long row0 = bt - t;
  WBlock<W> wb;
  wb.load(w, c0);
  float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int wi = 0; wi < W; ++wi) {
    const int ti = t + (W - 1) - wi;
    if (ti >= L) continue;
    const bf16x8 gv = *(const bf16x8*)(g + (row0 + ti) * C + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      acc[j] += bf2f(gv.v[j]) * wb.at(j, wi);
  }
  bf16x8 o;
#pragma unroll
  for (int j = 0; j < 8; ++j) o.v[j] = f2bf(acc[j]);
  *(bf16x8*)(dx + bt * C + c0) = o;
}

// pass 3: dw[c,i] += sum_{b,t} g[t,c] x[t-W+1+i,c]; db[c] += sum g[t,c]
__global__ void cconv_bwd_dwdb_kernel(const short* __restrict__ g,
                                      const short* __restrict__ x,
                                      float* __restrict__ dw,
                                      float* __restrict__ db,
                                      int L, int C, int W,
                                      long long rows, int rows_per_chunk) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const long long r0 = (long long)blockIdx.y * rows_per_chunk;
  const long long r1 = min(r0 + rows_per_chunk, rows);
  float accw[4] = {0.f, 0.f, 0.f, 0.f};  // W <= 4
  float accb = 0.f;
  for (long long bt = r0; bt < r1; ++bt) {
    const int t = (int)(bt % L);
    const long long row0 = bt - t;
    const float gv = bf2f(g[bt * C + c]);
    accb += gv;
    for (int wi = 0; wi < W; ++wi) {
      const int ti = t - W + 1 + wi;
      if (ti >= 0) accw[wi] += gv * bf2f(x[(row0 + ti) * C + c]);
    }
  }
  for (int wi = 0; wi < W; ++wi)
    atomicAdd(&dw[(long long)c * W + wi], accw[wi]);
  atomicAdd(&db[c], accb);
}

extern "C" {

void launch_cconv_fwd(const void* x, const void* w, const float* bias,
                      void* y, int BL, int L, int C, int W,
                      hipStream_t stream) {
  const long long total8 = (long long)BL * (C / 8);
  const int block = 256;
  const int grid = (int)((total8 + block - 1) / block);
  if (W == 4)
    cconv_fwd_kernel<4><<<grid, block, 0, stream>>>(
        (const short*)x, (const short*)w, bias, (short*)y, L, C, total8);
  else if (W == 3)
    cconv_fwd_kernel<3><<<grid, block, 0, stream>>>(
        (const short*)x, (const short*)w, bias, (short*)y, L, C, total8);
  else
    cconv_fwd_kernel<2><<<grid, block, 0, stream>>>(
        (const short*)x, (const short*)w, bias, (short*)y, L, C, total8);
}

void launch_cconv_bwd(const void* dy, const void* x, const void* w,
                      const float* bias, void* g, void* dx, float* dw,
                      float* db, int BL, int L, int C, int W,
                      hipStream_t stream) {
  const long long total8 = (long long)BL * (C / 8);
  const int block = 256;
  const int grid = (int)((total8 + block - 1) / block);
  if (W == 4) {
    cconv_bwd_g_kernel<4><<<grid, block, 0, stream>>>(
        (const short*)dy, (const short*)x, (const short*)w, bias, (short*)g,
        L, C, total8);
    cconv_bwd_dx_kernel<4><<<grid, block, 0, stream>>>(
        (const short*)g, (const short*)w, (short*)dx, L, C, total8);
  } else if (W == 3) {
    cconv_bwd_g_kernel<3><<<grid, block, 0, stream>>>(
        (const short*)dy, (const short*)x, (const short*)w, bias, (short*)g,
        L, C, total8);
    cconv_bwd_dx_kernel<3><<<grid, block, 0, stream>>>(
        (const short*)g, (const short*)w, (short*)dx, L, C, total8);
  } else {
    cconv_bwd_g_kernel<2><<<grid, block, 0, stream>>>(
        (const short*)dy, (const short*)x, (const short*)w, bias, (short*)g,
        L, C, total8);
    cconv_bwd_dx_kernel<2><<<grid, block, 0, stream>>>(
        (const short*)g, (const short*)w, (short*)dx, L, C, total8);
  }
  const int rpc = max(1, (int)((BL + 63) / 64));
  dim3 g2((C + 255) / 256, (BL + rpc - 1) / rpc);
  cconv_bwd_dwdb_kernel<<<g2, 256, 0, stream>>>(
      (const short*)g, (const short*)x, dw, db, L, C, W, BL, rpc);
}

}  // extern "C"
