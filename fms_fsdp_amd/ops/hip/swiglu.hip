// SwiGLU epilogue: h = silu(g) * u over fused (N, 2H) gate|up layout,
// and its backward. Memory-bound; bf16x8 vectorized (guide G13).
// Replaces the torch.compile fusion the reference leans on
// (SwiGLU MLP epilogue).
#include "common.h"

__global__ void swiglu_fwd_kernel(const bf16x8* __restrict__ gu,
                                  bf16x8* __restrict__ h,
                                  int H8, long long rows) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long total = rows * H8;
  if (i >= total) return;
  const long long row = i / H8;
  const int col = (int)(i % H8);
  const bf16x8 g = gu[row * 2 * H8 + col];
  const bf16x8 u = gu[row * 2 * H8 + H8 + col];
  bf16x8 o;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const float gf = bf2f(g.v[j]);
    const float s = 1.f / (1.f + __expf(-gf));
    o.v[j] = f2bf(gf * s * bf2f(u.v[j]));
  }
  h[i] = o;
}

// dg = dy * u * s * (1 + g*(1-s)),  du = dy * g * s,  s = sigmoid(g)
__global__ void swiglu_bwd_kernel(const bf16x8* __restrict__ dy,
                                  const bf16x8* __restrict__ gu,
                                  bf16x8* __restrict__ dgu,
                                  int H8, long long rows) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long total = rows * H8;
  if (i >= total) return;
  const long long row = i / H8;
  const int col = (int)(i % H8);
  const bf16x8 g = gu[row * 2 * H8 + col];
  const bf16x8 u = gu[row * 2 * H8 + H8 + col];
  const bf16x8 d = dy[i];
  bf16x8 og, ou;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const float gf = bf2f(g.v[j]);
    const float uf = bf2f(u.v[j]);
    const float df = bf2f(d.v[j]);
    const float s = 1.f / (1.f + __expf(-gf));
    og.v[j] = f2bf(df * uf * s * (1.f + gf * (1.f - s)));
    ou.v[j] = f2bf(df * gf * s);
  }
  dgu[row * 2 * H8 + col] = og;
  dgu[row * 2 * H8 + H8 + col] = ou;
}

extern "C" {

void launch_swiglu_fwd(const void* gu, void* h, long long rows, int H,
                       hipStream_t stream) {
  long long total = rows * (H / 8);
  int block = 256;
  swiglu_fwd_kernel<<<(int)((total + block - 1) / block), block, 0, stream>>>(
      (const bf16x8*)gu, (bf16x8*)h, H / 8, rows);
}

void launch_swiglu_bwd(const void* dy, const void* gu, void* dgu,
                       long long rows, int H, hipStream_t stream) {
  long long total = rows * (H / 8);
  int block = 256;
  swiglu_bwd_kernel<<<(int)((total + block - 1) / block), block, 0, stream>>>(
      (const bf16x8*)dy, (const bf16x8*)gu, (bf16x8*)dgu, H / 8, rows);
}

}  // extern "C"
