// RMSNorm forward/backward — one workgroup per row, wave64 reductions,
// bf16x8 vectorized loads (memory-bound: target HBM roofline).
// Replaces the reference's fms LayerNormParameterized call sites
// (SURVEY.md §2.3 kernel table: RMSNorm fwd/bwd).
#include "common.h"

// ---------------- forward ----------------
// x (N, H) bf16, w (H) bf16 -> y (N, H) bf16, rinv (N) f32
// H assumed % 8 == 0 (true for every llama dim); 256 threads/block.
__global__ void rmsnorm_fwd_kernel(const bf16x8* __restrict__ x,
                                   const bf16x8* __restrict__ w,
                                   bf16x8* __restrict__ y,
                                   float* __restrict__ rinv,
                                   int H8, float eps, int rows) {
  __shared__ float scratch[16];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const bf16x8* xr = x + (size_t)row * H8;
    bf16x8* yr = y + (size_t)row * H8;
    float ss = 0.f;
    for (int i = threadIdx.x; i < H8; i += blockDim.x) {
      bf16x8 v = xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(v.v[j]);
        ss += f * f;
      }
    }
    ss = block_reduce_sum(ss, scratch);
    float r = rsqrtf(ss / (H8 * 8) + eps);
    if (threadIdx.x == 0) rinv[row] = r;
    for (int i = threadIdx.x; i < H8; i += blockDim.x) {
      bf16x8 v = xr[i];
      bf16x8 wv = w[i];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.v[j] = f2bf(bf2f(v.v[j]) * r * bf2f(wv.v[j]));
      yr[i] = o;
    }
    __syncthreads();  // scratch reuse across row iterations
  }
}

// ---------------- fused residual-add + RMSNorm forward ----------------
// s = x + res;  y = s / rms(s) * w.  Saves the separate elementwise-add
// HBM pass (SURVEY.md §2.3 "fused residual-add variant" / mamba
// fused_add_norm). Emits both y and s (the residual stream).
__global__ void add_rmsnorm_fwd_kernel(const bf16x8* __restrict__ x,
                                       const bf16x8* __restrict__ res,
                                       const bf16x8* __restrict__ w,
                                       bf16x8* __restrict__ y,
                                       bf16x8* __restrict__ s_out,
                                       float* __restrict__ rinv,
                                       int H8, float eps, int rows) {
  __shared__ float scratch[16];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const bf16x8* xr = x + (size_t)row * H8;
    const bf16x8* rr = res + (size_t)row * H8;
    bf16x8* yr = y + (size_t)row * H8;
    bf16x8* sr = s_out + (size_t)row * H8;
    float ss = 0.f;
    for (int i = threadIdx.x; i < H8; i += blockDim.x) {
      const bf16x8 xv = xr[i];
      const bf16x8 rv = rr[i];
      bf16x8 sv;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = bf2f(xv.v[j]) + bf2f(rv.v[j]);
        sv.v[j] = f2bf(f);
        const float fb = bf2f(sv.v[j]);  // rms of the bf16-rounded sum
        ss += fb * fb;
      }
      sr[i] = sv;
    }
    ss = block_reduce_sum(ss, scratch);
    const float r = rsqrtf(ss / (H8 * 8) + eps);
    if (threadIdx.x == 0) rinv[row] = r;
    for (int i = threadIdx.x; i < H8; i += blockDim.x) {
      const bf16x8 sv = sr[i];
      const bf16x8 wv = w[i];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.v[j] = f2bf(bf2f(sv.v[j]) * r * bf2f(wv.v[j]));
      yr[i] = o;
    }
    __syncthreads();
  }
}

// ---------------- backward ----------------
// dx_i = r*w_i*dy_i - (r^3/H) * x_i * sum_j(dy_j*w_j*x_j)
__global__ void rmsnorm_bwd_dx_kernel(const bf16x8* __restrict__ dy,
                                      const bf16x8* __restrict__ x,
                                      const bf16x8* __restrict__ w,
                                      const float* __restrict__ rinv,
                                      const bf16x8* __restrict__ dextra,
                                      bf16x8* __restrict__ dx,
                                      int H8, int rows) {
  __shared__ float scratch[16];
  const int H = H8 * 8;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const bf16x8* dyr = dy + (size_t)row * H8;
    const bf16x8* xr = x + (size_t)row * H8;
    bf16x8* dxr = dx + (size_t)row * H8;
    const float r = rinv[row];
    float s = 0.f;
    for (int i = threadIdx.x; i < H8; i += blockDim.x) {
      bf16x8 d = dyr[i], xv = xr[i], wv = w[i];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        s += bf2f(d.v[j]) * bf2f(wv.v[j]) * bf2f(xv.v[j]);
    }
    s = block_reduce_sum(s, scratch);
    const float c = r * r * r * s / H;
    for (int i = threadIdx.x; i < H8; i += blockDim.x) {
      bf16x8 d = dyr[i], xv = xr[i], wv = w[i];
      bf16x8 o;
      if (dextra != nullptr) {
        const bf16x8 de = dextra[(size_t)row * H8 + i];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o.v[j] = f2bf(r * bf2f(wv.v[j]) * bf2f(d.v[j]) - c * bf2f(xv.v[j])
                        + bf2f(de.v[j]));
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o.v[j] = f2bf(r * bf2f(wv.v[j]) * bf2f(d.v[j]) - c * bf2f(xv.v[j]));
      }
      dxr[i] = o;
    }
    __syncthreads();
  }
}

// dw_j = sum_rows dy[i,j] * x[i,j] * rinv[i]; column-parallel with a
// row-chunk grid dim + fp32 atomics (few hundred K atomics total).
__global__ void rmsnorm_bwd_dw_kernel(const short* __restrict__ dy,
                                      const short* __restrict__ x,
                                      const float* __restrict__ rinv,
                                      float* __restrict__ dw,
                                      int H, int rows, int rows_per_chunk) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= H) return;
  const int r0 = blockIdx.y * rows_per_chunk;
  const int r1 = min(r0 + rows_per_chunk, rows);
  float acc = 0.f;
  for (int i = r0; i < r1; ++i) {
    size_t off = (size_t)i * H + col;
    acc += bf2f(dy[off]) * bf2f(x[off]) * rinv[i];
  }
  atomicAdd(&dw[col], acc);
}

extern "C" {

void launch_rmsnorm_fwd(const void* x, const void* w, void* y, float* rinv,
                        int rows, int H, float eps, hipStream_t stream) {
  int grid = min(rows, 2048);
  rmsnorm_fwd_kernel<<<grid, 256, 0, stream>>>(
      (const bf16x8*)x, (const bf16x8*)w, (bf16x8*)y, rinv, H / 8, eps, rows);
}

void launch_add_rmsnorm_fwd(const void* x, const void* res, const void* w,
                            void* y, void* s_out, float* rinv, int rows,
                            int H, float eps, hipStream_t stream) {
  int grid = min(rows, 2048);
  add_rmsnorm_fwd_kernel<<<grid, 256, 0, stream>>>(
      (const bf16x8*)x, (const bf16x8*)res, (const bf16x8*)w, (bf16x8*)y,
      (bf16x8*)s_out, rinv, H / 8, eps, rows);
}

void launch_rmsnorm_bwd(const void* dy, const void* x, const void* w,
                        const float* rinv, const void* dextra, void* dx,
                        float* dw, int rows, int H, hipStream_t stream) {
  int grid = min(rows, 2048);
  rmsnorm_bwd_dx_kernel<<<grid, 256, 0, stream>>>(
      (const bf16x8*)dy, (const bf16x8*)x, (const bf16x8*)w, rinv,
      (const bf16x8*)dextra, (bf16x8*)dx, H / 8, rows);
  int rows_per_chunk = max(1, (rows + 63) / 64);
  dim3 g2((H + 255) / 256, (rows + rows_per_chunk - 1) / rows_per_chunk);
  rmsnorm_bwd_dw_kernel<<<g2, 256, 0, stream>>>(
      (const short*)dy, (const short*)x, rinv, dw, H, rows, rows_per_chunk);
}

}  // extern "C"
