// Fused pieces of the Mamba2 SSD scan (SURVEY.md §2.3 selective-scan row).
// L = exp(segsum(cs)) materialized bf16 in ONE pass (the torch chain
// sub -> masked_fill -> exp -> cast costs ~4 full-tensor sweeps), plus
// the backward reductions d_cs = rowsum(g*L) - colsum(g*L) with L
// recomputed on the fly.
// cs (N, Q) fp32 cumulative sums; L (N, Q, Q) bf16 lower-triangular:
//   L[n,i,j] = exp(cs[n,i] - cs[n,j]) for j <= i else 0.
#include "common.h"

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

}  // extern "C"
