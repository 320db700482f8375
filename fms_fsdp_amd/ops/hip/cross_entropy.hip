// Fused chunked cross-entropy: in-place logits -> dlogits transform plus
// loss accumulation, one workgroup per row (online max+sumexp, single
// re-read for the dlogit write). The full fp32 softmax is never
// materialized (SURVEY.md hard-part 6; reference leans on `del output`,
// train_utils.py:92-93).
// logits (n, V) bf16 INOUT; labels (n) i64; loss_sum f32 scalar accum;
// dlogits = (softmax - onehot) / denom  (0 for ignored rows).
#include "common.h"

__global__ void ce_fwd_bwd_kernel(short* __restrict__ logits,
                                  const long long* __restrict__ labels,
                                  float* __restrict__ loss_sum,
                                  const float* __restrict__ denom_ptr,
                                  long long ignore_index,
                                  int V, int rows) {
  __shared__ float s_m, s_l;
  const int V8 = V / 8;
  const int tail = V - V8 * 8;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    short* lr = logits + (size_t)row * V;
    const long long lab = labels[row];
    if (lab == ignore_index) {
      // zero dlogits for ignored rows
      for (int i = threadIdx.x; i < V8; i += blockDim.x)
        ((bf16x8*)lr)[i] = bf16x8{};
      for (int i = threadIdx.x; i < tail; i += blockDim.x)
        lr[V8 * 8 + i] = 0;
      __syncthreads();
      continue;
    }
    // pass 1: online max + sumexp
    float m = -1e30f, l = 0.f;
    for (int i = threadIdx.x; i < V8; i += blockDim.x) {
      bf16x8 v = ((const bf16x8*)lr)[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = bf2f(v.v[j]);
        if (f > m) {
          l *= __expf(m - f);
          m = f;
        }
        l += __expf(f - m);
      }
    }
    for (int i = threadIdx.x; i < tail; i += blockDim.x) {
      const float f = bf2f(lr[V8 * 8 + i]);
      if (f > m) { l *= __expf(m - f); m = f; }
      l += __expf(f - m);
    }
    // reduce (m, l) across the block: l_total at global max
    {
      const int lane = threadIdx.x & (WAVE - 1);
      const int wid = threadIdx.x / WAVE;
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) {
        const float mo = __shfl_down(m, off, 64);
        const float lo = __shfl_down(l, off, 64);
        if (mo > m) { l = l * __expf(m - mo) + lo; m = mo; }
        else        { l = l + lo * __expf(mo - m); }
      }
      __shared__ float sm[16], sl[16];
      if (lane == 0) { sm[wid] = m; sl[wid] = l; }
      __syncthreads();
      if (threadIdx.x == 0) {
        const int nw = (blockDim.x + WAVE - 1) / WAVE;
        float M = sm[0], L = sl[0];
        for (int i = 1; i < nw; ++i) {
          if (sm[i] > M) { L = L * __expf(M - sm[i]) + sl[i]; M = sm[i]; }
          else           { L = L + sl[i] * __expf(sm[i] - M); }
        }
        s_m = M; s_l = L;
        const float lse = M + __logf(L);
        atomicAdd(loss_sum, lse - bf2f(lr[lab]));
      }
      __syncthreads();
    }
    const float M = s_m;
    const float inv_l = 1.f / s_l;
    const float inv_denom = 1.f / denom_ptr[0];
    // pass 2: dlogits = (softmax - onehot)/denom, written in place
    for (int i = threadIdx.x; i < V8; i += blockDim.x) {
      bf16x8 v = ((const bf16x8*)lr)[i];
      bf16x8 o;
      const long long base = (long long)i * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float p = __expf(bf2f(v.v[j]) - M) * inv_l;
        if (base + j == lab) p -= 1.f;
        o.v[j] = f2bf(p * inv_denom);
      }
      ((bf16x8*)lr)[i] = o;
    }
    for (int i = threadIdx.x; i < tail; i += blockDim.x) {
      const long long col = V8 * 8 + i;
      float p = __expf(bf2f(lr[col]) - M) * inv_l;
      if (col == lab) p -= 1.f;
      lr[col] = f2bf(p * inv_denom);
    }
    __syncthreads();
  }
}

extern "C" {

void launch_ce_fwd_bwd(void* logits, const long long* labels, float* loss_sum,
                       const float* denom_ptr, long long ignore_index, int rows, int V,
                       hipStream_t stream) {
  int grid = min(rows, 2048);
  ce_fwd_bwd_kernel<<<grid, 256, 0, stream>>>(
      (short*)logits, labels, loss_sum, denom_ptr, ignore_index, V, rows);
}

}  // extern "C"
