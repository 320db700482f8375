// Fused AdamW on flat fp32 master shards (+ bf16 or fp32 grads), and the
// multi-tensor sq-norm accumulator used by clip_grad_norm_.
// (AdamW step / grad-norm fused kernels; reference uses
// torch foreach AdamW, main_training_llama.py:113-115.)
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4v;
typedef __attribute__((ext_vector_type(4))) short s16x4v;



template <bool GRAD_BF16>
__global__ void adamw_kernel(float* __restrict__ p,
                             const void* __restrict__ g_,
                             float* __restrict__ m,
                             float* __restrict__ v,
                             short* __restrict__ p_bf16_out,
                             long long n4, float lr, float b1, float b2,
                             float eps, float wd, float bc1, float bc2,
                             const float* __restrict__ gscale) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n4) return;
  const float gs = gscale ? *gscale : 1.f;
  // every stream here is touched exactly once per step (189 GB/step on
  // a 7B shard): nontemporal hints keep them from thrashing L2
  f32x4v pv = __builtin_nontemporal_load((f32x4v*)p + i);
  f32x4v mv = __builtin_nontemporal_load((f32x4v*)m + i);
  f32x4v vv = __builtin_nontemporal_load((f32x4v*)v + i);
  float gf[4];
  if (GRAD_BF16) {
    const s16x4v g = __builtin_nontemporal_load((const s16x4v*)g_ + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) gf[j] = bf2f(g[j]) * gs;
  } else {
    const f32x4v g = __builtin_nontemporal_load((const f32x4v*)g_ + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) gf[j] = g[j] * gs;
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    float pj = pv[j] * (1.f - lr * wd);
    const float mj = b1 * mv[j] + (1.f - b1) * gf[j];
    const float vj = b2 * vv[j] + (1.f - b2) * gf[j] * gf[j];
    const float denom = sqrtf(vj / bc2) + eps;
    pj -= lr / bc1 * mj / denom;
    pv[j] = pj; mv[j] = mj; vv[j] = vj;
  }
  __builtin_nontemporal_store(pv, (f32x4v*)p + i);
  __builtin_nontemporal_store(mv, (f32x4v*)m + i);
  __builtin_nontemporal_store(vv, (f32x4v*)v + i);
  if (p_bf16_out != nullptr) {
    // publish the updated bf16 shard in the same pass (saves a separate
    // master->shard cast sweep: ~40 GB/step on a 7B model)
    s16x4v o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = f2bf(pv[j]);
    __builtin_nontemporal_store(o, (s16x4v*)p_bf16_out + i);
  }
}

template <bool BF16>
__global__ void sqnorm_kernel(const void* __restrict__ t, float* __restrict__ out,
                              long long n4) {
  float acc = 0.f;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (long long)gridDim.x * blockDim.x) {
    if (BF16) {
      const s16x4v v = __builtin_nontemporal_load((const s16x4v*)t + i);
#pragma unroll
      for (int j = 0; j < 4; ++j) { const float f = bf2f(v[j]); acc += f * f; }
    } else {
      const f32x4v v = __builtin_nontemporal_load((const f32x4v*)t + i);
#pragma unroll
      for (int j = 0; j < 4; ++j) acc += v[j] * v[j];
    }
  }
  __shared__ float scratch[16];
  acc = block_reduce_sum(acc, scratch);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

extern "C" {

void launch_adamw(float* p, const void* g, int grad_is_bf16, float* m,
                  float* v, void* p_bf16_out, long long n, float lr,
                  float b1, float b2, float eps, float wd, float bc1,
                  float bc2, const float* gscale, hipStream_t stream) {
  // flat shards are 128-element aligned; n % 4 == 0 guaranteed
  const long long n4 = n / 4;
  const int block = 256;
  const long long grid = (n4 + block - 1) / block;
  if (grad_is_bf16)
    adamw_kernel<true><<<(int)grid, block, 0, stream>>>(
        p, g, m, v, (short*)p_bf16_out, n4, lr, b1, b2, eps, wd, bc1, bc2,
        gscale);
  else
    adamw_kernel<false><<<(int)grid, block, 0, stream>>>(
        p, g, m, v, (short*)p_bf16_out, n4, lr, b1, b2, eps, wd, bc1, bc2,
        gscale);
}

void launch_sqnorm(const void* t, int is_bf16, float* out, long long n,
                   hipStream_t stream) {
  const long long n4 = n / 4;
  const int block = 256;
  const int grid = (int)min((long long)2048, (n4 + block - 1) / block);
  if (is_bf16)
    sqnorm_kernel<true><<<grid, block, 0, stream>>>(t, out, n4);
  else
    sqnorm_kernel<false><<<grid, block, 0, stream>>>(t, out, n4);
}

}  // extern "C"
