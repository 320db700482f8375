// Fused AdamW on flat fp32 master shards (+ bf16 / fp16 / fp32 grads), and
// the multi-tensor sq-norm accumulator used by clip_grad_norm_.
// (AdamW step / grad-norm fused kernels; reference uses
// torch foreach AdamW, main_training_llama.py:113-115. fp16 grads serve
// the fpSixteen mixed-precision policy, reference mixed_precision.py:5-9.)
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4v;
typedef __attribute__((ext_vector_type(4))) short s16x4v;

// dtype codes shared with bindings.cpp: 0 = fp32, 1 = bf16, 2 = fp16
__device__ __forceinline__ float h2f(short h) {
  _Float16 x = *reinterpret_cast<_Float16*>(&h);
  return (float)x;
}
__device__ __forceinline__ short f2h(float f) {
  _Float16 x = (_Float16)f;
  return *reinterpret_cast<short*>(&x);
}

template <int GDT, int ODT>
__global__ void adamw_kernel(float* __restrict__ p,
                             const void* __restrict__ g_,
                             float* __restrict__ m,
                             float* __restrict__ v,
                             short* __restrict__ p_lowp_out,
                             long long n4, float lr, float b1, float b2,
                             float eps, float wd, float bc1, float bc2,
                             const float* __restrict__ gscale) {
  // grid-stride: ~8 quads per thread on the pooled 7B shard keeps the
  // stream engines fed without a 1.7M-block launch
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n4; i += (long long)gridDim.x * blockDim.x) {
  const float gs = gscale ? *gscale : 1.f;
  // every stream here is touched exactly once per step (189 GB/step on
  // a 7B shard): nontemporal hints keep them from thrashing L2
  f32x4v pv = __builtin_nontemporal_load((f32x4v*)p + i);
  f32x4v mv = __builtin_nontemporal_load((f32x4v*)m + i);
  f32x4v vv = __builtin_nontemporal_load((f32x4v*)v + i);
  float gf[4];
  if (GDT == 1) {
    const s16x4v g = __builtin_nontemporal_load((const s16x4v*)g_ + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) gf[j] = bf2f(g[j]) * gs;
  } else if (GDT == 2) {
    const s16x4v g = __builtin_nontemporal_load((const s16x4v*)g_ + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) gf[j] = h2f(g[j]) * gs;
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
  if (ODT != 0 && p_lowp_out != nullptr) {
    // publish the updated low-precision shard in the same pass (saves a
    // separate master->shard cast sweep: ~40 GB/step on a 7B model)
    s16x4v o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = (ODT == 1) ? f2bf(pv[j]) : f2h(pv[j]);
    __builtin_nontemporal_store(o, (s16x4v*)p_lowp_out + i);
  }
  }
}

template <int DT>
__global__ void sqnorm_kernel(const void* __restrict__ t, float* __restrict__ out,
                              long long n4) {
  float acc = 0.f;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (long long)gridDim.x * blockDim.x) {
    if (DT == 1) {
      const s16x4v v = __builtin_nontemporal_load((const s16x4v*)t + i);
#pragma unroll
      for (int j = 0; j < 4; ++j) { const float f = bf2f(v[j]); acc += f * f; }
    } else if (DT == 2) {
      const s16x4v v = __builtin_nontemporal_load((const s16x4v*)t + i);
#pragma unroll
      for (int j = 0; j < 4; ++j) { const float f = h2f(v[j]); acc += f * f; }
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

void launch_adamw(float* p, const void* g, int grad_dt, float* m,
                  float* v, void* p_lowp_out, int out_dt, long long n,
                  float lr, float b1, float b2, float eps, float wd,
                  float bc1, float bc2, const float* gscale,
                  hipStream_t stream) {
  // flat shards are 128-element aligned; n % 4 == 0 guaranteed
  const long long n4 = n / 4;
  const int block = 256;
  const long long grid = (n4 + block - 1) / block;
#define ADAMW_CASE(GDT, ODT)                                                 \
  adamw_kernel<GDT, ODT><<<(int)grid, block, 0, stream>>>(                   \
      p, g, m, v, (short*)p_lowp_out, n4, lr, b1, b2, eps, wd, bc1, bc2,     \
      gscale)
  if (out_dt == 2)      { if (grad_dt == 2) ADAMW_CASE(2, 2); else if (grad_dt == 1) ADAMW_CASE(1, 2); else ADAMW_CASE(0, 2); }
  else if (out_dt == 1) { if (grad_dt == 2) ADAMW_CASE(2, 1); else if (grad_dt == 1) ADAMW_CASE(1, 1); else ADAMW_CASE(0, 1); }
  else                  { if (grad_dt == 2) ADAMW_CASE(2, 0); else if (grad_dt == 1) ADAMW_CASE(1, 0); else ADAMW_CASE(0, 0); }
#undef ADAMW_CASE
}

void launch_sqnorm(const void* t, int dt, float* out, long long n,
                   hipStream_t stream) {
  const long long n4 = n / 4;
  const int block = 256;
  const int grid = (int)min((long long)2048, (n4 + block - 1) / block);
  if (dt == 1)
    sqnorm_kernel<1><<<grid, block, 0, stream>>>(t, out, n4);
  else if (dt == 2)
    sqnorm_kernel<2><<<grid, block, 0, stream>>>(t, out, n4);
  else
    sqnorm_kernel<0><<<grid, block, 0, stream>>>(t, out, n4);
}

}  // extern "C"
