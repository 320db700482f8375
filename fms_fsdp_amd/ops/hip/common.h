// Common helpers for the CDNA4 (gfx950) kernel library.
// Wave size is 64 on CDNA (not 32); block sizes are multiples of 64.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

typedef __hip_bfloat16 bf16_t;

// 8 bf16 = 16 B per lane: the coalescing sweet spot for memory-bound
// kernels (guide G13: hipcc does not auto-vectorize bf16 loads).
struct alignas(16) bf16x8 {
  short v[8];
};

struct alignas(8) bf16x4 {
  short v[4];
};

struct alignas(16) f32x4 {
  float v[4];
};

__device__ __forceinline__ float bf2f(short u) {
  union { float f; unsigned int i; } c;
  c.i = ((unsigned int)(unsigned short)u) << 16;
  return c.f;
}

__device__ __forceinline__ short f2bf(float f) {
  // compiler-native RNE conversion: pairs fuse into v_cvt_pk_bf16_f32
  // (manual bit-math RNE costs ~4 VALU ops per element)
  __bf16 h = (__bf16)f;
  union { __bf16 b; short s; } c;
  c.b = h;
  return c.s;
}

// full-wave (64-lane) sum reduction
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x += __shfl_down(x, off, 64);
  return x;
}

__device__ __forceinline__ float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_down(x, off, 64));
  return x;
}

// block-level sum over up to 16 waves; result valid on every thread.
// `scratch` needs >= nwaves floats.
__device__ __forceinline__ float block_reduce_sum(float x, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = (blockDim.x + WAVE - 1) / WAVE;
  x = wave_reduce_sum(x);
  if (lane == 0) scratch[wid] = x;
  __syncthreads();
  float r = 0.f;
#pragma unroll 4
  for (int i = 0; i < nw; ++i) r += scratch[i];
  return r;
}

#define HIP_CHECK_LAUNCH()                                                   \
  do {                                                                       \
    hipError_t e = hipGetLastError();                                        \
    if (e != hipSuccess) {                                                   \
      printf("HIP launch error %s at %s:%d\n", hipGetErrorString(e),         \
             __FILE__, __LINE__);                                            \
    }                                                                        \
  } while (0)
