// RoPE apply (half-rotation convention), forward and adjoint (conj).
// q (b,s,h,d), k (b,s,kvh,d) bf16; cos/sin tables (s, d/2) fp32 host-built
// (guide App-B: precompute trig on host; on-device sinf/cosf turns a
// memory-bound op VALU-bound). conj=true applies the inverse rotation
// (the backward pass).
#include "common.h"

// One thread handles 4 rotation pairs: loads short4 from each half.
// in/out may be strided views at the (b, s) row level (slices of a fused
// qkv projection / fused dqkv grad buffer): in_stride / out_stride are
// element strides between successive (b, s) rows; heads within a row are
// contiguous.
__global__ void rope_kernel(const short* __restrict__ in,
                            short* __restrict__ out,
                            const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t,
                            int s, int heads, int d, int conj,
                            long long in_stride, long long out_stride,
                            long long total4) {
  const int d2 = d / 2;
  const long long idx4 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx4 >= total4) return;
  // idx4 indexes groups of 4 pairs: layout (b, s, h, d2/4)
  const int g_per_head = d2 / 4;
  const long long head_idx = idx4 / g_per_head;   // (b*s*heads + ...)
  const int g = (int)(idx4 % g_per_head);
  const long long row = head_idx / heads;
  const int hh = (int)(head_idx % heads);
  const int si = (int)(row % s);
  const long long hoff = (long long)hh * d + (long long)g * 4;
  const long long base = row * in_stride + hoff;
  const long long obase = row * out_stride + hoff;

  const bf16x4 x1 = *(const bf16x4*)(in + base);
  const bf16x4 x2 = *(const bf16x4*)(in + base + d2);
  const f32x4 c = *(const f32x4*)(cos_t + (size_t)si * d2 + g * 4);
  const f32x4 sn = *(const f32x4*)(sin_t + (size_t)si * d2 + g * 4);
  bf16x4 o1, o2;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const float a = bf2f(x1.v[j]);
    const float b = bf2f(x2.v[j]);
    const float sj = conj ? -sn.v[j] : sn.v[j];
    o1.v[j] = f2bf(a * c.v[j] - b * sj);
    o2.v[j] = f2bf(b * c.v[j] + a * sj);
  }
  *(bf16x4*)(out + obase) = o1;
  *(bf16x4*)(out + obase + d2) = o2;
}

extern "C" {

void launch_rope(const void* in, void* out, const float* cos_t,
                 const float* sin_t, int b, int s, int heads, int d,
                 int conj, long long in_stride, long long out_stride,
                 hipStream_t stream) {
  long long total4 = (long long)b * s * heads * (d / 2) / 4;
  int block = 256;
  long long grid = (total4 + block - 1) / block;
  rope_kernel<<<(int)grid, block, 0, stream>>>(
      (const short*)in, (short*)out, cos_t, sin_t, s, heads, d, conj,
      in_stride, out_stride, total4);
}

}  // extern "C"
