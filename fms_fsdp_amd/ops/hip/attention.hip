// Causal flash attention (forward + backward) for CDNA4/gfx950.
// MFMA v_mfma_f32_32x32x16_bf16 tiles, LDS-staged K/V with XOR swizzle,
// swapped-QK^T structure (compute S^T = K @ Q^T so each lane owns ONE
// q-row and softmax is lane-local: guide §B "swapped QK^T ... row-reduce
// is 31 fmax + 1 permlane32_swap").
//
// Layouts (bf16, contiguous): q (b, s, h, d), k/v (b, s, kvh, d), d in
// {64, 128}; causal; GQA via kvh | h. lse (b, h, s) fp32 saved for bwd.
// Replaces the reference's torch-SDPA flash call (SURVEY.md §2.3:
// "SDPA FlashAttention-v2 fwd+bwd ... MFMA tiled flash kernel, LDS
// double-buffering, causal block skipping, GQA head-broadcast").
//
// MFMA fragment maps used throughout (guide §3, cdna4 32x32x16 bf16):
//   A[i][k]: lane l -> i = l&31,          k = (l>>5)*8 + e   (e=0..7)
//   B[k][j]: lane l -> k = (l>>5)*8 + e,  j = l&31
//   D[i][j]: lane l, reg r -> j = l&31,   i = (r&3) + 8*(r>>2) + 4*(l>>5)
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(16))) float f32x16;

union U8 {
  bf16x8v v;
  unsigned u[4];
  bf16x8 s;
};

__device__ __forceinline__ unsigned pack2(float lo, float hi) {
  return ((unsigned)(unsigned short)f2bf(hi) << 16) |
         (unsigned short)f2bf(lo);
}

// Build the PV B-operand for one 16-key chunk from 8 P registers
// (D-layout rows (r&3)+8*(r>>2)+4*hb -> B[k=key][j=q]): cvt_pk pairs +
// permlane32_swap half-exchange (guide T12 / §B P->PV layout).
__device__ __forceinline__ bf16x8v pack_pT_chunk(const float* p) {
  unsigned w01 = pack2(p[0], p[1]);
  unsigned w23 = pack2(p[2], p[3]);
  unsigned w89 = pack2(p[4], p[5]);
  unsigned w1011 = pack2(p[6], p[7]);
  auto r1 = __builtin_amdgcn_permlane32_swap(w01, w89, false, false);
  auto r2 = __builtin_amdgcn_permlane32_swap(w23, w1011, false, false);
  U8 b;
  b.u[0] = r1[0];
  b.u[1] = r2[0];
  b.u[2] = r1[1];
  b.u[3] = r2[1];
  return b.v;
}

// row index inside a 32-row D tile for register r, half hb
#define DROW(r, hb) (((r) & 3) + 8 * ((r) >> 2) + 4 * (hb))

// ---------------------------------------------------------------------
// Forward. Block = 4 waves x 32 q-rows = 128 q rows; KV tile = 64 keys.
// grid.x = s/128, grid.y = b*h.
// ---------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256, 2) void attn_fwd_kernel(
    const short* __restrict__ qg, const short* __restrict__ kg,
    const short* __restrict__ vg, short* __restrict__ og,
    float* __restrict__ lseg, int B, int S, int H, int KVH, float scale) {
  constexpr int KVB = 64;
  constexpr int KSWZ = (D == 128) ? 15 : 7;  // XOR stays inside a D*2-byte row
  constexpr int NC = D / 16;   // QK^T k-chunks
  constexpr int NT = D / 32;   // 32-wide output tiles
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_lds = smem;                    // [KVB][D*2] swizzled, 256B rows @D=128
  char* vt_lds = smem + KVB * D * 2;     // [D][KVB*2] transposed V, 128B rows

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int col = lane & 31;   // q-row owner within wave tile
  const int hb = lane >> 5;

  const int bh = blockIdx.y;
  const int b = bh / H;
  const int h = bh % H;
  const int kvh = h / (H / KVH);
  const int q0 = blockIdx.x * 128;        // block q range [q0, q0+128)
  const int qw = q0 + wid * 32;           // wave q range
  const int my_q = qw + col;              // this lane's q row

  const long long qrow_stride = (long long)H * D;
  const long long krow_stride = (long long)KVH * D;
  const short* qbase = qg + ((long long)b * S * H + (long long)h) * D;
  const short* kbase = kg + ((long long)b * S * KVH + (long long)kvh) * D;
  const short* vbase = vg + ((long long)b * S * KVH + (long long)kvh) * D;

  // Q -> B-fragments (registers, reused all tiles)
  bf16x8v qb[NC];
  {
    const short* qp = qbase + (long long)my_q * qrow_stride + hb * 8;
#pragma unroll
    for (int c = 0; c < NC; ++c)
      qb[c] = *(const bf16x8v*)(qp + c * 16);
  }

  f32x16 accO[NT];
#pragma unroll
  for (int t = 0; t < NT; ++t) accO[t] = (f32x16)(0.f);
  float m_run = -1e30f, l_run = 0.f;

  const int ntiles = (q0 + 128 + KVB - 1) / KVB;
  for (int tile = 0; tile < ntiles; ++tile) {
    const int kv0 = tile * KVB;
    // ---- cooperative stage: K row-major (swizzled), V transposed ----
    {
      // K: 256 threads, each copies KVB*D*2/256 bytes in 16B units
      const int t256 = threadIdx.x;
      constexpr int BYTES_PER_ROW = D * 2;
      constexpr int CHUNKS = KVB * BYTES_PER_ROW / 16;  // 16B chunks
#pragma unroll
      for (int i = t256; i < CHUNKS; i += 256) {
        const int row = i / (BYTES_PER_ROW / 16);
        const int cb = (i % (BYTES_PER_ROW / 16)) * 16;
        const long long g = (long long)(kv0 + row) * krow_stride + cb / 2;
        *(f32x4*)(k_lds + ((row * BYTES_PER_ROW + cb) ^ ((row & KSWZ) << 4))) =
            *(const f32x4*)(kbase + g);
      }
      // V transposed: thread t: key = t&63, dv block = (t>>6)*32.
      // Loads are vectorized 16B (guide G13: never scalar bf16 loads);
      // the transposed LDS writes scatter but stay cheap vs the MFMAs.
      const int key = t256 & 63;
      const int dv0 = (t256 >> 6) * (D / 4);
      const short* vp = vbase + (long long)(kv0 + key) * krow_stride + dv0;
#pragma unroll
      for (int jj = 0; jj < D / 32; ++jj) {
        const bf16x8v vv = *(const bf16x8v*)(vp + jj * 8);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int dv = dv0 + jj * 8 + e;
          *(__bf16*)(vt_lds +
                     ((dv * (KVB * 2) + key * 2) ^ ((dv & 7) << 4))) = vv[e];
        }
      }
    }
    __syncthreads();

    // ---- two 32-key sub-tiles, each with its own online-softmax pass.
    // Register economy: one live accS/p set (16 regs) instead of two,
    // keeping total VGPR+AGPR under 256 for 2 waves/SIMD occupancy.
    // defer-max (guide T13, THR=8): the O/l rescale runs only when the
    // sub-tile max exceeds the running max by more than THR; P is then
    // bounded by e^THR which the fp32 accumulate tolerates. Decision is
    // made BEFORE this sub-tile's P is exponentiated (the safe order).
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
      const int kv32 = kv0 + kt * 32;
      if (kv32 > qw + 31) break;      // wave-uniform causal tile skip
      f32x16 accS = (f32x16)(0.f);
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        const int row = kt * 32 + col;
        const int inrow = c * 32 + hb * 16;
        const bf16x8v a =
            *(const bf16x8v*)(k_lds + ((row * (D * 2) + inrow) ^ ((row & KSWZ) << 4)));
        accS = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, qb[c], accS, 0, 0, 0);
      }
      float p[16];
      const bool partial = (kv32 + 32) > (qw + 1);
      float mt = -1e30f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float sv = accS[r] * scale;
        if (partial && (kv32 + DROW(r, hb)) > my_q) sv = -1e30f;
        p[r] = sv;
        mt = fmaxf(mt, sv);
      }
      mt = fmaxf(mt, __shfl_xor(mt, 32, 64));
      float alpha = 1.f;
      if (mt > m_run + 8.f) {          // defer-max threshold
        alpha = __expf(m_run - mt);
        m_run = mt;
#pragma unroll
        for (int t = 0; t < NT; ++t)
#pragma unroll
          for (int r = 0; r < 16; ++r) accO[t][r] *= alpha;
      }
      float s_own = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p[r] = __expf(p[r] - m_run);
        s_own += p[r];
      }
      l_run = l_run * alpha + s_own + __shfl_xor(s_own, 32, 64);

      // PV for this sub-tile: keys kv32..kv32+31 = chunks 2kt, 2kt+1
      const bf16x8v pb0 = pack_pT_chunk(p);
      const bf16x8v pb1 = pack_pT_chunk(p + 8);
#pragma unroll
      for (int t = 0; t < NT; ++t) {
        const int row = t * 32 + col;   // dv row
        const int inrow0 = (kt * 2) * 32 + hb * 16;
        const bf16x8v a0 = *(const bf16x8v*)(
            vt_lds + ((row * (KVB * 2) + inrow0) ^ ((row & 7) << 4)));
        accO[t] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, pb0, accO[t], 0, 0, 0);
        const int inrow1 = (kt * 2 + 1) * 32 + hb * 16;
        const bf16x8v a1 = *(const bf16x8v*)(
            vt_lds + ((row * (KVB * 2) + inrow1) ^ ((row & 7) << 4)));
        accO[t] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, pb1, accO[t], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: o[q][dv] = accO^T / l ----
  const float inv_l = 1.f / l_run;
  short* op = og + ((long long)b * S * H + (long long)h) * D +
              (long long)my_q * qrow_stride;
#pragma unroll
  for (int t = 0; t < NT; ++t) {
#pragma unroll
    for (int g = 0; g < 4; ++g) {   // reg groups of 4 -> dv consecutive 4
      const int dv0 = t * 32 + 8 * g + 4 * hb;
      bf16x4 w;
#pragma unroll
      for (int j = 0; j < 4; ++j)
        w.v[j] = f2bf(accO[t][g * 4 + j] * inv_l);
      *(bf16x4*)(op + dv0) = w;
    }
  }
  if (hb == 0)
    lseg[((long long)bh) * S + my_q] = m_run + __logf(l_run);
}

// ---------------------------------------------------------------------
// Backward preprocess: delta[b,h,s] = rowsum(dO * O) fp32
// ---------------------------------------------------------------------
__global__ void attn_bwd_delta_kernel(const short* __restrict__ dog,
                                      const short* __restrict__ og,
                                      float* __restrict__ delta,
                                      int D, int S, int H, long long rows) {
  // one wave per row; input rows are (b, s, h) order, delta is (b, h, s)
  const long long row = ((long long)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const short* dop = dog + row * D;
  const short* op = og + row * D;
  float acc = 0.f;
  for (int i = lane * 2; i < D; i += 128) {
    acc += bf2f(dop[i]) * bf2f(op[i]) + bf2f(dop[i + 1]) * bf2f(op[i + 1]);
  }
  acc = wave_reduce_sum(acc);
  if (lane == 0) {
    const long long b = row / ((long long)S * H);
    const int si = (int)((row / H) % S);
    const int hi = (int)(row % H);
    delta[(b * H + hi) * (long long)S + si] = acc;
  }
}

// ---------------------------------------------------------------------
// Backward dq: grid like forward (q-tiles); recompute P and dP, then
// dq^T = K^T @ dS^T. No atomics: each wave owns its 32 q rows.
//   P^T  = exp(scale*K@Q^T - lse)          (A=K rows from k_lds)
//   dP^T = V @ dO^T                        (A=V rows from v_lds)
//   dS^T = P^T * (dP^T - delta) * scale
//   dq^T[dk][q] = sum_key K^T[dk][key] dS^T[key][q]
//                 (A=K^T from kt_lds, B=pack(dS^T))
// ---------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256) void attn_bwd_dq_kernel(
    const short* __restrict__ dog, const short* __restrict__ qg,
    const short* __restrict__ kg, const short* __restrict__ vg,
    const float* __restrict__ lseg, const float* __restrict__ deltag,
    short* __restrict__ dqg, int B, int S, int H, int KVH, float scale) {
  constexpr int KVB = 32;
  constexpr int KSWZ = (D == 128) ? 15 : 7;
  constexpr int NC = D / 16;
  constexpr int NT = D / 32;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_lds = smem;                       // [KVB][D*2] swizzled
  char* v_lds = smem + KVB * D * 2;         // [KVB][D*2] swizzled
  char* kt_lds = smem + 2 * KVB * D * 2;    // [D][KVB*2] transposed K

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int col = lane & 31;
  const int hb = lane >> 5;

  const int bh = blockIdx.y;
  const int b = bh / H;
  const int h = bh % H;
  const int kvh = h / (H / KVH);
  const int q0 = blockIdx.x * 128;
  const int qw = q0 + wid * 32;
  const int my_q = qw + col;

  const long long qrow_stride = (long long)H * D;
  const long long krow_stride = (long long)KVH * D;
  const short* qbase = qg + ((long long)b * S * H + (long long)h) * D;
  const short* dobase = dog + ((long long)b * S * H + (long long)h) * D;
  const short* kbase = kg + ((long long)b * S * KVH + (long long)kvh) * D;
  const short* vbase = vg + ((long long)b * S * KVH + (long long)kvh) * D;

  bf16x8v qb[NC], dob[NC];
  {
    const short* qp = qbase + (long long)my_q * qrow_stride + hb * 8;
    const short* dp = dobase + (long long)my_q * qrow_stride + hb * 8;
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      qb[c] = *(const bf16x8v*)(qp + c * 16);
      dob[c] = *(const bf16x8v*)(dp + c * 16);
    }
  }
  const float my_lse = lseg[((long long)bh) * S + my_q];
  const float my_delta = deltag[((long long)bh) * S + my_q];

  f32x16 accDQ[NT];
#pragma unroll
  for (int t = 0; t < NT; ++t) accDQ[t] = (f32x16)(0.f);

  const int ntiles = (q0 + 128 + KVB - 1) / KVB;
  for (int tile = 0; tile < ntiles; ++tile) {
    const int kv0 = tile * KVB;
    // stage K, V row-major (swizzled) + K transposed
    {
      const int t256 = threadIdx.x;
      constexpr int BPR = D * 2;
      constexpr int CHUNKS = KVB * BPR / 16;
#pragma unroll
      for (int i = t256; i < CHUNKS; i += 256) {
        const int row = i / (BPR / 16);
        const int cb = (i % (BPR / 16)) * 16;
        const long long g = (long long)(kv0 + row) * krow_stride + cb / 2;
        *(f32x4*)(k_lds + ((row * BPR + cb) ^ ((row & KSWZ) << 4))) =
            *(const f32x4*)(kbase + g);
        *(f32x4*)(v_lds + ((row * BPR + cb) ^ ((row & KSWZ) << 4))) =
            *(const f32x4*)(vbase + g);
      }
      // K transposed: thread t: key = t % KVB, dk group = t/KVB
      const int key = t256 % KVB;
      const int ng = 256 / KVB;              // thread groups over dk
      const int dk0 = (t256 / KVB) * (D / ng);
      const short* kp = kbase + (long long)(kv0 + key) * krow_stride + dk0;
#pragma unroll
      for (int jj = 0; jj < D / ng / 8; ++jj) {
        const bf16x8v kv8 = *(const bf16x8v*)(kp + jj * 8);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int dk = dk0 + jj * 8 + e;
          *(__bf16*)(kt_lds + ((dk * (KVB * 2) + key * 2) ^ ((dk & 3) << 4))) =
              kv8[e];
        }
      }
    }
    __syncthreads();

    // S^T and dP^T (one 32-key tile)
    f32x16 accS = (f32x16)(0.f), accDP = (f32x16)(0.f);
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      const int row = col;
      const int inrow = c * 32 + hb * 16;
      const bf16x8v ka =
          *(const bf16x8v*)(k_lds + ((row * (D * 2) + inrow) ^ ((row & KSWZ) << 4)));
      const bf16x8v va =
          *(const bf16x8v*)(v_lds + ((row * (D * 2) + inrow) ^ ((row & KSWZ) << 4)));
      accS = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qb[c], accS, 0, 0, 0);
      accDP = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va, dob[c], accDP, 0, 0, 0);
    }

    // dS^T = P * (dP - delta) * scale  (0 where masked)
    float ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int key = kv0 + DROW(r, hb);
      const float pval = __expf(accS[r] * scale - my_lse);
      float v = pval * (accDP[r] - my_delta) * scale;
      if (key > my_q) v = 0.f;
      ds[r] = v;
    }

    // dq^T += K^T @ dS^T
    bf16x8v dsb[2];
    dsb[0] = pack_pT_chunk(ds);
    dsb[1] = pack_pT_chunk(ds + 8);
#pragma unroll
    for (int t = 0; t < NT; ++t) {
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        const int row = t * 32 + col;   // dk row
        const int inrow = kc * 32 + hb * 16;
        const bf16x8v a = *(const bf16x8v*)(
            kt_lds + ((row * (KVB * 2) + inrow) ^ ((row & 3) << 4)));
        accDQ[t] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, dsb[kc], accDQ[t], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // write dq (bf16): lane owns q row my_q; dq^T D-layout: dk = DROW + 32t
  short* dqp = dqg + ((long long)b * S * H + (long long)h) * D +
               (long long)my_q * qrow_stride;
#pragma unroll
  for (int t = 0; t < NT; ++t) {
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      const int dk0 = t * 32 + 8 * g + 4 * hb;
      bf16x4 w;
#pragma unroll
      for (int j = 0; j < 4; ++j) w.v[j] = f2bf(accDQ[t][g * 4 + j]);
      *(bf16x4*)(dqp + dk0) = w;
    }
  }
}

// ---------------------------------------------------------------------
// Backward dk/dv: grid over KV tiles; each wave owns 32 keys, loops all
// q-tiles >= its kv tile (causal). dK/dV accumulated in registers; P and
// dS^T cross the lane<->reg transpose through a small LDS buffer.
//   dV[key][dv] = sum_q P^T[key][q] dO[q][dv]   (A=P from p_lds, B=dOt)
//   dK[key][dk] = sum_q dS^T[key][q] Q[q][dk]   (A=dS from p_lds, B=Qt)
// ---------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256, 2) void attn_bwd_dkv_kernel(
    const short* __restrict__ dog, const short* __restrict__ qg,
    const short* __restrict__ kg, const short* __restrict__ vg,
    const float* __restrict__ lseg, const float* __restrict__ deltag,
    float* __restrict__ dkg, float* __restrict__ dvg, int B, int S, int H,
    int KVH, float scale) {
  constexpr int KVB = 32;   // keys per wave; block = 4 waves = 128 keys
  constexpr int KSWZ = (D == 128) ? 15 : 7;
  constexpr int NC = D / 16;
  constexpr int NT = D / 32;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // per-wave carves
  char* k_lds = smem;                                  // [4][KVB][D*2]
  char* v_lds = smem + 4 * KVB * D * 2;                // [4][KVB][D*2]
  // ONE shared transpose buffer, staged with dO^T then re-staged with Q^T
  // each q-iteration: keeps total LDS at 80 KB -> 2 blocks/CU.
  char* t_lds = smem + 8 * KVB * D * 2;                // [D][32*2] shared
  char* p_lds = t_lds + D * 64;                        // [4][KVB][32*2]

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int col = lane & 31;
  const int hb = lane >> 5;

  const int bh = blockIdx.y;
  const int b = bh / H;
  const int h = bh % H;
  const int kvh = h / (H / KVH);
  const int ngrp = H / KVH;
  const int kv0 = blockIdx.x * 128 + wid * KVB;  // this wave's keys

  const long long qrow_stride = (long long)H * D;
  const long long krow_stride = (long long)KVH * D;
  const short* qbase = qg + ((long long)b * S * H + (long long)h) * D;
  const short* dobase = dog + ((long long)b * S * H + (long long)h) * D;
  const short* kbase = kg + ((long long)b * S * KVH + (long long)kvh) * D;
  const short* vbase = vg + ((long long)b * S * KVH + (long long)kvh) * D;

  char* my_k = k_lds + wid * KVB * D * 2;
  char* my_v = v_lds + wid * KVB * D * 2;
  char* my_p = p_lds + wid * KVB * 64;

  // stage this wave's K/V rows once (each lane 32 16B chunks / wave)
  {
    constexpr int BPR = D * 2;
    constexpr int CHUNKS = KVB * BPR / 16;
#pragma unroll
    for (int i = lane; i < CHUNKS; i += 64) {
      const int row = i / (BPR / 16);
      const int cb = (i % (BPR / 16)) * 16;
      const long long g = (long long)(kv0 + row) * krow_stride + cb / 2;
      *(f32x4*)(my_k + ((row * BPR + cb) ^ ((row & KSWZ) << 4))) =
          *(const f32x4*)(kbase + g);
      *(f32x4*)(my_v + ((row * BPR + cb) ^ ((row & KSWZ) << 4))) =
          *(const f32x4*)(vbase + g);
    }
  }

  f32x16 accDV[NT], accDK[NT];
#pragma unroll
  for (int t = 0; t < NT; ++t) {
    accDV[t] = (f32x16)(0.f);
    accDK[t] = (f32x16)(0.f);
  }

  // q tiles: causal => q >= kv0 of the BLOCK's first wave; all waves walk
  // the same q range (block-uniform barriers), masking handles the rest.
  const int q_start = (blockIdx.x * 128) / 32 * 32;
  for (int q0 = q_start; q0 < S; q0 += 32) {
    // stage dO^T (transposed, shared): thread t: q = t&31, dk grp t>>5
    {
      const int t256 = threadIdx.x;
      const int q = t256 & 31;
      const int ng = 256 / 32;                  // 8 groups over dk
      const int dk0 = (t256 >> 5) * (D / ng);
      const short* dp = dobase + (long long)(q0 + q) * qrow_stride + dk0;
#pragma unroll
      for (int jj = 0; jj < D / ng / 8; ++jj) {
        const bf16x8v dv8 = *(const bf16x8v*)(dp + jj * 8);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int dk = dk0 + jj * 8 + e;
          *(__bf16*)(t_lds + ((dk * 64 + q * 2) ^ ((dk & 3) << 4))) = dv8[e];
        }
      }
    }
    __syncthreads();

    // S^T, dP^T for (my 32 keys) x (32 q)
    f32x16 accS = (f32x16)(0.f), accDP = (f32x16)(0.f);
    // B operands: Q / dO rows q0+col, read from qt/dot (B[k=dk][j=q]):
    // lane: dk=(l>>5)*8+e, q=l&31 -> qt[dk][q] strided... use global regs:
    bf16x8v qb2[NC], dob2[NC];
    {
      const short* qp = qbase + (long long)(q0 + col) * qrow_stride + hb * 8;
      const short* dp = dobase + (long long)(q0 + col) * qrow_stride + hb * 8;
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        qb2[c] = *(const bf16x8v*)(qp + c * 16);
        dob2[c] = *(const bf16x8v*)(dp + c * 16);
      }
    }
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      const int row = col;
      const int inrow = c * 32 + hb * 16;
      const bf16x8v ka =
          *(const bf16x8v*)(my_k + ((row * (D * 2) + inrow) ^ ((row & KSWZ) << 4)));
      const bf16x8v va =
          *(const bf16x8v*)(my_v + ((row * (D * 2) + inrow) ^ ((row & KSWZ) << 4)));
      accS = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qb2[c], accS, 0, 0, 0);
      accDP = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va, dob2[c], accDP, 0, 0, 0);
    }

    // P^T and dS^T; write P to p_lds for the A-operand transpose
    const float lse_q = lseg[((long long)bh) * S + q0 + col];
    const float delta_q = deltag[((long long)bh) * S + q0 + col];
    float pv[16], ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int key = kv0 + DROW(r, hb);
      const int q = q0 + col;
      float pval = __expf(accS[r] * scale - lse_q);
      if (key > q) pval = 0.f;
      pv[r] = pval;
      ds[r] = pval * (accDP[r] - delta_q) * scale;
    }
    // p_lds layout [key][q], rows 64B, swizzle ((key&3)<<4) — write P
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int key = DROW(r, hb);
      *(short*)(my_p + ((key * 64 + col * 2) ^ ((key & 3) << 4))) = f2bf(pv[r]);
    }
    // (p_lds is wave-private: lgkmcnt ordering suffices, no barrier)
    // dV[key][dv] += P(A) @ dOt(B): A[i=key][k=q] from p_lds
    {
      bf16x8v pa[2];
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        const int row = col;  // key row
        const int inrow = kc * 32 + hb * 16;
        pa[kc] = *(const bf16x8v*)(my_p + ((row * 64 + inrow) ^ ((row & 3) << 4)));
      }
#pragma unroll
      for (int t = 0; t < NT; ++t) {
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          // B[k=q][j=dv] from dot_lds[dv... B read: lane: q=(l>>5)*8+e,
          // dv=l&31: dot_lds holds [dk][q] (transposed dO): B[k=q][j=dv]
          // = dO[q][dv] = dot_lds[dv][q] -> lane reads row dv=col... but
          // j=l&31 must be dv: row = t*32+col, q chunk = kc*32+hb*16
          const int row = t * 32 + col;
          const int inrow = kc * 32 + hb * 16;
          const bf16x8v bb = *(const bf16x8v*)(
              t_lds + ((row * 64 + inrow) ^ ((row & 3) << 4)));
          accDV[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[kc], bb,
                                                             accDV[t], 0, 0, 0);
        }
      }
    }
    __syncthreads();  // every wave done with the dO image
    // re-stage t_lds with Q^T; overwrite p_lds with dS^T
    {
      const int t256 = threadIdx.x;
      const int q = t256 & 31;
      const int ng = 256 / 32;
      const int dk0 = (t256 >> 5) * (D / ng);
      const short* qp = qbase + (long long)(q0 + q) * qrow_stride + dk0;
#pragma unroll
      for (int jj = 0; jj < D / ng / 8; ++jj) {
        const bf16x8v qv8 = *(const bf16x8v*)(qp + jj * 8);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int dk = dk0 + jj * 8 + e;
          *(__bf16*)(t_lds + ((dk * 64 + q * 2) ^ ((dk & 3) << 4))) = qv8[e];
        }
      }
    }
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int key = DROW(r, hb);
      *(short*)(my_p + ((key * 64 + col * 2) ^ ((key & 3) << 4))) = f2bf(ds[r]);
    }
    __syncthreads();
    {
      bf16x8v da[2];
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        const int row = col;
        const int inrow = kc * 32 + hb * 16;
        da[kc] = *(const bf16x8v*)(my_p + ((row * 64 + inrow) ^ ((row & 3) << 4)));
      }
#pragma unroll
      for (int t = 0; t < NT; ++t) {
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          const int row = t * 32 + col;
          const int inrow = kc * 32 + hb * 16;
          const bf16x8v bb = *(const bf16x8v*)(
              t_lds + ((row * 64 + inrow) ^ ((row & 3) << 4)));
          accDK[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da[kc], bb,
                                                             accDK[t], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // write dK/dV. D-layout: row i = key_rel = DROW(r,hb), col j = feature
  // = t*32 + (lane&31). GQA head-groups collide on (b, key, kvh) ->
  // atomicAdd fp32 when ngrp>1, plain add otherwise (buffers zeroed).
#pragma unroll
  for (int t = 0; t < NT; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int key_abs = kv0 + DROW(r, hb);
      const int feat = t * 32 + col;
      float* dkp = dkg + ((long long)b * S + key_abs) * (long long)(KVH * D) +
                   (long long)kvh * D;
      float* dvp = dvg + ((long long)b * S + key_abs) * (long long)(KVH * D) +
                   (long long)kvh * D;
      if (ngrp > 1) {
        atomicAdd(&dkp[feat], accDK[t][r]);
        atomicAdd(&dvp[feat], accDV[t][r]);
      } else {
        dkp[feat] += accDK[t][r];
        dvp[feat] += accDV[t][r];
      }
    }
  }
}

// ---------------------------------------------------------------------
extern "C" {

void launch_attn_fwd(const void* q, const void* k, const void* v, void* o,
                     float* lse, int B, int S, int H, int KVH, int D,
                     float scale, hipStream_t stream) {
  dim3 grid(S / 128, B * H);
  const int lds = 64 * D * 2 + D * 64 * 2;  // K + Vt
  if (D == 128)
    attn_fwd_kernel<128><<<grid, 256, lds, stream>>>(
        (const short*)q, (const short*)k, (const short*)v, (short*)o, lse, B,
        S, H, KVH, scale);
  else
    attn_fwd_kernel<64><<<grid, 256, lds, stream>>>(
        (const short*)q, (const short*)k, (const short*)v, (short*)o, lse, B,
        S, H, KVH, scale);
}

void launch_attn_bwd(const void* do_, const void* q, const void* k,
                     const void* v, const void* o, const float* lse, void* dq,
                     void* dk, void* dv, float* delta_ws, int B, int S, int H,
                     int KVH, int D, float scale, hipStream_t stream) {
  const long long rows = (long long)B * S * H;
  attn_bwd_delta_kernel<<<(int)((rows * 64 + 255) / 256), 256, 0, stream>>>(
      (const short*)do_, (const short*)o, delta_ws, D, S, H, rows);
  dim3 grid(S / 128, B * H);
  if (D == 128) {
    const int lds_dq = 2 * 32 * 128 * 2 + 128 * 64;   // k + v + kt
    attn_bwd_dq_kernel<128><<<grid, 256, lds_dq, stream>>>(
        (const short*)do_, (const short*)q, (const short*)k, (const short*)v,
        lse, delta_ws, (short*)dq, B, S, H, KVH, scale);
    const int lds_dkv = 8 * 32 * 128 * 2 + 128 * 64 + 4 * 32 * 64;
    attn_bwd_dkv_kernel<128><<<grid, 256, lds_dkv, stream>>>(
        (const short*)do_, (const short*)q, (const short*)k, (const short*)v,
        lse, delta_ws, (float*)dk, (float*)dv, B, S, H, KVH, scale);
  } else {
    const int lds_dq = 2 * 32 * 64 * 2 + 64 * 64;
    attn_bwd_dq_kernel<64><<<grid, 256, lds_dq, stream>>>(
        (const short*)do_, (const short*)q, (const short*)k, (const short*)v,
        lse, delta_ws, (short*)dq, B, S, H, KVH, scale);
    const int lds_dkv = 8 * 32 * 64 * 2 + 64 * 64 + 4 * 32 * 64;
    attn_bwd_dkv_kernel<64><<<grid, 256, lds_dkv, stream>>>(
        (const short*)do_, (const short*)q, (const short*)k, (const short*)v,
        lse, delta_ws, (float*)dk, (float*)dv, B, S, H, KVH, scale);
  }
}

}  // extern "C"
