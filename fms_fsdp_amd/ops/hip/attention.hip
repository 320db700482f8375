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
// Shared helpers for the attention kernels: SUBTILED LDS images.
// A [R rows][D cols] bf16 tile is stored as [R/4][D/16][4][16] (each
// subtile 128 B contiguous). This single layout serves BOTH fragment
// patterns:
//  - plain B/A reads (row = lane&31, 8 consecutive cols): 16 B contiguous
//  - transposed reads via gfx950 ds_read_b64_tr_b16: a 16-lane group's
//    lane j at subtile_base + j*8 receives COLUMN j of the 4x16 subtile
//    (semantics verified on hardware: tools/tr_probe.hip).
// ---------------------------------------------------------------------
// column-block-major subtiles: tile = colblk*R4 + rowgrp (R4 = image
// rows / 4). A plain 16-lane-group read then touches 4 CONSECUTIVE
// tiles (alternating LDS bank halves); the rowgrp-bit1 16B XOR spreads
// the remaining overlap. (PMC-driven: row-group-major was 4-way.)
#define SUBT_OFF(row, col, R4) \
  ((((col) >> 4) * (R4) + ((row) >> 2)) * 128 + \
   ((((row) & 3) * 32 + ((col) & 15) * 2) ^ ((((row) >> 2) & 2) << 3)))

union U2x64 {
  unsigned long long u[2];
  bf16x8v v;
};

// Two transpose reads batched behind ONE lgkmcnt wait (each read's 4
// shorts = column lane&15 of the 4x16 subtile at its address).
__device__ __forceinline__ bf16x8v tr_read2(const char* base, int off0,
                                            int off1) {
  U2x64 r;
  const unsigned a0 = (unsigned)(unsigned long long)(base + off0);
  const unsigned a1 = (unsigned)(unsigned long long)(base + off1);
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=&v"(r.u[0]), "=&v"(r.u[1])
      : "v"(a0), "v"(a1)
      : "memory");
  return r.v;
}

// Four transpose reads -> TWO A-fragments behind ONE lgkmcnt wait: the
// LDS pipe streams all four b64 reads before the single serialization
// point, so back-to-back mfmas consume both fragments densely.
struct TR4 {
  bf16x8v a, b;
};
__device__ __forceinline__ TR4 tr_read4(const char* base, int off0, int off1,
                                        int off2, int off3) {
  U2x64 r0, r1;
  const unsigned a0 = (unsigned)(unsigned long long)(base + off0);
  const unsigned a1 = (unsigned)(unsigned long long)(base + off1);
  const unsigned a2 = (unsigned)(unsigned long long)(base + off2);
  const unsigned a3 = (unsigned)(unsigned long long)(base + off3);
  asm volatile(
      "ds_read_b64_tr_b16 %0, %4\n\t"
      "ds_read_b64_tr_b16 %1, %5\n\t"
      "ds_read_b64_tr_b16 %2, %6\n\t"
      "ds_read_b64_tr_b16 %3, %7\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=&v"(r0.u[0]), "=&v"(r0.u[1]), "=&v"(r1.u[0]), "=&v"(r1.u[1])
      : "v"(a0), "v"(a1), "v"(a2), "v"(a3)
      : "memory");
  TR4 out;
  out.a = r0.v;
  out.b = r1.v;
  return out;
}

// ---- immediate-offset, software-pipelined transpose reads ------------
// The 16 tr-read addresses of a PV/dq/dkv output loop share ONE runtime
// base (swizzle ^ half-select ^ 16-lane-block, all lane-derived); the
// remaining terms are compile-time. Issuing through `offset:` immediates
// removes the per-read VALU address chain (PMC round-1: VALU is the
// limiter, profiles/attn_pmc_counters_r01.md), and splitting issue from
// wait lets fragment t+1 stream from LDS while fragment t feeds mfmas.
// Counted waits are safe here because DS ops retire IN ORDER: "allow 4
// outstanding" implies everything older (incl. staging ds_writes) is
// done. No SMEM is issued inside these windows.
template <int O0, int O1, int O2, int O3>
__device__ __forceinline__ void tr_issue4(unsigned base, U2x64& r0,
                                          U2x64& r1) {
  asm volatile(
      "ds_read_b64_tr_b16 %0, %4 offset:%c5\n\t"
      "ds_read_b64_tr_b16 %1, %4 offset:%c6\n\t"
      "ds_read_b64_tr_b16 %2, %4 offset:%c7\n\t"
      "ds_read_b64_tr_b16 %3, %4 offset:%c8"
      : "=&v"(r0.u[0]), "=&v"(r0.u[1]), "=&v"(r1.u[0]), "=&v"(r1.u[1])
      : "v"(base), "i"(O0), "i"(O1), "i"(O2), "i"(O3)
      : "memory");
}

template <int CNT>
__device__ __forceinline__ void tr_wait(U2x64& r0, U2x64& r1) {
  asm volatile("s_waitcnt lgkmcnt(%c4)"
               : "+v"(r0.u[0]), "+v"(r0.u[1]), "+v"(r1.u[0]), "+v"(r1.u[1])
               : "i"(CNT)
               : "memory");
}

// lane-derived base for the subtiled-image transpose reads: row-group
// XOR (constant 16*hb for every fragment of these loops), the in-subtile
// lane column, the hb*8 key offset (rg = hb*2 -> 256*hb bytes) and the
// 16-lane block select. R4B = R4*128 bytes per column block.
template <int R4B>
__device__ __forceinline__ unsigned tr_base(int lane, int col, int hb) {
  return (unsigned)((((lane & 15) * 8) ^ (16 * hb)) + hb * 256 +
                    (col >> 4) * R4B);
}


// ---------------------------------------------------------------------
// Forward. Block = 4 waves x 32 q-rows = 128 q rows; KV tile = 64 keys.
// grid.x = s/128, grid.y = b*h.
// ---------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256, 3) void attn_fwd_kernel(
    const short* __restrict__ qg, const short* __restrict__ kg,
    const short* __restrict__ vg, short* __restrict__ og,
    float* __restrict__ lseg, int B, int S, int H, int KVH, float scale,
    long long vstride) {
  constexpr int KVB = 64;
  constexpr int NC = D / 16;   // QK^T k-chunks
  constexpr int NT = D / 32;   // 32-wide output tiles
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // SUBTILED K/V images (same layout as the backward kernels: coalesced
  // b128 staging, plain reads for the S^T A-operand, hardware transpose
  // reads for the PV A-operand). LDS diet for occupancy: K is
  // SINGLE-buffered, V double-buffered (48 KB at D=128 -> 3 blocks/CU,
  // up from 2 at the fully double-buffered 64 KB; VGPRs fit 3 waves).
  // K(t+1) is staged mid-tile right after the last S-phase read of
  // K(t), costing a second barrier per tile.
  constexpr int KB = KVB * D * 2;
#define KLDS() (smem)
#define VLDS(buf) (smem + KB + ((buf) ? KB : 0))

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
  // v may be a strided view (a slice of the fused qkv projection)
  const short* vbase = vg + (long long)b * S * vstride + (long long)kvh * D;

  // softmax runs in base 2 (v_exp_f32 is natively 2^x; folding log2(e)
  // into the scale drops one v_mul per exponential)
  const float scale2 = scale * 1.4426950408889634f;
  // Q -> B-fragments (registers, reused all tiles), PRE-SCALED by
  // scale*log2e: accS arrives already in softmax units, dropping 16
  // VALU muls per sub-tile (PMC: every kernel is VALU-bound). The bf16
  // rounding of q*scale2 is within the operands' own precision.
  bf16x8v qb[NC];
  {
    const short* qp = qbase + (long long)my_q * qrow_stride + hb * 8;
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      bf16x8v qv = *(const bf16x8v*)(qp + c * 16);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        qv[e] = (__bf16)((float)qv[e] * scale2);
      qb[c] = qv;
    }
  }

  f32x16 accO[NT];
#pragma unroll
  for (int t = 0; t < NT; ++t) accO[t] = (f32x16)(0.f);
  float m_run = -1e30f, l_run = 0.f;

  const int ntiles = (q0 + 128 + KVB - 1) / KVB;
  const int t256 = threadIdx.x;
  // subtiled staging: thread t: key = t&63, colblk pair = t>>6 (each
  // thread copies 2x16B per image; 64-key rows x D cols). Global
  // pointers advance incrementally (no per-tile 64-bit muls) and the
  // LDS offsets are per-thread constants computed once.
  const int skey = t256 & 63;
  const int sd0 = (t256 >> 6) * (D / 4);   // 32 cols per thread @D=128
  const short* kp_s = kbase + (long long)skey * krow_stride + sd0;
  const short* vp_s = vbase + (long long)skey * vstride + sd0;
  const long long kstep = (long long)KVB * krow_stride;
  const long long vstep = (long long)KVB * vstride;
  int soff[D / 64][2];
#pragma unroll
  for (int c2 = 0; c2 < D / 64; ++c2) {
    soff[c2][0] = SUBT_OFF(skey, sd0 + c2 * 16, 16);
    soff[c2][1] = SUBT_OFF(skey, sd0 + c2 * 16 + 8, 16);
  }
  auto stage_k = [&]() {
#pragma unroll
    for (int c2 = 0; c2 < D / 64; ++c2) {
      *(f32x4*)(KLDS() + soff[c2][0]) = *(const f32x4*)(kp_s + c2 * 16);
      *(f32x4*)(KLDS() + soff[c2][1]) = *(const f32x4*)(kp_s + c2 * 16 + 8);
    }
    kp_s += kstep;
  };
  auto stage_v = [&](int buf) {
#pragma unroll
    for (int c2 = 0; c2 < D / 64; ++c2) {
      *(f32x4*)(VLDS(buf) + soff[c2][0]) = *(const f32x4*)(vp_s + c2 * 16);
      *(f32x4*)(VLDS(buf) + soff[c2][1]) = *(const f32x4*)(vp_s + c2 * 16 + 8);
    }
    vp_s += vstep;
  };
  stage_k();
  stage_v(0);
  __syncthreads();
  // per-buffer tr-read bases for the PV loops (R4 = KVB/4 = 16)
  const unsigned pv_swz = tr_base<2048>(lane, col, hb);
  const unsigned pvb[2] = {(unsigned)(unsigned long long)VLDS(0) + pv_swz,
                           (unsigned)(unsigned long long)VLDS(1) + pv_swz};
  int cur = 0;
  for (int tile = 0; tile < ntiles; ++tile) {
    const int kv0 = tile * KVB;
    if (tile + 1 < ntiles) stage_v(cur ^ 1);

    // ---- two 32-key sub-tiles, each with its own online-softmax pass.
    // Register economy: one live accS/p set (16 regs) instead of two.
    // defer-max (guide T13, THR=8 nats = 11.5 bits): the O/l rescale
    // runs only when the sub-tile max exceeds the running max by more
    // than THR; P is then bounded by e^THR which the fp32 accumulate
    // tolerates. Decision is made BEFORE this sub-tile's P is
    // exponentiated (the safe order).
    // S-phase + softmax + pack for one sub-tile; PV split out so kt=1's
    // PV can run after the mid-tile K barrier (single-buffered K).
    bf16x8v pb1_0, pb1_1;
    const bool act1 = (kv0 + 32) <= qw + 31;   // wave-uniform causal
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
      const int kv32 = kv0 + kt * 32;
      if (kt == 1 && !act1) break;
      f32x16 accS = (f32x16)(0.f);
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        const int row = kt * 32 + col;
        const bf16x8v a = *(const bf16x8v*)(
            KLDS() + SUBT_OFF(row, c * 16 + hb * 8, 16));
        accS = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, qb[c], accS, 0, 0, 0);
      }
      float p[16];
      const bool partial = (kv32 + 32) > (qw + 1);
      float mt = -1e30f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float sv = accS[r];                   // q pre-scaled at load
        if (partial && (kv32 + DROW(r, hb)) > my_q) sv = -1e30f;
        p[r] = sv;
        mt = fmaxf(mt, sv);
      }
      mt = fmaxf(mt, __shfl_xor(mt, 32, 64));
      float alpha = 1.f;
      if (mt > m_run + 11.5f) {        // defer-max threshold (base-2)
        alpha = exp2f(m_run - mt);
        m_run = mt;
#pragma unroll
        for (int t = 0; t < NT; ++t)
#pragma unroll
          for (int r = 0; r < 16; ++r) accO[t][r] *= alpha;
      }
      float s_own = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p[r] = exp2f(p[r] - m_run);
        s_own += p[r];
      }
      l_run = l_run * alpha + s_own + __shfl_xor(s_own, 32, 64);

      const unsigned pvbase = pvb[cur] + kt * 1024;   // rg += kt*8
      if (kt == 0) {
        // PV immediately (software-pipelined immediate-offset reads;
        // first issue BEFORE the P pack so LDS latency hides behind
        // the pack VALU work)
        U2x64 fr[2][2];
        tr_issue4<0, 128, 512, 640>(pvbase, fr[0][0], fr[0][1]);
        const bf16x8v pb0 = pack_pT_chunk(p);
        const bf16x8v pb1 = pack_pT_chunk(p + 8);
#pragma unroll
        for (int t = 0; t < NT; ++t) {
          const int pp = t & 1;
          if (t + 1 < NT) {
            // offsets: (t+1)*2*(KVB/4)*128 = (t+1)*4096 + rg*128
            if (t + 1 == 1)
              tr_issue4<4096, 4096 + 128, 4096 + 512, 4096 + 640>(
                  pvbase, fr[1][0], fr[1][1]);
            else if (t + 1 == 2)
              tr_issue4<8192, 8192 + 128, 8192 + 512, 8192 + 640>(
                  pvbase, fr[0][0], fr[0][1]);
            else
              tr_issue4<12288, 12288 + 128, 12288 + 512, 12288 + 640>(
                  pvbase, fr[1][0], fr[1][1]);
            tr_wait<4>(fr[pp][0], fr[pp][1]);
          } else {
            tr_wait<0>(fr[pp][0], fr[pp][1]);
          }
          accO[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(fr[pp][0].v, pb0,
                                                            accO[t], 0, 0, 0);
          accO[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(fr[pp][1].v, pb1,
                                                            accO[t], 0, 0, 0);
        }
      } else {
        // pack only; PV runs after the K barrier below
        pb1_0 = pack_pT_chunk(p);
        pb1_1 = pack_pT_chunk(p + 8);
      }
    }
    // all waves are done reading K(tile): restage the single K image
    __syncthreads();
    if (tile + 1 < ntiles) stage_k();
    if (act1) {
      const unsigned pvbase = pvb[cur] + 1024;
      U2x64 fr[2][2];
      tr_issue4<0, 128, 512, 640>(pvbase, fr[0][0], fr[0][1]);
#pragma unroll
      for (int t = 0; t < NT; ++t) {
        const int pp = t & 1;
        if (t + 1 < NT) {
          if (t + 1 == 1)
            tr_issue4<4096, 4096 + 128, 4096 + 512, 4096 + 640>(
                pvbase, fr[1][0], fr[1][1]);
          else if (t + 1 == 2)
            tr_issue4<8192, 8192 + 128, 8192 + 512, 8192 + 640>(
                pvbase, fr[0][0], fr[0][1]);
          else
            tr_issue4<12288, 12288 + 128, 12288 + 512, 12288 + 640>(
                pvbase, fr[1][0], fr[1][1]);
          tr_wait<4>(fr[pp][0], fr[pp][1]);
        } else {
          tr_wait<0>(fr[pp][0], fr[pp][1]);
        }
        accO[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(fr[pp][0].v, pb1_0,
                                                          accO[t], 0, 0, 0);
        accO[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(fr[pp][1].v, pb1_1,
                                                          accO[t], 0, 0, 0);
      }
    }
    __syncthreads();
    cur ^= 1;
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
  if (hb == 0)   // exported in natural-log units (tests, bwd contract)
    lseg[((long long)bh) * S + my_q] =
        (m_run + __log2f(l_run)) * 0.6931471805599453f;
}
#undef KLDS
#undef VLDS

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
// Backward dq, consuming the dS workspace the dkv kernel published:
//   dq^T[dk][q] = sum_key K^T[dk][key] * dS^T[key][q]
// No S/dP recompute, no softmax, no lse/delta reads — the backward
// drops from 7 gemm chains to 5 (dkv computes dS once for both sides).
// A = K^T via tr reads on the K image; B = dS^T via tr reads on the
// staged dS tile. grid over q-tiles; each wave owns its 32 q rows.
// ---------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256, 2) void attn_bwd_dq_kernel(
    const short* __restrict__ kg, const short* __restrict__ dsw,
    short* __restrict__ dqg, int B, int S, int H, int KVH) {
  constexpr int KVB = 32;
  constexpr int NT = D / 32;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // double-buffered subtiled images: K (KVB x D) + dS (KVB x 128 q)
  constexpr int KIMGB = KVB * D * 2;
  constexpr int SIMGB = KVB * 128 * 2;
#define KIMG(buf) (smem + ((buf) ? KIMGB + SIMGB : 0))
#define SIMG(buf) (smem + KIMGB + ((buf) ? KIMGB + SIMGB : 0))

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
  const short* kbase = kg + ((long long)b * S * KVH + (long long)kvh) * D;
  const short* dsbase = dsw + (long long)bh * S * S + q0;

  f32x16 accDQ[NT];
#pragma unroll
  for (int t = 0; t < NT; ++t) accDQ[t] = (f32x16)(0.f);

  const int ntiles = (q0 + 128 + KVB - 1) / KVB;
  const int t256 = threadIdx.x;
  // staging: K rows (t&31, D cols split over 8 groups) and dS rows
  // (t&31 keys x 128 q cols split over 8 groups of 16)
  const int skey = t256 & 31;
  const int sgrp = t256 >> 5;            // 0..7
  const int skd0 = sgrp * (D / 8);
  const short* kp_s = kbase + (long long)skey * krow_stride + skd0;
  const long long kstep = (long long)KVB * krow_stride;
  const short* sp_s = dsbase + (long long)skey * S + sgrp * 16;
  const long long sstep = (long long)KVB * S;
  const int koff = SUBT_OFF(skey, skd0, 8);
  const int koff1 = SUBT_OFF(skey, skd0 + 8, 8);
  const int soff0 = SUBT_OFF(skey, sgrp * 16, 8);
  const int soff1 = SUBT_OFF(skey, sgrp * 16 + 8, 8);
  auto stage = [&](int buf) {
    if constexpr (D == 128) {
      *(f32x4*)(KIMG(buf) + koff) = *(const f32x4*)(kp_s);
      *(f32x4*)(KIMG(buf) + koff1) = *(const f32x4*)(kp_s + 8);
    } else {
      *(f32x4*)(KIMG(buf) + koff) = *(const f32x4*)(kp_s);
    }
    *(f32x4*)(SIMG(buf) + soff0) = *(const f32x4*)(sp_s);
    *(f32x4*)(SIMG(buf) + soff1) = *(const f32x4*)(sp_s + 8);
    kp_s += kstep;
    sp_s += sstep;
  };
  stage(0);
  __syncthreads();
  // tr-read bases (R4 = 8 for both images)
  const unsigned ktr = tr_base<1024>(lane, col, hb);
  const unsigned kib[2] = {(unsigned)(unsigned long long)KIMG(0) + ktr,
                           (unsigned)(unsigned long long)KIMG(1) + ktr};
  // dS image cols = the wave's q window: base col = wid*32
  const unsigned str = (unsigned)((((lane & 15) * 8) ^ (16 * hb)) +
                                  hb * 256 + ((col >> 4) << 10) +
                                  (wid * 32) * 64);
  const unsigned sib[2] = {(unsigned)(unsigned long long)SIMG(0) + str,
                           (unsigned)(unsigned long long)SIMG(1) + str};

  int cur = 0;
  for (int tile = 0; tile < ntiles; ++tile) {
    if (tile + 1 < ntiles) stage(cur ^ 1);
    // B fragments: dS^T, two 16-key chunks
    U2x64 bs[2];
    tr_issue4<0, 128, 512, 512 + 128>(sib[cur], bs[0], bs[1]);
    tr_wait<0>(bs[0], bs[1]);
    const unsigned kbase_t = kib[cur];
    U2x64 fr[2][2];
    tr_issue4<0, 128, 512, 640>(kbase_t, fr[0][0], fr[0][1]);
#pragma unroll
    for (int t = 0; t < NT; ++t) {
      const int pp = t & 1;
      if (t + 1 < NT) {
        if (t + 1 == 1)
          tr_issue4<2048, 2048 + 128, 2048 + 512, 2048 + 640>(
              kbase_t, fr[1][0], fr[1][1]);
        else if (t + 1 == 2)
          tr_issue4<4096, 4096 + 128, 4096 + 512, 4096 + 640>(
              kbase_t, fr[0][0], fr[0][1]);
        else
          tr_issue4<6144, 6144 + 128, 6144 + 512, 6144 + 640>(
              kbase_t, fr[1][0], fr[1][1]);
        tr_wait<4>(fr[pp][0], fr[pp][1]);
      } else {
        tr_wait<0>(fr[pp][0], fr[pp][1]);
      }
      accDQ[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(fr[pp][0].v,
                                                         bs[0].v, accDQ[t],
                                                         0, 0, 0);
      accDQ[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(fr[pp][1].v,
                                                         bs[1].v, accDQ[t],
                                                         0, 0, 0);
    }
    __syncthreads();
    cur ^= 1;
  }

  // dq was accumulated UNSCALED dS? No — dkv stored ds already *scale.
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
#undef KIMG
#undef SIMG

// ---------------------------------------------------------------------
// Backward dk/dv: grid over KV tiles; each wave owns 32 keys and walks
// all q-tiles >= its block's kv base. dK/dV accumulate in registers;
// V stays in registers; K in a per-wave swizzled LDS tile; Q/dO arrive
// per q-iteration in shared SUBTILED images (coalesced vector staging,
// plain reads for the S/dP B-operands, tr_read for the dV/dK
// B-operands). The P/dS lane<->reg transpose goes through a small
// wave-private LDS buffer. TWO barriers per q-iteration.
// ---------------------------------------------------------------------
template <int D, bool OUT_BF16>
__global__ __launch_bounds__(256, 2) void attn_bwd_dkv_kernel(
    const short* __restrict__ dog, const short* __restrict__ qg,
    const short* __restrict__ kg, const short* __restrict__ vg,
    const float* __restrict__ lseg, const float* __restrict__ deltag,
    void* __restrict__ dkg, void* __restrict__ dvg,
    short* __restrict__ dsw, int B, int S, int H,
    int KVH, float scale, long long vstride, long long dvstride) {
  constexpr int KVB = 32;   // keys per wave; block = 4 waves = 128 keys
  constexpr int NC = D / 16;
  constexpr int NT = D / 32;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_lds = smem;                           // [4][KVB][D*2] per-wave
  // double-buffered subtiled Q/dO images (stage q-tile t+1 during t)
  constexpr int IMGB = 32 * D * 2;
#define QIMG(buf) (smem + 4 * KVB * D * 2 + ((buf) ? 2 * IMGB : 0))
#define DOIMG(buf) (smem + 4 * KVB * D * 2 + IMGB + ((buf) ? 2 * IMGB : 0))
  // P/dS transpose buffer: rows padded 64 -> 80 B so banks rotate 20
  // per row — the 16-lane transposed reads land on 16 distinct 4-bank
  // groups with no XOR (the old 64 B rows + 2-bit XOR were 4-way
  // conflicted; dkv had 2x the bank conflicts of the other kernels,
  // profiles/attn_pmc_counters_r01.md). 80 preserves 16 B alignment.
  char* p_lds = smem + 4 * KVB * D * 2 + 4 * IMGB;  // [4][KVB][80]

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int col = lane & 31;
  const int hb = lane >> 5;

  const int bh = blockIdx.y;
  const int b = bh / H;
  const int h = bh % H;
  const int kvh = h / (H / KVH);
  const int ngrp = H / KVH;
  const int kv0 = blockIdx.x * 128 + wid * KVB;

  const long long qrow_stride = (long long)H * D;
  const long long krow_stride = (long long)KVH * D;
  const short* qbase = qg + ((long long)b * S * H + (long long)h) * D;
  const short* dobase = dog + ((long long)b * S * H + (long long)h) * D;
  const short* kbase = kg + ((long long)b * S * KVH + (long long)kvh) * D;
  const short* vbase = vg + (long long)b * S * vstride + (long long)kvh * D;

  char* my_k = k_lds + wid * KVB * D * 2;
  char* my_p = p_lds + wid * KVB * 80;

  // stage this wave's K rows once, in the SUBTILED image format (same
  // as the Q/dO images): the S-phase plain reads hit the proven
  // conflict-free pattern (PMC: the old row-swizzled tile carried the
  // residual bank conflicts)
  {
    constexpr int CHUNKS = KVB * D / 8;     // 16 B chunks
    for (int i = lane; i < CHUNKS; i += 64) {
      const int row = i / (D / 8);
      const int cb = (i % (D / 8)) * 8;
      const long long g = (long long)(kv0 + row) * krow_stride + cb;
      *(f32x4*)(my_k + SUBT_OFF(row, cb, 8)) = *(const f32x4*)(kbase + g);
    }
  }
  // V fragments stay in registers (A-operand rows = this lane's key)
  bf16x8v vreg[NC];
  {
    const short* vp = vbase + (long long)(kv0 + col) * vstride + hb * 8;
#pragma unroll
    for (int c = 0; c < NC; ++c) vreg[c] = *(const bf16x8v*)(vp + c * 16);
  }

  f32x16 accDV[NT], accDK[NT];
#pragma unroll
  for (int t = 0; t < NT; ++t) {
    accDV[t] = (f32x16)(0.f);
    accDK[t] = (f32x16)(0.f);
  }

  const int q_start = (blockIdx.x * 128) / 32 * 32;
  const int t256 = threadIdx.x;
  // thread t owns q row t&31 and ADJACENT column chunks (see the dq
  // kernel's staging comment: cacheline locality matters here);
  // incremental global pointers + precomputed LDS offsets
  const int sq = t256 & 31;
  const int sd0 = (t256 >> 5) * ((D == 128) ? 16 : 8);
  const short* qp_s = qbase + ((long long)(q_start + sq)) * qrow_stride + sd0;
  const short* dop_s = dobase + ((long long)(q_start + sq)) * qrow_stride + sd0;
  const long long qstep = 32 * qrow_stride;
  const int soff0 = SUBT_OFF(sq, sd0, 8);
  const int soff1 = SUBT_OFF(sq, sd0 + 8, 8);
  auto stage = [&](int buf) {
    if constexpr (D == 128) {
      *(f32x4*)(QIMG(buf) + soff0) = *(const f32x4*)(qp_s);
      *(f32x4*)(QIMG(buf) + soff1) = *(const f32x4*)(qp_s + 8);
      *(f32x4*)(DOIMG(buf) + soff0) = *(const f32x4*)(dop_s);
      *(f32x4*)(DOIMG(buf) + soff1) = *(const f32x4*)(dop_s + 8);
    } else {  // D=64: 32 rows x 8 chunks, one 16B chunk/thread
      *(f32x4*)(QIMG(buf) + soff0) = *(const f32x4*)(qp_s);
      *(f32x4*)(DOIMG(buf) + soff0) = *(const f32x4*)(dop_s);
    }
    qp_s += qstep;
    dop_s += qstep;
  };
  stage(0);
  __syncthreads();
  // tr-read base offset into the Q/dO images (R4 = 8)
  const unsigned kv_swz = tr_base<1024>(lane, col, hb);
  int cur = 0;
  for (int q0 = q_start; q0 < S; q0 += 32) {
    if (q0 + 32 < S) stage(cur ^ 1);

    // S^T (A=K lds, B=imgq) and dP^T (A=V regs, B=imgdo)
    f32x16 accS = (f32x16)(0.f), accDP = (f32x16)(0.f);
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      const int boff = SUBT_OFF(col, c * 16 + hb * 8, 8);
      const bf16x8v ka = *(const bf16x8v*)(my_k + boff);
      const bf16x8v qbf = *(const bf16x8v*)(QIMG(cur) + boff);
      const bf16x8v dbf = *(const bf16x8v*)(DOIMG(cur) + boff);
      accS = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qbf, accS, 0, 0, 0);
      accDP = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vreg[c], dbf, accDP,
                                                      0, 0, 0);
    }

    const float lse_q2 =
        lseg[((long long)bh) * S + q0 + col] * 1.4426950408889634f;
    const float delta_q = deltag[((long long)bh) * S + q0 + col];
    const float scale2 = scale * 1.4426950408889634f;
    float pv[16], ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int key = kv0 + DROW(r, hb);
      const int q = q0 + col;
      float pval = exp2f(accS[r] * scale2 - lse_q2);
      if (key > q) pval = 0.f;
      pv[r] = pval;
      ds[r] = pval * (accDP[r] - delta_q) * scale;
    }
    // P -> my_p (wave-private; in-wave LDS ordering suffices)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int key = DROW(r, hb);
      *(short*)(my_p + key * 80 + col * 2) = f2bf(pv[r]);
    }
    // dV[key][dv] += P(A) @ tr(imgdo)(B). Immediate-offset reads (one
    // base add per t) — the dkv kernel sits at the 255-VGPR edge, so no
    // extra pipeline slots here, but the address chains still go.
#define DKV_TRLOOP(ACC, A0, A1, BASE)                                        \
  do {                                                                       \
    _Pragma("unroll") for (int t = 0; t < NT; ++t) {                         \
      U2x64 f0, f1;                                                          \
      const unsigned b_t = (BASE) + t * 2048;                                \
      tr_issue4<0, 128, 512, 640>(b_t, f0, f1);                              \
      tr_wait<0>(f0, f1);                                                    \
      ACC[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(A0, f0.v, ACC[t],     \
                                                       0, 0, 0);             \
      ACC[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(A1, f1.v, ACC[t],     \
                                                       0, 0, 0);             \
    }                                                                        \
  } while (0)
    {
      bf16x8v pa[2];
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        const int inrow = kc * 32 + hb * 16;
        pa[kc] = *(const bf16x8v*)(my_p + col * 80 + inrow);
      }
      const unsigned dob = (unsigned)(unsigned long long)DOIMG(cur) + kv_swz;
      DKV_TRLOOP(accDV, pa[0], pa[1], dob);
    }
    // dS -> my_p, then dK[key][dk] += dS(A) @ tr(imgq)(B)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int key = DROW(r, hb);
      *(short*)(my_p + key * 80 + col * 2) = f2bf(ds[r]);
    }
    // publish this (32-key x 32-q) dS tile to the global workspace: the
    // dq kernel consumes it instead of recomputing the S and dP chains
    // (7 gemm chains across the backward -> 5). Vectorized readback from
    // the my_p transpose buffer: lane l covers key l&31, 16B q-chunk
    // (l>>5)*2 + c.
    {
      const int key = lane & 31;
      const long long grow = ((long long)bh * S + kv0 + key) * S + q0;
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        const int qc = ((lane >> 5) * 2 + c) * 8;
        *(f32x4*)(dsw + grow + qc) =
            *(const f32x4*)(my_p + key * 80 + qc * 2);
      }
    }
    {
      bf16x8v da[2];
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        const int inrow = kc * 32 + hb * 16;
        da[kc] = *(const bf16x8v*)(my_p + col * 80 + inrow);
      }
      const unsigned qib = (unsigned)(unsigned long long)QIMG(cur) + kv_swz;
      DKV_TRLOOP(accDK, da[0], da[1], qib);
    }
#undef DKV_TRLOOP
    __syncthreads();
    cur ^= 1;
  }
#undef QIMG
#undef DOIMG

  // write dK/dV. D-layout: row i = key_rel = DROW(r,hb), col j = feature
  // = t*32 + (lane&31). OUT_BF16 (no GQA): each element written exactly
  // once -> direct bf16 stores, no fp32 scratch/memset/convert pass.
  // GQA (ngrp>1): head-groups collide on (b, key, kvh) -> fp32 atomics.
#pragma unroll
  for (int t = 0; t < NT; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int key_abs = kv0 + DROW(r, hb);
      const int feat = t * 32 + col;
      const long long off =
          ((long long)b * S + key_abs) * (long long)(KVH * D) +
          (long long)kvh * D + feat;
      // dv may target a strided slice of a fused dqkv buffer
      const long long offv = ((long long)b * S + key_abs) * dvstride +
                             (long long)kvh * D + feat;
      if (OUT_BF16) {
        ((short*)dkg)[off] = f2bf(accDK[t][r]);
        ((short*)dvg)[offv] = f2bf(accDV[t][r]);
      } else if (ngrp > 1) {
        atomicAdd((float*)dkg + off, accDK[t][r]);
        atomicAdd((float*)dvg + offv, accDV[t][r]);
      } else {
        ((float*)dkg)[off] += accDK[t][r];
        ((float*)dvg)[offv] += accDV[t][r];
      }
    }
  }
}

extern "C" {

void launch_attn_fwd(const void* q, const void* k, const void* v, void* o,
                     float* lse, int B, int S, int H, int KVH, int D,
                     float scale, long long vstride, hipStream_t stream) {
  dim3 grid(S / 128, B * H);
  const int lds = 3 * 64 * D * 2;  // single K image + dbuf V
  if (D == 128)
    attn_fwd_kernel<128><<<grid, 256, lds, stream>>>(
        (const short*)q, (const short*)k, (const short*)v, (short*)o, lse, B,
        S, H, KVH, scale, vstride);
  else
    attn_fwd_kernel<64><<<grid, 256, lds, stream>>>(
        (const short*)q, (const short*)k, (const short*)v, (short*)o, lse, B,
        S, H, KVH, scale, vstride);
}

void launch_attn_bwd(const void* do_, const void* q, const void* k,
                     const void* v, const void* o, const float* lse, void* dq,
                     void* dk, void* dv, float* delta_ws, void* ds_ws,
                     int B, int S, int H,
                     int KVH, int D, float scale, int out_bf16,
                     long long vstride, long long dvstride,
                     hipStream_t stream) {
  const long long rows = (long long)B * S * H;
  attn_bwd_delta_kernel<<<(int)((rows * 64 + 255) / 256), 256, 0, stream>>>(
      (const short*)do_, (const short*)o, delta_ws, D, S, H, rows);
  dim3 grid(S / 128, B * H);
  // dkv FIRST: it publishes the dS workspace the dq kernel consumes
  if (D == 128) {
    const int lds_dkv = 4 * 32 * 128 * 2 + 4 * 32 * 128 * 2 + 4 * 32 * 80;
    if (out_bf16)
      attn_bwd_dkv_kernel<128, true><<<grid, 256, lds_dkv, stream>>>(
          (const short*)do_, (const short*)q, (const short*)k,
          (const short*)v, lse, delta_ws, dk, dv, (short*)ds_ws, B, S, H,
          KVH, scale, vstride, dvstride);
    else
      attn_bwd_dkv_kernel<128, false><<<grid, 256, lds_dkv, stream>>>(
          (const short*)do_, (const short*)q, (const short*)k,
          (const short*)v, lse, delta_ws, dk, dv, (short*)ds_ws, B, S, H,
          KVH, scale, vstride, dvstride);
    const int lds_dq = 2 * (32 * 128 * 2 + 32 * 128 * 2);
    attn_bwd_dq_kernel<128><<<grid, 256, lds_dq, stream>>>(
        (const short*)k, (const short*)ds_ws, (short*)dq, B, S, H, KVH);
  } else {
    const int lds_dkv = 4 * 32 * 64 * 2 + 4 * 32 * 64 * 2 + 4 * 32 * 80;
    if (out_bf16)
      attn_bwd_dkv_kernel<64, true><<<grid, 256, lds_dkv, stream>>>(
          (const short*)do_, (const short*)q, (const short*)k,
          (const short*)v, lse, delta_ws, dk, dv, (short*)ds_ws, B, S, H,
          KVH, scale, vstride, dvstride);
    else
      attn_bwd_dkv_kernel<64, false><<<grid, 256, lds_dkv, stream>>>(
          (const short*)do_, (const short*)q, (const short*)k,
          (const short*)v, lse, delta_ws, dk, dv, (short*)ds_ws, B, S, H,
          KVH, scale, vstride, dvstride);
    const int lds_dq = 2 * (32 * 64 * 2 + 32 * 128 * 2);
    attn_bwd_dq_kernel<64><<<grid, 256, lds_dq, stream>>>(
        (const short*)k, (const short*)ds_ws, (short*)dq, B, S, H, KVH);
  }
}

}  // extern "C"
