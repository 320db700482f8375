// Hand-written CDNA4 GEMM for the dgrad layout (VERDICT round-1 item 6):
//   C[M,N] = A[M,K] @ B[K,N], all row-major bf16, fp32 accumulate.
// (dx = dy @ W with W stored [out,in]: the B operand is k-strided.)
//
// STATUS: correctness-exact (bit-equal to torch.mm on 4 of 5 bench
// shapes) but NOT wired into the hot path. Measured on MI355X
// (tools/bench_gemm.py GEMM_CUSTOM=1): plain-load staging 517-576 TF;
// this glds version 922-1012 TF (+80-90%, exactly the guide's "staging
// pipeline is the lever"); hipBLASLt on the same warm shapes: 1120-1432
// TF. The TunableOp-table times that motivated the attempt (1.09-1.19
// PF "NT class") reflect tuning-loop conditions, not the library's warm
// steady state — the library's Tensile asm kernels hold a ~1.2-1.4x
// margin over this HIP-level implementation, consistent with the CDNA4
// guide's own HIP-template ceiling (~1.16-1.22 PF at 8192^3) vs its
// asm figure (2.0 PF). Kept as the documented experiment + test target.
//
// Design (guide §5 canonical GEMM, attention-kernel idioms):
// - 256(M) x 256(N) C-tile per workgroup, BK=64, double-buffered LDS
//   images (128 KB -> 1 block/CU), 8 waves as 4(M) x 2(N): each wave
//   owns 64 x 128 of C = eight 32x32 MFMA tiles (128 accum VGPRs).
// - A tile staged row-major-subtiled, fragments by plain b128 reads.
// - B tile staged into the same subtiled format and read with
//   ds_read_b64_tr_b16 immediate-offset groups (the attention-V recipe):
//   the hardware transpose turns the k-strided operand into clean
//   B-fragments with no VALU address chains.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(16))) float f32x16;

#define SUBT_OFF(row, col, R4) \
  ((((col) >> 4) * (R4) + ((row) >> 2)) * 128 + \
   ((((row) & 3) * 32 + ((col) & 15) * 2) ^ ((((row) >> 2) & 2) << 3)))

union GU2x64 {
  unsigned long long u[2];
  bf16x8v v;
};

template <int O0, int O1, int O2, int O3>
__device__ __forceinline__ void g_tr_issue4(unsigned base, GU2x64& r0,
                                            GU2x64& r1) {
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
__device__ __forceinline__ void g_tr_wait(GU2x64& r0, GU2x64& r1) {
  asm volatile("s_waitcnt lgkmcnt(%c4)"
               : "+v"(r0.u[0]), "+v"(r0.u[1]), "+v"(r1.u[0]), "+v"(r1.u[1])
               : "i"(CNT)
               : "memory");
}

// D-layout row index for reg r, half hb (see attention.hip)
#define GDROW(r, hb) (((r) & 3) + 8 * ((r) >> 2) + 4 * (hb))

// async global->LDS DMA, 16 B per lane, destination = uniform base +
// lane*16 (guide §5: the glds staging pipeline is the GEMM throughput
// lever; plain-load + ds_write staging measured 520 TF vs the library's
// 1060-1335 on these shapes)
__device__ __forceinline__ void glds16(const short* g, char* l) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)g,
      (__attribute__((address_space(3))) void*)l, 16, 0, 0);
}

// invert the subtiled-image layout: linear 16B chunk index -> (row, col8)
// so each lane's GLOBAL address is pre-swizzled while LDS stays
// lane-linear (glds cannot scatter)
__device__ __forceinline__ void subt_invert(int chunk, int R4, int& row,
                                            int& col8) {
  const int st = chunk >> 3;          // 128 B subtile index
  const int sub16 = chunk & 7;        // 16 B slot within the subtile
  const int colblk = st / R4;
  const int rowgrp = st % R4;
  const int x = sub16 ^ ((rowgrp >> 1) & 1);   // undo the 16 B XOR
  row = rowgrp * 4 + (x >> 1);
  col8 = colblk * 16 + (x & 1) * 8;
}

__global__ __launch_bounds__(512, 1) void gemm_nt_kernel(
    const short* __restrict__ Ag, const short* __restrict__ Bg,
    short* __restrict__ Cg, int M, int N, int K) {
  constexpr int BK = 64;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // A image: 256 rows(m) x 64 cols(k)  -> R4 = 64, 32 KB
  // B image: 64 rows(k) x 256 cols(n)  -> R4 = 16, 32 KB
  constexpr int AB = 256 * BK * 2;
#define ALDS(buf) (smem + ((buf) ? 2 * AB : 0))
#define BLDS(buf) (smem + AB + ((buf) ? 2 * AB : 0))

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;          // 0..7
  const int col = lane & 31;
  const int hb = lane >> 5;

  const int m0 = blockIdx.x * 256;           // C tile origin
  const int n0 = blockIdx.y * 256;
  const int wm = (wid & 3) * 64;             // wave's m offset in tile
  const int wn = (wid >> 2) * 128;           // wave's n offset in tile

  // ---- glds staging: per wave, 4 DMA instructions per image fill a
  // lane-linear 1 KB span; the subtile swizzle lives on the SOURCE
  // addresses (subt_invert). Per-thread pointers advance incrementally.
  const short* ga[4];
  const short* gb[4];
  int lofs[4];
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    const int chunk = (wid * 4 + q) * 64 + lane;
    int row, col8;
    subt_invert(chunk, 64, row, col8);
    ga[q] = Ag + (long long)(m0 + row) * K + col8;
    subt_invert(chunk, 16, row, col8);
    gb[q] = Bg + (long long)row * N + n0 + col8;
    lofs[q] = (wid * 4 + q) * 1024;
  }
  const long long badv = (long long)BK * N;
  auto stage = [&](int buf) {
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      glds16(ga[q], ALDS(buf) + lofs[q]);
      glds16(gb[q], BLDS(buf) + lofs[q]);
      ga[q] += BK;
      gb[q] += badv;
    }
  };

  f32x16 acc[2][4];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x16)(0.f);

  stage(0);
  __syncthreads();   // drains the DMA (vmcnt 0) + barrier
  // tr-read base into the B image (R4 = 16): swizzle ^ k-half ^ 16-lane
  // block ^ the wave's 128-col offset (wn*128 bytes)
  const unsigned btr0 = (unsigned)(unsigned long long)BLDS(0) +
                        ((((lane & 15) * 8) ^ (16 * hb)) + hb * 256 +
                         ((col >> 4) << 11) + wn * 128);
  const unsigned bstep = (unsigned)(2 * AB);

  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += BK) {
    if (k0 + BK < K) stage(cur ^ 1);   // fire-and-forget DMA
    const unsigned bbase = btr0 + (cur ? bstep : 0);
#pragma unroll
    for (int kc = 0; kc < BK / 16; ++kc) {
      // A fragments: rows wm+mt*32+col, cols kc*16 + hb*8 (plain reads)
      bf16x8v af[2];
#pragma unroll
      for (int mt = 0; mt < 2; ++mt)
        af[mt] = *(const bf16x8v*)(
            ALDS(cur) + SUBT_OFF(wm + mt * 32 + col, kc * 16 + hb * 8, 64));
      // B fragments, 4 n-tiles: per (nt, kc) two tr reads at
      // nt*4096 + kc*512 + {0, 128} off the folded base
      GU2x64 f0a, f0b, f1a, f1b;
      if (kc == 0) {
        g_tr_issue4<0, 128, 4096, 4096 + 128>(bbase, f0a, f0b);
        g_tr_issue4<8192, 8192 + 128, 12288, 12288 + 128>(bbase, f1a, f1b);
      } else if (kc == 1) {
        g_tr_issue4<512, 512 + 128, 4608, 4608 + 128>(bbase, f0a, f0b);
        g_tr_issue4<8704, 8704 + 128, 12800, 12800 + 128>(bbase, f1a, f1b);
      } else if (kc == 2) {
        g_tr_issue4<1024, 1024 + 128, 5120, 5120 + 128>(bbase, f0a, f0b);
        g_tr_issue4<9216, 9216 + 128, 13312, 13312 + 128>(bbase, f1a, f1b);
      } else {
        g_tr_issue4<1536, 1536 + 128, 5632, 5632 + 128>(bbase, f0a, f0b);
        g_tr_issue4<9728, 9728 + 128, 13824, 13824 + 128>(bbase, f1a, f1b);
      }
      g_tr_wait<4>(f0a, f0b);   // group 0 (nt 0,1) landed
#pragma unroll
      for (int mt = 0; mt < 2; ++mt) {
        acc[mt][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            af[mt], f0a.v, acc[mt][0], 0, 0, 0);
        acc[mt][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            af[mt], f0b.v, acc[mt][1], 0, 0, 0);
      }
      g_tr_wait<0>(f1a, f1b);   // group 1 (nt 2,3)
#pragma unroll
      for (int mt = 0; mt < 2; ++mt) {
        acc[mt][2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            af[mt], f1a.v, acc[mt][2], 0, 0, 0);
        acc[mt][3] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            af[mt], f1b.v, acc[mt][3], 0, 0, 0);
      }
    }
    __syncthreads();   // vmcnt(0): next buffer's DMA landed, LDS quiesced
    cur ^= 1;
  }

  // ---- epilogue: D-layout -> bf16 C ----
#pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int n = n0 + wn + nt * 32 + col;
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        const int m = m0 + wm + mt * 32 + 8 * g + 4 * hb;
        bf16x4 w4;
#pragma unroll
        for (int j = 0; j < 4; ++j)
          w4.v[j] = f2bf(acc[mt][nt][g * 4 + j]);
        // D rows are m (A rows), cols n: 4 consecutive m rows
#pragma unroll
        for (int j = 0; j < 4; ++j)
          Cg[(long long)(m + j) * N + n] = w4.v[j];
      }
    }
  }
}

extern "C" {

int launch_gemm_nt(const void* A, const void* B, void* C, int M, int N,
                   int K, hipStream_t stream) {
  if (M % 256 || N % 256 || K % 64) return -1;
  dim3 grid(M / 256, N / 256);
  const int lds = 4 * 256 * 64 * 2;
  gemm_nt_kernel<<<grid, 512, lds, stream>>>((const short*)A,
                                             (const short*)B, (short*)C, M,
                                             N, K);
  return 0;
}

}  // extern "C"
