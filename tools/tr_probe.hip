// Probe the exact lane<->element mapping of ds_read_b64_tr_b16 on gfx950.
// LDS filled with lds[i] = i (shorts); three addressing schemes dumped.
// Build+run standalone: hipcc --offload-arch=gfx950 tools/tr_probe.hip -o
//   /tmp/trp && /tmp/trp
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void tr_probe(short* out) {
  __shared__ short lds[4096];
  for (int i = threadIdx.x; i < 4096; i += 64) lds[i] = (short)i;
  __syncthreads();
  const int l = threadIdx.x;
  unsigned base = (unsigned)(unsigned long long)(&lds[0]);
  for (int scheme = 0; scheme < 4; ++scheme) {
    unsigned addr = base;
    if (scheme == 1) addr += (l & 15) * 8;
    else if (scheme == 2) addr += l * 8;
    else if (scheme == 3) addr += (l & 15) * 2 + (l >> 4) * 128;
    unsigned long long v;
    asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                 : "=v"(v)
                 : "v"(addr)
                 : "memory");
    __builtin_amdgcn_sched_barrier(0);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      out[(scheme * 64 + l) * 4 + j] = (short)(v >> (16 * j));
  }
}

int main() {
  short* d;
  hipMalloc(&d, 4 * 64 * 4 * sizeof(short));
  tr_probe<<<1, 64>>>(d);
  short h[4 * 64 * 4];
  hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  const char* names[4] = {"uniform base", "(l&15)*8", "l*8",
                          "(l&15)*2+(l>>4)*128"};
  for (int s = 0; s < 4; ++s) {
    printf("== scheme %d: addr = base + %s\n", s, names[s]);
    for (int l = 0; l < 64; ++l) {
      printf("l%02d:[%4d %4d %4d %4d] ", l, h[(s * 64 + l) * 4],
             h[(s * 64 + l) * 4 + 1], h[(s * 64 + l) * 4 + 2],
             h[(s * 64 + l) * 4 + 3]);
      if (l % 4 == 3) printf("\n");
    }
  }
  return 0;
}
