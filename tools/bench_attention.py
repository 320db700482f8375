"""Standalone attention kernel microbenchmark (bench shape: b2 h32 kvh32
s4096 d128 — the llama2-7b hot shape) with TF/s at causal flop count.
Run on GPU: python tools/bench_attention.py"""

import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from fms_fsdp_amd import _C


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters


def main():
    torch.manual_seed(0)
    for (b, s, h, kvh, d) in [(2, 4096, 32, 32, 128), (2, 4096, 32, 8, 128)]:
        q = torch.randn(b, s, h, d, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(b, s, kvh, d, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(b, s, kvh, d, device="cuda", dtype=torch.bfloat16)
        do = torch.randn_like(q)
        o, lse = _C.attn_fwd(q, k, v)

        fwd_flops = 2 * 2 * b * h * s * s * d * 0.5  # QK^T+PV, causal
        t = bench(lambda: _C.attn_fwd(q, k, v))
        print(f"fwd  b{b} s{s} h{h} kvh{kvh}: {t*1e3:8.3f} ms  "
              f"{fwd_flops/t/1e12:7.1f} TF/s")

        bwd_flops = fwd_flops * 2.5
        t = bench(lambda: _C.attn_bwd(do, q, k, v, o, lse, None), iters=10)
        print(f"bwd  b{b} s{s} h{h} kvh{kvh}: {t*1e3:8.3f} ms  "
              f"{bwd_flops/t/1e12:7.1f} TF/s (dq+dkv+delta)")

        # SDPA comparison (aotriton flash, for reference only)
        qt, kt, vt = (x.transpose(1, 2) for x in (q, k, v))
        t = bench(lambda: torch.nn.functional.scaled_dot_product_attention(
            qt, kt, vt, is_causal=True, enable_gqa=(kvh != h)))
        print(f"sdpa fwd reference:       {t*1e3:8.3f} ms  "
              f"{fwd_flops/t/1e12:7.1f} TF/s")


if __name__ == "__main__":
    main()
