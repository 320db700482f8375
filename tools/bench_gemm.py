"""Microbenchmark the exact GEMM shapes of the llama2-7b training step
(hipBLASLt via torch.mm) to separate library efficiency from pipeline
overhead. Run on GPU: python tools/bench_gemm.py"""

import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

M = 8192  # b2 x s4096 tokens


def bench(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters


def gemm_case(name, m, k, n, ta=False, tb=False):
    a = torch.randn((k, m) if ta else (m, k), device="cuda", dtype=torch.bfloat16)
    b = torch.randn((n, k) if tb else (k, n), device="cuda", dtype=torch.bfloat16)
    A = a.t() if ta else a
    B = b.t() if tb else b
    t = bench(lambda: torch.mm(A, B))
    tf = 2 * m * k * n / t / 1e12
    print(f"{name:28s} M{m:6d} K{k:6d} N{n:6d} {'T' if ta else 'N'}{'T' if tb else 'N'}"
          f" {t*1e3:8.3f} ms {tf:7.0f} TF/s")
    return t


def main():
    torch.manual_seed(0)
    total = 0.0
    # forward
    total += gemm_case("qkv fwd", M, 4096, 6144)
    total += gemm_case("proj fwd", M, 4096, 4096)
    total += gemm_case("wg1 fwd", M, 4096, 22016)
    total += gemm_case("w2 fwd", M, 11008, 4096)
    # dgrad (dy @ W)
    total += gemm_case("qkv dgrad", M, 6144, 4096)
    total += gemm_case("proj dgrad", M, 4096, 4096)
    total += gemm_case("wg1 dgrad", M, 22016, 4096)
    total += gemm_case("w2 dgrad", M, 4096, 11008)
    # wgrad (dy^T @ x)
    total += gemm_case("qkv wgrad", 6144, M, 4096, ta=True)
    total += gemm_case("proj wgrad", 4096, M, 4096, ta=True)
    total += gemm_case("wg1 wgrad", 22016, M, 4096, ta=True)
    total += gemm_case("w2 wgrad", 4096, M, 11008, ta=True)
    print(f"\nper-layer GEMM total: {total*1e3:.2f} ms -> x32 layers = "
          f"{total*32*1e3:.1f} ms/step")
    # CE path
    t = gemm_case("ce logits", 2048, 4096, 32000)
    t += gemm_case("ce dx", 2048, 32000, 4096)
    t += gemm_case("ce dw", 32000, 2048, 4096, ta=True)
    print(f"CE chunk x4: {t*4*1e3:.1f} ms/step")


def bench_custom():
    """A/B the hand-written NT-class dgrad kernel vs hipBLASLt."""
    from fms_fsdp_amd import _C
    torch.manual_seed(0)
    print("\n-- custom gemm_nt vs torch.mm (NT dgrad class) --")
    for (name, m, k, n) in [("proj dgrad", M, 4096, 4096),
                            ("qkv dgrad", M, 12288, 4096),
                            ("wg1 dgrad", M, 22016, 4096),
                            ("w2 dgrad", M, 4096, 11008),
                            ("ce dx", M, 32000, 4096)]:
        a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16) * 0.1
        b = torch.randn(k, n, device="cuda", dtype=torch.bfloat16) * 0.1
        ref = torch.mm(a, b)
        got = _C.gemm_nt(a, b)
        err = ((got.float() - ref.float()).abs().max()
               / ref.float().abs().max().clamp(min=1e-6)).item()
        t_lib = bench(lambda: torch.mm(a, b))
        t_cus = bench(lambda: _C.gemm_nt(a, b))
        fl = 2 * m * k * n
        print(f"{name:12s} M{m} K{k:6d} N{n:6d}  lib {t_lib*1e3:7.3f} ms "
              f"({fl/t_lib/1e12:6.0f} TF)  custom {t_cus*1e3:7.3f} ms "
              f"({fl/t_cus/1e12:6.0f} TF)  relerr {err:.3e}")


if __name__ == "__main__":
    if os.environ.get("GEMM_CUSTOM") == "1":
        bench_custom()
    else:
        main()
