"""Micro-bench for the fused AdamW kernel (7B-scale flat shard).

Traffic/step = 4+4+4 fp32 r/w + 2 bf16 r/w bytes/elem = 28 B/elem.
"""
import os
import sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from fms_fsdp_amd import _C


def main():
    n = 6_738_415_616 // 4 * 4  # 7B params, 4-aligned
    dev = "cuda:0"
    p = torch.randn(n, device=dev, dtype=torch.float32)
    m = torch.zeros(n, device=dev, dtype=torch.float32)
    v = torch.zeros(n, device=dev, dtype=torch.float32)
    g = torch.randn(n, device=dev, dtype=torch.bfloat16)
    pb = torch.empty(n, device=dev, dtype=torch.bfloat16)
    for _ in range(3):
        _C.adamw(p, g, m, v, 1.0, 3e-4, 0.9, 0.95, 1e-8, 0.1, None, pb)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    iters = 10
    t0.record()
    for i in range(iters):
        _C.adamw(p, g, m, v, i + 2, 3e-4, 0.9, 0.95, 1e-8, 0.1, None, pb)
    t1.record()
    torch.cuda.synchronize()
    ms = t0.elapsed_time(t1) / iters
    gb = n * 28 / 1e9
    print(f"n={n} {ms:.2f} ms/iter  {gb / ms * 1000 / 1000:.2f} TB/s")


if __name__ == "__main__":
    main()
