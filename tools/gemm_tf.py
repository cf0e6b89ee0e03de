#!/usr/bin/env python3
"""Pure-GEMM throughput probe for the MFMA tile core (run on a GPU box).

Separates core efficiency from conv-gather overhead: the guide's reference
points are ~122 TF for an untuned 32x32x2 LDS kernel and ~147 TF tuned at
4096^3 f32; torch.matmul (rocBLAS) gives the library number on the same box.
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from split_learning_amd.ops import native  # noqa: E402


def bench(fn, iters=20, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    dev = torch.device("cuda:0")
    for n in (1024, 2048, 4096):
        a = torch.randn(n, n, device=dev)
        b = torch.randn(n, n, device=dev)
        flops = 2.0 * n * n * n
        ext = native()  # raw op: bypasses the library-GEMM routing on purpose

        def ours():
            ext.matmul_f32(a, b, False, False, None, False)

        def lib():
            torch.matmul(a, b)

        t_ours = bench(ours)
        t_lib = bench(lib)
        print(f"n={n}: ours {flops / t_ours / 1e12:7.1f} TF   "
              f"rocBLAS {flops / t_lib / 1e12:7.1f} TF", flush=True)


if __name__ == "__main__":
    main()
