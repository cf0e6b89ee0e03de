#!/usr/bin/env python3
"""Winograd F(2x2,3x3) numerics check + per-shape timing vs the direct
implicit-GEMM conv kernels (GPU box).

RUN WITH SLK_WINO=0 so the routed entry points stay on the direct kernels
(the "dir" columns); the wino columns call the conv2d_wino* ops directly."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from split_learning_amd.ops.functional import native

SHAPES = [
    (32, 64, 32, 32, 64),
    (32, 64, 16, 16, 128),
    (32, 128, 16, 16, 128),
    (32, 128, 8, 8, 256),
    (32, 256, 8, 8, 256),
    (32, 256, 4, 4, 512),
    (32, 512, 4, 4, 512),
    (32, 512, 2, 2, 512),
]


def timed(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    n = native()
    dev = "cuda:0"
    torch.manual_seed(3)
    print(f"{'shape':>20} {'dir_f':>7} {'win_f':>7} {'dir_bd':>7} {'win_bd':>7}"
          f" {'dir_bw':>7} {'win_bw':>7}  maxerr_f maxerr_bd maxerr_bw")
    for (B, Ci, H, W, Co) in SHAPES:
        x = torch.randn(B, Ci, H, W, device=dev)
        w = torch.randn(Co, Ci, 3, 3, device=dev) * 0.1
        bias = torch.randn(Co, device=dev)
        y_dir = n.conv2d_fwd(x, w, bias, 1, 1)
        y_win = n.conv2d_wino(x, w, bias, 1, False)
        ref = torch.nn.functional.conv2d(x, w, bias, 1, 1)
        err_f = (y_win - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
        gy = torch.randn_like(y_dir)
        gx_dir = n.conv2d_bwd_data(gy, w, 1, 1, H, W)
        gx_win = n.conv2d_wino(gy, w, None, 1, True)
        err_bd = (gx_win - gx_dir).abs().max().item() / (gx_dir.abs().max().item() + 1e-9)
        gw_dir = n.conv2d_bwd_weight(gy, x, 3, 3, 1, 1)
        gw_win = n.conv2d_wino_bwdw(gy, x, 1)
        err_bw = (gw_win - gw_dir).abs().max().item() / (gw_dir.abs().max().item() + 1e-9)
        tf = timed(lambda: n.conv2d_fwd(x, w, bias, 1, 1))
        twf = timed(lambda: n.conv2d_wino(x, w, bias, 1, False))
        tbd = timed(lambda: n.conv2d_bwd_data(gy, w, 1, 1, H, W))
        twbd = timed(lambda: n.conv2d_wino(gy, w, None, 1, True))
        tbw = timed(lambda: n.conv2d_bwd_weight(gy, x, 3, 3, 1, 1))
        twbw = timed(lambda: n.conv2d_wino_bwdw(gy, x, 1))
        print(f"{B}x{Ci}x{H}x{W}->{Co}".rjust(20)
              + f" {tf:7.1f} {twf:7.1f} {tbd:7.1f} {twbd:7.1f}"
              f" {tbw:7.1f} {twbw:7.1f}"
              f"  {err_f:.2e} {err_bd:.2e} {err_bw:.2e}", flush=True)


if __name__ == "__main__":
    main()
