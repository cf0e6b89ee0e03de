#!/usr/bin/env python3
"""Per-shape conv kernel microbenchmark (run on the GPU box).

Times the three conv kernels (fwd / bwd-data / bwd-weight) on each VGG16
layer shape with hipEvents, reports us + effective TF, and A/Bs against
torch's aten conv (MIOpen) on the same shapes.

  python tools/conv_microbench.py            # timing table
  python tools/conv_microbench.py --loop N --op fwd --shape 1
      # run one (op, shape) in a bare loop N times: the rocprofv3 --pmc target
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from split_learning_amd.ops.functional import native

# (B, Ci, H, W, Co, stride): the VGG16/CIFAR10 batch-32 conv stack (+ stem)
SHAPES = [
    (32, 3, 32, 32, 64, 1),
    (32, 64, 32, 32, 64, 1),
    (32, 64, 16, 16, 128, 1),
    (32, 128, 16, 16, 128, 1),
    (32, 128, 8, 8, 256, 1),
    (32, 256, 8, 8, 256, 1),
    (32, 256, 4, 4, 512, 1),
    (32, 512, 4, 4, 512, 1),
    (32, 512, 2, 2, 512, 1),
]


def timed(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--loop", type=int, default=0)
    ap.add_argument("--op", default="fwd", choices=["fwd", "bwdd", "bwdw"])
    ap.add_argument("--shape", type=int, default=1)
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    dev = torch.device("cuda:0")
    n = native()

    if args.loop:
        B, Ci, H, W, Co, s = SHAPES[args.shape]
        x = torch.randn(B, Ci, H, W, device=dev)
        w = torch.randn(Co, Ci, 3, 3, device=dev)
        y = n.conv2d_fwd(x, w, None, s, 1)
        for _ in range(args.loop):
            if args.op == "fwd":
                n.conv2d_fwd(x, w, None, s, 1)
            elif args.op == "bwdd":
                n.conv2d_bwd_data(y, w, s, 1, H, W)
            else:
                n.conv2d_bwd_weight(y, x, 3, 3, s, 1)
        torch.cuda.synchronize()
        return

    print(f"{'shape':>22} {'op':>5} {'ours_us':>9} {'torch_us':>9} "
          f"{'ours_TF':>8} {'torch_TF':>9}")
    tot = {"fwd": [0.0, 0.0], "bwdd": [0.0, 0.0], "bwdw": [0.0, 0.0]}
    for si, (B, Ci, H, W, Co, s) in enumerate(SHAPES):
        x = torch.randn(B, Ci, H, W, device=dev)
        w = torch.randn(Co, Ci, 3, 3, device=dev) * 0.05
        y = n.conv2d_fwd(x, w, None, s, 1)
        OH = y.shape[2]
        flops = 2.0 * B * Co * Ci * 9 * OH * OH

        xt = x.clone().requires_grad_(True)
        wt = w.clone().requires_grad_(True)
        yt = torch.nn.functional.conv2d(xt, wt, None, s, 1)
        gy = torch.randn_like(yt)

        # end-to-end per-call cost (any pad/flip prep of the SLK_CONV_PAD=1
        # path is charged to the op, as in the training step)
        ops = {
            "fwd": (lambda: n.conv2d_fwd(x, w, None, s, 1),
                    lambda: torch.nn.functional.conv2d(x, w, None, s, 1)),
            "bwdd": (lambda: n.conv2d_bwd_data(y, w, s, 1, H, W),
                     lambda: torch.ops.aten.convolution_backward(
                         gy, xt, wt, None, (s, s), (1, 1), (1, 1), False,
                         (0, 0), 1, (True, False, False))),
            "bwdw": (lambda: n.conv2d_bwd_weight(y, x, 3, 3, s, 1),
                     lambda: torch.ops.aten.convolution_backward(
                         gy, xt, wt, None, (s, s), (1, 1), (1, 1), False,
                         (0, 0), 1, (False, True, False))),
        }
        for op, (f_ours, f_torch) in ops.items():
            us_o = timed(f_ours, iters=args.iters)
            us_t = timed(f_torch, iters=args.iters)
            tot[op][0] += us_o
            tot[op][1] += us_t
            print(f"[{si}] {B}x{Ci}x{H}x{W}->{Co}" .rjust(22)
                  + f" {op:>5} {us_o:9.1f} {us_t:9.1f} "
                  f"{flops / us_o / 1e6:8.1f} {flops / us_t / 1e6:9.1f}",
                  flush=True)
    print("---- totals over stack (us) ----")
    for op, (o, t) in tot.items():
        print(f"{op:>5}: ours {o:8.1f}  torch {t:8.1f}  ratio {t / o:5.2f}x")


if __name__ == "__main__":
    main()
