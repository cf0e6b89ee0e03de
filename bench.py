#!/usr/bin/env python3
"""Flagship benchmark: images/sec for VGG16/CIFAR10 split training at cut=7.

BASELINE.json metric: "images/sec (whole node) VGG16/CIFAR10 cut=7 at 1/2/4/8
MI355X".  The reference publishes no numbers, so this measures
our engine's own headline config: batch 32 fp32 (the reference's compute
dtype — pure fp32 PyTorch), SGD momentum 0.5, control-count 3, synthetic
CIFAR10-shaped data, random-init weights.

Topology (weak scaling):
  N=1: both stages colocated on cuda:0 (loopback plane, zero-copy).
  N>=2 (torchrun, one rank per GPU): ranks [0, N/2) are stage-1 clients,
  ranks [N/2, N) stage-2; pipeline pairs (r, r+N/2) exchange cut-layer
  activations/gradients via RCCL p2p over xGMI.

One "step" = one microbatch (batch 32) through forward+backward+optimizer on
BOTH stages of every pipeline.  Timed region: barrier+sync, run K steps to
full pipeline drain, barrier+sync; value = K * 32 * n_pipelines / elapsed_max.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch


def log(msg):
    print(msg, file=sys.stderr, flush=True)


BATCH = 32
CUT = 7
LR = 5e-4
MOMENTUM = 0.5
CONTROL_COUNT = 3


def build_stage(layers, device):
    from split_learning_amd.models import build_partition
    from split_learning_amd.parallel.optim import FusedSGD
    torch.manual_seed(1234 + layers[0])
    model = build_partition("VGG16", "CIFAR10", layers).to(device).train()
    opt = FusedSGD(model.parameters(), lr=LR, momentum=MOMENTUM)
    return model, opt


def make_batches(device, n, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    x = torch.randn(n, BATCH, 3, 32, 32, generator=g).to(device)
    y = torch.randint(0, 10, (n, BATCH), generator=g).to(device)
    return x, y


def run_colocated(device, steps):
    """Both stages on one GPU, serial in-process pipeline (N=1 path)."""
    from split_learning_amd.ops import functional as hf
    s1_model, s1_opt = build_stage([0, CUT], device)
    s2_model, s2_opt = build_stage([CUT, -1], device)
    xs, ys = make_batches(device, steps)
    nan_flag = torch.zeros((), dtype=torch.bool, device=device)

    def ce(logits, labels):
        if logits.is_cuda:
            return hf.cross_entropy(logits, labels)
        return torch.nn.functional.cross_entropy(logits, labels)

    for i in range(steps):
        x, y = xs[i], ys[i]
        with torch.no_grad():
            act = s1_model(x)
        # stage 2: fwd + loss + bwd + step
        act_in = act.detach().requires_grad_(True)
        s2_opt.zero_grad()
        logits = s2_model(act_in)
        loss = ce(logits, y)
        nan_flag |= torch.isnan(loss)
        loss.backward()
        s2_opt.step()
        # stage 1: recompute fwd with grad, bwd from cut gradient, step
        s1_opt.zero_grad()
        out = s1_model(x)
        out.backward(gradient=act_in.grad)
        s1_opt.step()
    return nan_flag


def run_distributed(rank, world, device, steps):
    """One rank per GPU; pair (r, r + world/2) forms a pipeline."""
    import torch.distributed as dist
    from split_learning_amd.ops import functional as hf

    half = world // 2
    is_first = rank < half
    peer = rank + half if is_first else rank - half
    act_shape = (BATCH, 64, 16, 16)  # VGG16 cut=7 boundary ([B,64,16,16])

    if is_first:
        model, opt = build_stage([0, CUT], device)
        xs, ys = make_batches(device, steps, seed=rank)
        grad_buf = torch.zeros(act_shape, device=device)
        inflight = []
        for i in range(steps):
            x, y = xs[i], ys[i]
            with torch.no_grad():
                act = s_out = model(x)
            dist.send(act.contiguous(), dst=peer)
            dist.send(y, dst=peer)
            inflight.append(x)
            if len(inflight) >= CONTROL_COUNT or i == steps - 1:
                while inflight:
                    dist.recv(grad_buf, src=peer)
                    xo = inflight.pop(0)
                    opt.zero_grad()
                    out = model(xo)
                    out.backward(gradient=grad_buf)
                    opt.step()
                    if len(inflight) < CONTROL_COUNT - 1 and i < steps - 1:
                        break
        return torch.zeros((), dtype=torch.bool, device=device)
    else:
        model, opt = build_stage([CUT, -1], device)
        act_buf = torch.zeros(act_shape, device=device)
        y_buf = torch.zeros(BATCH, dtype=torch.int64, device=device)
        nan_flag = torch.zeros((), dtype=torch.bool, device=device)
        for i in range(steps):
            dist.recv(act_buf, src=peer)
            dist.recv(y_buf, src=peer)
            act = act_buf.clone().requires_grad_(True)
            opt.zero_grad()
            logits = model(act)
            loss = hf.cross_entropy(logits, y_buf) if logits.is_cuda else \
                torch.nn.functional.cross_entropy(logits, y_buf)
            nan_flag |= torch.isnan(loss)
            loss.backward()
            opt.step()
            dist.send(act.grad.contiguous(), dst=peer)
        return nan_flag


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=64)
    ap.add_argument("--warmup", type=int, default=16)
    ap.add_argument("--device", default=None)
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    n_gpus = max(args.gpus, world)

    have_gpu = torch.cuda.is_available()
    if args.device:
        device = torch.device(args.device)
    elif have_gpu:
        device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", "0")))
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
        log("WARNING: no GPU visible; CPU fallback run (numbers not comparable)")

    dist_mode = world > 1
    if dist_mode:
        import torch.distributed as dist
        backend = "nccl" if have_gpu else "gloo"
        dist.init_process_group(backend)
        if world % 2 != 0:
            raise SystemExit("bench: WORLD_SIZE must be even for the split pipeline")

    def sync():
        if have_gpu:
            torch.cuda.synchronize()
        if dist_mode:
            import torch.distributed as dist
            dist.barrier()
            if have_gpu:
                torch.cuda.synchronize()

    runner = (lambda n: run_distributed(rank, world, device, n)) if dist_mode \
        else (lambda n: run_colocated(device, n))

    log(f"[bench] warmup {args.warmup} steps (rank {rank}/{world}, {device})")
    nan_w = runner(args.warmup)
    sync()
    t0 = time.perf_counter()
    nan_flag = runner(args.steps)
    sync()
    t1 = time.perf_counter()
    elapsed = t1 - t0

    if dist_mode:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if have_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    n_pipelines = max(world // 2, 1)
    images = args.steps * BATCH * n_pipelines
    value = images / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if bool(nan_flag.item()):
        log("WARNING: NaN loss during timed region")

    if rank == 0:
        print(json.dumps({
            "metric": "images/sec (whole node) VGG16/CIFAR10 cut=7",
            "value": round(value, 2),
            "unit": "images/sec",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 4),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic (random CIFAR10-shaped, random-init weights)",
            "config": {"model": "VGG16_CIFAR10", "global_batch": BATCH * n_pipelines,
                       "image": "3x32x32", "cut_layer": CUT,
                       "optimizer": "SGD(lr=5e-4, momentum=0.5)",
                       "control_count": CONTROL_COUNT,
                       "parallelism": f"split2 x dp{n_pipelines}"},
        }), flush=True)

    if dist_mode:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
