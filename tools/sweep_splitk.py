#!/usr/bin/env python3
"""In-process split-K sweep for the bench step (run on a GPU box)."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import bench as B  # noqa: E402


def run(target, cap, steps=128, warmup=24):
    os.environ["SLK_SPLIT_TARGET"] = str(target)
    os.environ["SLK_SPLIT_CAP"] = str(cap)
    dev = torch.device("cuda:0")
    pipe = B.ColocatedPipeline(dev, use_graphs=False)
    pipe.run(warmup)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    pipe.run(steps)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    ips = steps * B.BATCH / dt
    print(f"target={target:5d} cap={cap:3d}: {dt / steps * 1e3:7.3f} ms/step "
          f"{ips:9.1f} img/s", flush=True)
    return ips


if __name__ == "__main__":
    # interleaved A/B rounds (guide §5.4 rule 24)
    configs = [(768, 48), (1024, 64), (1280, 64), (1536, 64), (2048, 128)]
    results = {c: [] for c in configs}
    for rnd in range(2):
        for c in configs:
            results[c].append(run(*c))
    print("=== medians ===")
    for c, vals in results.items():
        vals.sort()
        print(f"target={c[0]:5d} cap={c[1]:3d}: {vals[len(vals)//2]:9.1f} img/s")
