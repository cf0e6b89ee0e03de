#!/usr/bin/env python3
"""Launch a deployment config on this node: one rank per GPU (or per CPU
process with --cpu N), rank 0 hosting the control-plane server.

    python tools/run_config.py configs/baseline3_8gpu_noniid.yaml
    python tools/run_config.py configs/baseline1_loopback_cpu.yaml   # in-process
    torchrun --nproc-per-node 8 tools/run_config.py <cfg>            # explicit
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from split_learning_amd.config import load_config
from split_learning_amd.parallel.launch import run_loopback, run_p2p_client


def main():
    cfg_path = sys.argv[1]
    cfg = load_config(cfg_path)
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if cfg["transport"]["kind"] == "loopback" or world == 1:
        if cfg["transport"]["kind"] != "loopback":
            # single process: collapse to loopback (all stages share one device)
            cfg["transport"]["kind"] = "loopback"
        device = "cuda:0" if torch.cuda.is_available() else "cpu"
        server, _ = run_loopback(cfg, device=device)
        print(f"done: round={server.round}")
        return
    import torch.distributed as dist
    rank = int(os.environ["RANK"])
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    if torch.cuda.is_available():
        device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", "0")))
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
    dist.init_process_group(backend)
    run_p2p_client(cfg, rank, world, device,
                   os.environ.get("MASTER_ADDR", "127.0.0.1"),
                   int(cfg["transport"]["master-port"]))
    dist.destroy_process_group()
    if rank == 0:
        print("done")


if __name__ == "__main__":
    main()
