#!/usr/bin/env python3
"""RCCL-backend smoke for a 1-GPU lease (VERDICT round-1 item 2: de-risk the
first real multi-GPU run).  Exercises every nccl-backend assumption the
engine makes that multi-process CPU tests (gloo) cannot:

  1. single-rank nccl init_process_group over a TCP store;
  2. make_p2p_groups eager communicator warmup (new_group x2 + all_reduce on
     each — the exact call sequence every bench/engine rank replays);
  3. Work.is_completed() polling semantics on nccl (RecvRing relies on it:
     data_plane.py) — verified here with a self-contained isend/irecv pair
     via two groups on world 1 is impossible, so we verify is_completed on
     an all_reduce work handle instead;
  4. StoreControl end-to-end (chunked blob round trip on the rendezvous
     TCPStore);
  5. grouped all-reduce FedAvg math (allreduce_fedavg_ on a world-1 group ==
     identity) on device tensors.

Run on the GPU box:   python tools/rccl_smoke.py
Expected output:      "RCCL SMOKE OK" + per-phase timings.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    t0 = time.perf_counter()

    def phase(name):
        print(f"[smoke +{time.perf_counter() - t0:6.2f}s] {name}", flush=True)

    assert torch.cuda.is_available(), "rccl_smoke needs a GPU"
    device = torch.device("cuda", 0)
    torch.cuda.set_device(device)

    import torch.distributed as dist
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29471")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    phase("init_process_group(nccl) ...")
    dist.init_process_group("nccl")
    assert dist.get_backend() == "nccl"
    phase("init ok")

    # 2. the engine's eager communicator warmup
    from split_learning_amd.parallel.launch import make_p2p_groups
    gf, gb = make_p2p_groups(device)
    phase("make_p2p_groups warmup ok (fwd+bwd comms live)")

    # 3. Work.is_completed() semantics on a real RCCL work object
    t = torch.ones(1 << 20, device=device)
    w = dist.all_reduce(t, group=gf, async_op=True)
    spins = 0
    while not w.is_completed():
        spins += 1
        if spins > 50_000_000:
            raise RuntimeError("is_completed() never turned true on RCCL")
    torch.cuda.synchronize()
    assert float(t[0]) == 1.0
    phase(f"async all_reduce is_completed() ok after {spins} polls")

    # 4. StoreControl chunked-blob round trip
    from split_learning_amd.parallel.control import StoreControl
    ctl_srv = StoreControl.create("127.0.0.1", 29473, is_server=True)
    ctl_cli = StoreControl.create("127.0.0.1", 29473, is_server=False)
    big = {"action": "START", "parameters": {f"k{i}": torch.randn(64, 64)
                                             for i in range(32)}}
    ctl_srv.send("client_0", big)
    got = ctl_cli.recv("client_0", block=True, timeout=60.0)
    assert got["action"] == "START" and len(got["parameters"]) == 32
    assert torch.equal(got["parameters"]["k3"], big["parameters"]["k3"])
    phase("StoreControl chunked blob round trip ok (~1 MB)")

    # 5. FedAvg all-reduce on device (world-1 group: identity)
    from split_learning_amd.parallel.fedavg import allreduce_fedavg_
    m = torch.nn.Linear(128, 128).to(device)
    before = {k: v.detach().clone() for k, v in m.state_dict().items()}
    allreduce_fedavg_(m, 17.0, group=gf)
    after = m.state_dict()
    for k in before:
        assert torch.allclose(before[k], after[k], atol=1e-6), k
    phase("allreduce_fedavg_ on device ok")

    dist.barrier()
    dist.destroy_process_group()
    phase("teardown ok")
    print("RCCL SMOKE OK", flush=True)


if __name__ == "__main__":
    sys.exit(main())
