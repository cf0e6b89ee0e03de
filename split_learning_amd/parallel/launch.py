"""Deployment launchers.

run_loopback(): server + all clients in ONE process (threads + in-process
queues) — the BASELINE.json config-1 slice ("server+clients on CPU via
in-process loopback transport") and the single-GPU deployment (all stages share
cuda:0; tensors pass by reference, zero copies).

p2p helpers: build the per-rank pieces for one-process-per-GPU runs (torchrun),
where the control plane is a TCPStore and the data plane is RCCL p2p.
"""

from __future__ import annotations

import threading
from typing import Any, Dict, List, Optional

import torch

from ..utils import Logger
from .client import ClientRuntime
from .control import InProcControl
from .data_plane import LoopbackData, P2PData
from .policies import make_server
from .server import Server


def assign_clients(config) -> List[dict]:
    """Deterministic client-id assignment: ids 0..N-1, stage-major, and cluster
    fill order following manual.cluster.infor-cluster when cluster-mode is on."""
    clients_per_stage = config["server"]["clients"]
    n_stages = len(clients_per_stage)
    manual = config["server"]["manual"]
    cluster_mode = (not config["server"]["auto-mode"]) and manual["cluster-mode"]
    infor = manual["cluster"]["infor-cluster"] if cluster_mode else None

    recs = []
    cid = 0
    for stage in range(1, n_stages + 1):
        remaining = []
        if cluster_mode:
            for k, row in enumerate(infor):
                remaining.extend([k] * row[stage - 1])
        for i in range(clients_per_stage[stage - 1]):
            cluster = remaining[i] if cluster_mode else None
            recs.append({"client_id": cid, "layer_id": stage, "cluster": cluster})
            cid += 1
    return recs


def run_loopback(config: Dict[str, Any], device: str = "cpu",
                 max_batches: Optional[int] = None, on_step=None,
                 checkpoint_dir: str = ".", logger: Optional[Logger] = None,
                 scheduler_cfg: Optional[dict] = None,
                 client_specs: Optional[List[dict]] = None):
    """Run the full protocol in-process. Returns (server, runtimes)."""
    control = InProcControl()
    plane = LoopbackData()
    logger = logger or Logger(f"{config['log_path']}/app.log", config["debug_mode"])
    server = make_server(config, control, logger=logger,
                         checkpoint_dir=checkpoint_dir)

    runtimes = []
    threads = []
    dev = torch.device(device)
    for rec in (client_specs or assign_clients(config)):
        rt = ClientRuntime(rec["client_id"], rec["layer_id"], control, plane,
                           dev, cluster=rec.get("cluster"), logger=logger,
                           scheduler_cfg=scheduler_cfg or config.get("scheduler"))
        if rec.get("out_cluster") is not None:
            rt.out_cluster = rec["out_cluster"]
        if rec.get("select") is not None:
            rt.select = rec["select"]
        runtimes.append(rt)
        t = threading.Thread(target=rt.run,
                             kwargs={"max_batches": max_batches, "on_step": on_step},
                             daemon=True)
        threads.append(t)

    for t in threads:
        t.start()
    for rt in runtimes:
        rt.register()
    server.run()
    for t in threads:
        t.join(timeout=60.0)
    return server, runtimes


def make_p2p_groups(device: Optional[torch.device] = None):
    """Create the forward/backward process groups (collective: every rank must
    call this once, in the same order, right after init_process_group).

    NCCL initialises communicators LAZILY on first use and init is a blocking
    collective — if stage-1 ranks first touch only the bwd group while stage-2
    ranks first touch the fwd group, comm init cross-deadlocks.  So both comms
    are warmed up EAGERLY here with a tiny all-reduce every rank joins."""
    import torch.distributed as dist
    world = list(range(dist.get_world_size()))
    group_fwd = dist.new_group(world)
    group_bwd = dist.new_group(world)
    if dist.get_backend() == "nccl":
        dev = device if device is not None else torch.device(
            "cuda", torch.cuda.current_device())
        t = torch.zeros(1, device=dev)
        dist.all_reduce(t, group=group_fwd)
        dist.all_reduce(t, group=group_bwd)
        torch.cuda.synchronize()
    return group_fwd, group_bwd


def make_p2p_plane_factory(rank: int, device: torch.device, group_fwd, group_bwd,
                           depth: int = 4):
    """plane_factory for ClientRuntime in one-process-per-GPU mode: builds a
    P2PData from the routing dict the server ships in START."""

    def factory(routing: dict) -> P2PData:
        return P2PData(
            my_rank=rank, device=device, batch=routing["batch"],
            down_peer=routing["down_peer"], up_peers=routing["up_peers"],
            act_shape_out=routing["act_shape_out"],
            act_shape_in=routing["act_shape_in"],
            grad_from_down=routing["act_shape_out"] is not None,
            group_fwd=group_fwd, group_bwd=group_bwd, depth=depth,
        )

    return factory


def run_p2p_client(config: Dict[str, Any], rank: int, world: int,
                   device: torch.device, store_addr: str, store_port: int,
                   max_batches: Optional[int] = None, on_step=None,
                   checkpoint_dir: str = ".", logger=None):
    """One-process-per-GPU worker: rank 0 also runs the server in a thread.

    Stage assignment: ranks are stage-major in registration order — the first
    clients[0] ranks are stage 1, the next clients[1] stage 2, etc.
    client_id == rank (the p2p routing the server computes uses client ids as
    ranks).
    """
    from .control import StoreControl

    clients_per_stage = config["server"]["clients"]
    assert sum(clients_per_stage) == world, \
        f"sum(clients)={sum(clients_per_stage)} must equal world size {world}"
    layer_id = None
    acc = 0
    for stage, n in enumerate(clients_per_stage, start=1):
        if rank < acc + n:
            layer_id = stage
            break
        acc += n
    group_fwd, group_bwd = make_p2p_groups(device)
    control = StoreControl.create(store_addr, store_port, is_server=(rank == 0))
    logger = logger or Logger(f"{config['log_path']}/app.log", config["debug_mode"])

    server_thread = None
    server = None
    if rank == 0:
        # the server thread gets its OWN store connection (a TCPStore object is
        # a single socket and must not be shared across threads)
        server_control = StoreControl.create(store_addr, store_port,
                                             is_server=False)
        server = make_server(config, server_control, logger=logger,
                             checkpoint_dir=checkpoint_dir)
        server_thread = threading.Thread(target=server.run, daemon=True)
        server_thread.start()

    rt = ClientRuntime(
        rank, layer_id, control, None, device, logger=logger,
        scheduler_cfg=config.get("scheduler"),
        plane_factory=make_p2p_plane_factory(
            rank, device, group_fwd, group_bwd,
            depth=max(2, int(config["learning"].get("control-count", 3)) + 1)))
    rt.register()
    rt.run(max_batches=max_batches, on_step=on_step)
    if server_thread is not None:
        server_thread.join(timeout=120.0)
    import torch.distributed as dist
    if dist.is_initialized():
        dist.barrier()  # keep rank 0's master store alive until all ranks finish
    return server, rt
