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
                 scheduler_cfg: Optional[dict] = None):
    """Run the full protocol in-process. Returns (server, runtimes)."""
    control = InProcControl()
    plane = LoopbackData()
    logger = logger or Logger(f"{config['log_path']}/app.log", config["debug_mode"])
    server = Server(config, control, logger=logger, checkpoint_dir=checkpoint_dir)

    runtimes = []
    threads = []
    dev = torch.device(device)
    for rec in assign_clients(config):
        rt = ClientRuntime(rec["client_id"], rec["layer_id"], control, plane,
                           dev, cluster=rec["cluster"], logger=logger,
                           scheduler_cfg=scheduler_cfg or config.get("scheduler"))
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


def make_p2p_plane_factory(rank: int, device: torch.device):
    """plane_factory for ClientRuntime in one-process-per-GPU mode: builds a
    P2PData from the routing dict the server ships in START."""

    def factory(routing: dict) -> P2PData:
        return P2PData(
            my_rank=rank, device=device, batch=routing["batch"],
            down_peer=routing["down_peer"], up_peers=routing["up_peers"],
            act_shape_out=routing["act_shape_out"],
            act_shape_in=routing["act_shape_in"],
            grad_from_down=routing["act_shape_out"] is not None,
        )

    return factory
