"""Config loader, schema-compatible with the reference config.yaml.

The reference loads one YAML file in both server.py:12-13 and client.py:20-21 and
indexes it as a plain dict (``config["server"]["clients"]`` etc.).  We keep the
exact same key layout (see reference config.yaml:1-55) but add:

* defaults for every key, so partial configs work;
* an optional ``device`` section (MI355X-specific: streams, dtype, graph capture);
* a ``transport`` selection (``loopback`` | ``rccl``) replacing the reference's
  ``rabbit`` section (which is accepted and ignored for compatibility).
"""

from __future__ import annotations

import copy
import os
from typing import Any, Dict, Optional

import yaml

DEFAULT_CONFIG: Dict[str, Any] = {
    "name": "Split Learning",
    "server": {
        "global-round": 1,
        # clients[i] = number of clients at stage i+1 (reference config.yaml:4-6)
        "clients": [1, 1],
        "auto-mode": False,
        "model": "VGG16",
        "data-name": "CIFAR10",
        "parameters": {"load": True, "save": True},
        "validation": True,
        "data-distribution": {
            "non-iid": False,
            "num-sample": 5000,
            "num-label": 10,
            "dirichlet": {"alpha": 1},
            "refresh": True,
        },
        "random-seed": 1,
        "manual": {
            "cluster-mode": False,
            "no-cluster": {"cut-layers": [1]},
            "cluster": {
                "num-cluster": 1,
                "cut-layers": [[1]],
                "infor-cluster": [[1, 1]],
            },
        },
        "cluster-selection": {
            "num-cluster": 1,
            "algorithm-cluster": "KMeans",
            "selection-mode": False,
        },
    },
    # Accepted for reference compatibility; unused by the RCCL/loopback transports.
    "rabbit": {
        "address": "127.0.0.1",
        "username": "admin",
        "password": "admin",
        "virtual-host": "/",
    },
    "log_path": ".",
    "debug_mode": False,
    "learning": {
        "learning-rate": 0.0005,
        "weight-decay": 0.01,
        "momentum": 0.5,
        "batch-size": 32,
        "control-count": 3,
    },
    # MI355X-native additions (absent keys fall back to these defaults).
    "transport": {
        # "loopback": in-process queues (CPU tests / single-process runs)
        # "rccl": torch.distributed NCCL(=RCCL) p2p over xGMI, one process per GPU
        "kind": "loopback",
        "master-addr": "127.0.0.1",
        "master-port": 29571,
        # FedAvg path: "control" ships state dicts through the control plane
        # (reference semantics); "rccl" all-reduces weight*size over the
        # stage-group communicator (sequential policies force "control")
        "fedavg": "control",
        # START parameter delivery: "auto" broadcasts the full model over
        # RCCL when safe (concurrent policy, all ranks accepted, dist live),
        # "control" always ships per-client slices, "rccl" forces broadcast
        "params": "auto",
    },
    "device": {
        "dtype": "float32",
        "graphs": False,      # capture the per-microbatch step in a hipGraph
        "side-streams": True, # p2p on dedicated HIP streams overlapped with compute
    },
    # Variant scheduling policy (reference forks under other/ become policies):
    # "main" | "vanilla" | "cluster_fsl" | "dcsl" | "flex" | "2ls"
    "scheduler": {
        "policy": "main",
        # stage-1 backward recomputes the forward with current weights
        # (reference semantics, src/train/VGG16.py:89-91); False stashes the
        # autograd graph (valid only with control-count 1)
        "recompute": True,
        # Vanilla_SL / Cluster_FSL extras:
        "epochs": 1,
        "limited-time": None,         # wall-clock seconds cap per round (Vanilla)
        "clip-grad-norm": None,
        # DCSL extras:
        "local-round": 1,
        # FLEX extras (periodic aggregation):
        "t-c": 1,
        "t-g": 1,
    },
}


def _merge(base: Dict[str, Any], override: Dict[str, Any]) -> Dict[str, Any]:
    out = copy.deepcopy(base)
    for k, v in override.items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _merge(out[k], v)
        else:
            out[k] = copy.deepcopy(v)
    return out


def load_config(path: Optional[str] = None, overrides: Optional[Dict[str, Any]] = None) -> Dict[str, Any]:
    """Load a reference-compatible config.yaml, filling defaults.

    ``path=None`` returns pure defaults (optionally merged with ``overrides``).
    """
    cfg: Dict[str, Any] = copy.deepcopy(DEFAULT_CONFIG)
    if path is not None and os.path.exists(path):
        with open(path, "r") as f:
            user = yaml.safe_load(f) or {}
        cfg = _merge(cfg, user)
    if overrides:
        cfg = _merge(cfg, overrides)
    validate_config(cfg)
    return cfg


def validate_config(cfg: Dict[str, Any]) -> None:
    srv = cfg["server"]
    clients = srv["clients"]
    if not isinstance(clients, list) or not clients or any(int(c) < 0 for c in clients):
        raise ValueError(f"server.clients must be a list of non-negative ints, got {clients!r}")
    if srv["model"] not in ("VGG16", "BERT", "KWT", "MobileNetv1", "ViT"):
        raise ValueError(f"Unknown model {srv['model']!r}")
    if not srv["auto-mode"]:
        man = srv["manual"]
        if man["cluster-mode"]:
            ncl = man["cluster"]["num-cluster"]
            cls_cuts = man["cluster"]["cut-layers"]
            if len(cls_cuts) != ncl:
                raise ValueError("manual.cluster.cut-layers must have num-cluster entries")
            for cuts in cls_cuts:
                _check_cuts(cuts, len(clients))
        else:
            _check_cuts(man["no-cluster"]["cut-layers"], len(clients))
    pol = cfg["scheduler"]["policy"]
    if pol not in ("main", "vanilla", "cluster_fsl", "dcsl", "flex", "2ls"):
        raise ValueError(f"Unknown scheduler policy {pol!r}")


def _check_cuts(cuts, n_stages: int) -> None:
    if len(cuts) != n_stages - 1:
        raise ValueError(
            f"cut-layers {cuts!r} must have len(clients)-1 = {n_stages - 1} entries "
            f"(stage k trains layers (cut[k-1], cut[k]])"
        )
    if any(cuts[i] > cuts[i + 1] for i in range(len(cuts) - 1)):
        raise ValueError(f"cut-layers must be non-decreasing, got {cuts!r}")


def stage_ranges(cuts, n_stages: int, total_units: int):
    """[start, end] unit range per stage, matching reference src/Server.py:221-228.

    Stage 1: [0, cuts[0]]; stage k (middle): [cuts[k-2], cuts[k-1]];
    last stage: [cuts[-1], -1] where -1 means "through the final unit".
    """
    out = []
    for stage in range(1, n_stages + 1):
        if stage == 1:
            out.append([0, cuts[0]])
        elif stage == n_stages:
            out.append([cuts[-1], -1])
        else:
            out.append([cuts[stage - 2], cuts[stage - 1]])
    return out
