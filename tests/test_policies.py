"""Variant scheduling policy tests (loopback, CPU) — one round (or several for
FLEX periodicity) of each of the five reference forks' semantics."""

import os

import torch

from split_learning_amd.config import load_config
from split_learning_amd.parallel.launch import run_loopback


def _cfg(tmp_path, policy, model="ViT", data="CIFAR10", clients=(1, 1), cut=6,
         rounds=1, num_sample=40, **sched):
    return load_config(None, overrides={
        "server": {
            "global-round": rounds, "clients": list(clients), "model": model,
            "data-name": data, "validation": False,
            "parameters": {"load": True, "save": True},
            "data-distribution": {"num-sample": num_sample, "num-label": 10,
                                  "non-iid": False, "dirichlet": {"alpha": 1},
                                  "refresh": True},
            "manual": {"cluster-mode": False, "no-cluster": {"cut-layers": [cut]}},
        },
        "log_path": str(tmp_path), "debug_mode": False,
        "learning": {"batch-size": 8, "control-count": 3,
                     "learning-rate": 5e-4, "momentum": 0.5, "weight-decay": 0.01},
        "scheduler": {"policy": policy, "recompute": True, **sched},
    })


def _ckpt(tmp_path, model="ViT", data="CIFAR10"):
    return os.path.join(str(tmp_path), f"{model}_{data}.pth")


def test_vanilla_sequential(tmp_path):
    """3 edge devices one-at-a-time + 1 resident stage-2 device; epochs=2."""
    cfg = _cfg(tmp_path, "vanilla", clients=(3, 1), epochs=2)
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))
    assert server.round == 0
    assert os.path.exists(_ckpt(tmp_path))
    assert len(server._edge_parts) == 3


def test_vanilla_time_limit(tmp_path):
    cfg = _cfg(tmp_path, "vanilla", clients=(2, 1), epochs=50)
    cfg["scheduler"]["limited-time"] = 2.0  # seconds per device
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))
    assert server.round == 0


def test_cluster_fsl_sequential_clusters(tmp_path):
    cfg = _cfg(tmp_path, "cluster_fsl", clients=(2, 2))
    cfg["server"]["manual"] = {
        "cluster-mode": True,
        "no-cluster": {"cut-layers": [6]},
        "cluster": {"num-cluster": 2, "cut-layers": [[6], [6]],
                    "infor-cluster": [[1, 1], [1, 1]]},
    }
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))
    assert server.round == 0
    assert len(server._cluster_avgs) == 2
    assert os.path.exists(_ckpt(tmp_path))


def test_dcsl_sda_and_sync(tmp_path):
    """2 stage-1 clients -> 1 stage-2 with SDA concat batching, sync first layer."""
    cfg = _cfg(tmp_path, "dcsl", clients=(2, 1), **{"local-round": 1})
    cfg["server"]["manual"] = {
        "cluster-mode": True,
        "no-cluster": {"cut-layers": [6]},
        "cluster": {"num-cluster": 1, "cut-layers": [[6]],
                    "infor-cluster": [[2, 1]]},
    }
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))
    assert server.round == 0
    assert os.path.exists(_ckpt(tmp_path))


def test_flex_periodic_aggregation(tmp_path):
    """t-c=2: rounds 1 and 3 skip the upload, rounds 2 and 4 aggregate."""
    cfg = _cfg(tmp_path, "flex", clients=(1, 1), rounds=4,
               **{"t-c": 2, "t-g": 2})
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))
    assert server.round == 0
    assert os.path.exists(_ckpt(tmp_path))


def test_flex_select_reject(tmp_path):
    """A client registering with select=False is rejected up front."""
    cfg = _cfg(tmp_path, "flex", clients=(2, 1))
    specs = [
        {"client_id": 0, "layer_id": 1, "cluster": None, "select": True},
        {"client_id": 1, "layer_id": 1, "cluster": None, "select": False},
        {"client_id": 2, "layer_id": 2, "cluster": None, "select": True},
    ]
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path),
                             client_specs=specs)
    assert server.round == 0
    rejected = [c for c in server.list_clients if not c["train"]]
    assert len(rejected) == 1 and rejected[0]["client_id"] == 1


def test_2ls_fedasync(tmp_path):
    """Two out-clusters sequential + FedAsync fold into the global model."""
    cfg = _cfg(tmp_path, "2ls", clients=(2, 2))
    cfg["server"]["manual"] = {
        "cluster-mode": True,
        "no-cluster": {"cut-layers": [6]},
        "cluster": {"num-cluster": 2, "cut-layers": [[6], [6]],
                    "infor-cluster": [[1, 1], [1, 1]]},
    }
    specs = [
        {"client_id": 0, "layer_id": 1, "cluster": 0, "out_cluster": 0},
        {"client_id": 1, "layer_id": 1, "cluster": 1, "out_cluster": 1},
        {"client_id": 2, "layer_id": 2, "cluster": 0, "out_cluster": 0},
        {"client_id": 3, "layer_id": 2, "cluster": 1, "out_cluster": 1},
    ]
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path),
                             client_specs=specs)
    assert server.round == 0
    assert server._arrival == 2  # two FedAsync folds
    assert os.path.exists(_ckpt(tmp_path))


def test_vanilla_vgg16(tmp_path):
    """Vanilla with the flagship model, 2 sequential edges."""
    cfg = _cfg(tmp_path, "vanilla", model="VGG16", cut=7, clients=(2, 1),
               num_sample=30)
    cfg["learning"]["batch-size"] = 8
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))
    assert server.round == 0
    assert os.path.exists(_ckpt(tmp_path, "VGG16"))
