"""End-to-end protocol tests on the CPU loopback transport — BASELINE config 1:
VGG16/CIFAR10, 1+1 clients, cut=7, full REGISTER->START->SYN->train->NOTIFY->
PAUSE->UPDATE->aggregate->save round, single process, no GPU."""

import os

import pytest
import torch

from split_learning_amd.config import load_config
from split_learning_amd.models import get_model_class
from split_learning_amd.parallel.launch import run_loopback


def _base_config(tmp_path, **over):
    cfg = load_config(None, overrides={
        "server": {
            "global-round": 1,
            "clients": [1, 1],
            "model": "VGG16",
            "data-name": "CIFAR10",
            "parameters": {"load": True, "save": True},
            "validation": False,
            "data-distribution": {"num-sample": 60, "num-label": 10,
                                  "non-iid": False,
                                  "dirichlet": {"alpha": 1}, "refresh": True},
            "manual": {"cluster-mode": False, "no-cluster": {"cut-layers": [7]}},
        },
        "log_path": str(tmp_path),
        "learning": {"batch-size": 16, "control-count": 3,
                     "learning-rate": 5e-4, "momentum": 0.5,
                     "weight-decay": 0.01},
        **over,
    })
    return cfg


def test_vgg16_single_round(tmp_path):
    cfg = _base_config(tmp_path)
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))
    ckpt = os.path.join(str(tmp_path), "VGG16_CIFAR10.pth")
    assert os.path.exists(ckpt)
    sd = torch.load(ckpt, weights_only=True)
    full_keys = set(get_model_class("VGG16", "CIFAR10")().state_dict().keys())
    assert set(sd.keys()) == full_keys
    assert server.round == 0


def test_vgg16_two_rounds_resume(tmp_path):
    cfg = _base_config(tmp_path)
    cfg["server"]["global-round"] = 2
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))
    assert server.round == 0
    assert os.path.exists(os.path.join(str(tmp_path), "VGG16_CIFAR10.pth"))


def test_three_stage_pipeline(tmp_path):
    """2-stage cut [7,14]: middle-stage relay (absent from the reference,
    SURVEY.md §7 step 7)."""
    cfg = _base_config(tmp_path)
    cfg["server"]["clients"] = [1, 1, 1]
    cfg["server"]["manual"]["no-cluster"]["cut-layers"] = [7, 14]
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))
    sd = torch.load(os.path.join(str(tmp_path), "VGG16_CIFAR10.pth"),
                    weights_only=True)
    full_keys = set(get_model_class("VGG16", "CIFAR10")().state_dict().keys())
    assert set(sd.keys()) == full_keys


def test_multi_client_fedavg(tmp_path):
    """2+1 clients: two stage-1 clients feed one stage-2 client; weighted
    FedAvg across the stage-1 pair."""
    cfg = _base_config(tmp_path)
    cfg["server"]["clients"] = [2, 1]
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))
    assert server.round == 0


def test_cluster_mode_two_clusters(tmp_path):
    """BASELINE config 4 shape: 2 clusters with different cut layers."""
    cfg = _base_config(tmp_path)
    cfg["server"]["clients"] = [2, 2]
    cfg["server"]["manual"] = {
        "cluster-mode": True,
        "no-cluster": {"cut-layers": [7]},
        "cluster": {"num-cluster": 2, "cut-layers": [[7], [14]],
                    "infor-cluster": [[1, 1], [1, 1]]},
    }
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))
    assert server.round == 0
    assert os.path.exists(os.path.join(str(tmp_path), "VGG16_CIFAR10.pth"))


def test_non_iid_dirichlet(tmp_path):
    cfg = _base_config(tmp_path)
    cfg["server"]["data-distribution"]["non-iid"] = True
    cfg["server"]["data-distribution"]["dirichlet"]["alpha"] = 0.5
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))
    assert server.round == 0


@pytest.mark.parametrize("model,data,cut", [("ViT", "CIFAR10", 6),
                                            ("KWT", "SPEECHCOMMANDS", 7)])
def test_other_models_round(tmp_path, model, data, cut):
    cfg = _base_config(tmp_path)
    cfg["server"]["model"] = model
    cfg["server"]["data-name"] = data
    cfg["server"]["manual"]["no-cluster"]["cut-layers"] = [cut]
    cfg["server"]["data-distribution"]["num-sample"] = 40
    server, _ = run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))
    assert server.round == 0
    assert os.path.exists(os.path.join(str(tmp_path), f"{model}_{data}.pth"))
