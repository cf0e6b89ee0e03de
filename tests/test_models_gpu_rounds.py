"""Full protocol rounds on GPU (loopback, cuda:0) for every model family —
exercises the native kernel path inside the real server/client state machine:
conv/BN/pool (VGG16, MobileNetv1), attention/LayerNorm/GELU (ViT, KWT),
embeddings + LoRA + AdamW (BERT), checkpoint save/load."""

import os

import pytest
import torch

from split_learning_amd.config import load_config
from split_learning_amd.models import get_model_class
from split_learning_amd.parallel.launch import run_loopback

pytestmark = pytest.mark.gpu


def _cfg(tmp_path, model, data, cut, num_sample=32, batch=8, rounds=1):
    return load_config(None, overrides={
        "server": {
            "global-round": rounds, "clients": [1, 1], "model": model,
            "data-name": data, "validation": False,
            "parameters": {"load": True, "save": True},
            "data-distribution": {"num-sample": num_sample,
                                  "num-label": 10 if data != "AGNEWS" else 4,
                                  "non-iid": False, "dirichlet": {"alpha": 1},
                                  "refresh": True},
            "manual": {"cluster-mode": False, "no-cluster": {"cut-layers": [cut]}},
        },
        "log_path": str(tmp_path), "debug_mode": False,
        "learning": {"batch-size": batch, "control-count": 2,
                     "learning-rate": 5e-4, "momentum": 0.5, "weight-decay": 0.01},
    })


@pytest.mark.parametrize("model,data,cut", [
    ("VGG16", "CIFAR10", 7),
    ("MobileNetv1", "CIFAR10", 40),
    ("ViT", "CIFAR10", 6),
    ("KWT", "SPEECHCOMMANDS", 7),
    ("BERT", "AGNEWS", 2),
])
def test_gpu_round(tmp_path, model, data, cut):
    cfg = _cfg(tmp_path, model, data, cut)
    server, _ = run_loopback(cfg, device="cuda:0", checkpoint_dir=str(tmp_path))
    assert server.round == 0
    ckpt = os.path.join(str(tmp_path), f"{model}_{data}.pth")
    assert os.path.exists(ckpt)
    sd = torch.load(ckpt, weights_only=True)
    full_keys = set(get_model_class(model, data)().state_dict().keys())
    assert set(sd.keys()) == full_keys
    for v in sd.values():
        assert not torch.isnan(v.float()).any()


def test_gpu_three_stage_mnist(tmp_path):
    """BASELINE config-5 analog on one GPU: VGG16/MNIST 3-stage split
    cuts [7, 14], middle-relay on the loopback plane with per-thread streams."""
    cfg = _cfg(tmp_path, "VGG16", "MNIST", 7)
    cfg["server"]["clients"] = [1, 1, 1]
    cfg["server"]["manual"]["no-cluster"]["cut-layers"] = [7, 14]
    server, _ = run_loopback(cfg, device="cuda:0", checkpoint_dir=str(tmp_path))
    assert server.round == 0
    sd = torch.load(os.path.join(str(tmp_path), "VGG16_MNIST.pth"),
                    weights_only=True)
    assert set(sd.keys()) == set(
        get_model_class("VGG16", "MNIST")().state_dict().keys())
