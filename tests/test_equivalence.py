"""Numerics equivalence: the split pipeline (loopback, control-count 1) produces
EXACTLY the same final weights as a hand-rolled loop implementing split-SGD
directly — same init, same data order, same optimizer math (SURVEY.md §4 test
strategy item 5).  ViT is dropout-free, so CPU fp32 runs are bit-deterministic.
"""

import os

import torch

from split_learning_amd.config import load_config
from split_learning_amd.data import data_loader
from split_learning_amd.models import build_partition
from split_learning_amd.parallel.launch import run_loopback
from split_learning_amd.parallel.optim import FusedSGD

CUT = 6
LR = 5e-4
MOM = 0.5
BATCH = 8
NUM_SAMPLE = 40


def _make_ckpt(path):
    torch.manual_seed(42)
    full = build_partition("ViT", "CIFAR10", [0, 0])
    torch.save(full.state_dict(), path)
    return full.state_dict()


def _hand_rolled(init_sd, label_count):
    s1 = build_partition("ViT", "CIFAR10", [0, CUT])
    s2 = build_partition("ViT", "CIFAR10", [CUT, -1])
    s1.load_state_dict({k: init_sd[k] for k in s1.state_dict()})
    s2.load_state_dict({k: init_sd[k] for k in s2.state_dict()})
    s1.train()
    s2.train()
    o1 = FusedSGD(s1.parameters(), lr=LR, momentum=MOM)
    o2 = FusedSGD(s2.parameters(), lr=LR, momentum=MOM)
    loader = data_loader("CIFAR10", BATCH, label_count, train=True, seed=0)
    for x, y in loader:
        # stage-1 forward (no grad) -> send
        with torch.no_grad():
            act = s1(x)
        # stage-2: fwd + CE + bwd + step; grad of the cut activation flows back
        act_in = act.detach().requires_grad_(True)
        loss = torch.nn.functional.cross_entropy(s2(act_in), y)
        loss.backward()
        o2.step()
        # stage-1: recompute with grad, backward from cut gradient, step
        out = s1(x)
        out.backward(gradient=act_in.grad)
        o1.step()
    merged = {}
    merged.update({k: v.detach().clone() for k, v in s1.state_dict().items()})
    merged.update({k: v.detach().clone() for k, v in s2.state_dict().items()})
    return merged


def test_pipeline_equals_hand_rolled(tmp_path):
    ckpt = os.path.join(str(tmp_path), "ViT_CIFAR10.pth")
    init_sd = _make_ckpt(ckpt)

    label_count = [NUM_SAMPLE // 10] * 10
    expected = _hand_rolled(init_sd, label_count)

    cfg = load_config(None, overrides={
        "server": {
            "global-round": 1, "clients": [1, 1], "model": "ViT",
            "data-name": "CIFAR10", "validation": False,
            "parameters": {"load": True, "save": True},
            "data-distribution": {"num-sample": NUM_SAMPLE, "num-label": 10,
                                  "non-iid": False, "dirichlet": {"alpha": 1},
                                  "refresh": True},
            "manual": {"cluster-mode": False, "no-cluster": {"cut-layers": [CUT]}},
        },
        "log_path": str(tmp_path), "debug_mode": False,
        "learning": {"batch-size": BATCH, "control-count": 1,
                     "learning-rate": LR, "momentum": MOM, "weight-decay": 0.01},
        "scheduler": {"policy": "main", "recompute": True},
    })
    run_loopback(cfg, device="cpu", checkpoint_dir=str(tmp_path))

    got = torch.load(ckpt, weights_only=True)
    assert set(got.keys()) == set(expected.keys())
    for k in expected:
        assert torch.allclose(got[k].float(), expected[k].float(),
                              atol=1e-6, rtol=1e-6), f"mismatch at {k}"


def test_loss_decreases_over_round(tmp_path):
    """Sanity: a few rounds of pipeline training reduce CE loss on the training
    distribution (synthetic but learnable label structure is absent, so compare
    against the initial-model loss on the SAME fixed batch: optimizer must
    reduce it)."""
    torch.manual_seed(0)
    s1 = build_partition("ViT", "CIFAR10", [0, CUT]).train()
    s2 = build_partition("ViT", "CIFAR10", [CUT, -1]).train()
    o1 = FusedSGD(s1.parameters(), lr=5e-3, momentum=0.9)
    o2 = FusedSGD(s2.parameters(), lr=5e-3, momentum=0.9)
    x = torch.randn(16, 3, 32, 32)
    y = torch.randint(0, 10, (16,))
    losses = []
    for _ in range(12):
        act = s1(x).detach().requires_grad_(True)
        loss = torch.nn.functional.cross_entropy(s2(act), y)
        losses.append(float(loss))
        loss.backward()
        o2.step()
        out = s1(x)
        out.backward(gradient=act.grad)
        o1.step()
    assert losses[-1] < losses[0] * 0.8, losses
