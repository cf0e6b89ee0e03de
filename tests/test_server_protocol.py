"""Server state-machine unit tests: drive REGISTER/NOTIFY/UPDATE messages
directly (no client threads) and assert protocol behaviour, including the
NaN round-skip semantics (reference src/Server.py:155-212)."""

import os
import threading

import torch

from split_learning_amd.config import load_config
from split_learning_amd.models import build_partition
from split_learning_amd.parallel.control import InProcControl
from split_learning_amd.parallel.server import Server


def _cfg(tmp_path, rounds=1, validation=False):
    return load_config(None, overrides={
        "server": {
            "global-round": rounds, "clients": [1, 1], "model": "ViT",
            "data-name": "CIFAR10", "validation": validation,
            "parameters": {"load": True, "save": True},
            "data-distribution": {"num-sample": 20, "num-label": 10,
                                  "non-iid": False, "dirichlet": {"alpha": 1},
                                  "refresh": True},
            "manual": {"cluster-mode": False, "no-cluster": {"cut-layers": [6]}},
        },
        "log_path": str(tmp_path), "debug_mode": False,
        "learning": {"batch-size": 8, "control-count": 3,
                     "learning-rate": 5e-4, "momentum": 0.5, "weight-decay": 0.01},
    })


class FakeClients:
    """Responds to START/SYN/PAUSE like two well-behaved clients, with
    scriptable UPDATE results."""

    def __init__(self, control, results=(True, True)):
        self.control = control
        self.results = results
        self.started = {}
        self.stopped = set()

    def pump(self):
        done = False
        while not done:
            for cid, stage in ((0, 1), (1, 2)):
                msg = self.control.recv(f"client_{cid}", block=False)
                if msg is None:
                    continue
                a = msg["action"]
                if a == "START":
                    self.started[cid] = msg
                    self.control.send("server", {"action": "READY",
                                                 "client_id": cid,
                                                 "layer_id": stage})
                elif a == "SYN":
                    if stage == 1:
                        self.control.send("server", {
                            "action": "NOTIFY", "client_id": cid,
                            "layer_id": 1, "cluster": 0})
                elif a == "PAUSE":
                    sd = build_partition(
                        "ViT", "CIFAR10",
                        self.started[cid]["layers"]).state_dict()
                    sd = {k: v.cpu() for k, v in sd.items()}
                    self.control.send("server", {
                        "action": "UPDATE", "client_id": cid, "layer_id": stage,
                        "cluster": 0, "result": self.results[cid], "size": 20,
                        "parameters": sd, "message": "m"})
                elif a == "STOP":
                    self.stopped.add(cid)
                    if len(self.stopped) == 2:
                        done = True


def _run(server, clients):
    t = threading.Thread(target=clients.pump, daemon=True)
    t.start()
    server.run()
    t.join(timeout=30)


def test_happy_round_saves_checkpoint(tmp_path):
    control = InProcControl()
    server = Server(_cfg(tmp_path), control, checkpoint_dir=str(tmp_path))
    clients = FakeClients(control)
    for cid, stage in ((0, 1), (1, 2)):
        control.send("server", {"action": "REGISTER", "client_id": cid,
                                "layer_id": stage, "profile": {}, "cluster": None})
    _run(server, clients)
    assert server.round == 0
    assert os.path.exists(os.path.join(str(tmp_path), "ViT_CIFAR10.pth"))
    # stage-2 PAUSE happens only after the stage-1 NOTIFY -> both got PAUSE+UPDATE
    assert clients.stopped == {0, 1}


def test_nan_round_skips_save(tmp_path):
    """A client reporting result=False (NaN detected) makes the server skip
    aggregation/saving for the round (reference src/Server.py:162-196)."""
    control = InProcControl()
    server = Server(_cfg(tmp_path), control, checkpoint_dir=str(tmp_path))
    clients = FakeClients(control, results=(True, False))
    for cid, stage in ((0, 1), (1, 2)):
        control.send("server", {"action": "REGISTER", "client_id": cid,
                                "layer_id": stage, "profile": {}, "cluster": None})
    _run(server, clients)
    assert server.round == 0  # round still consumed
    assert not os.path.exists(os.path.join(str(tmp_path), "ViT_CIFAR10.pth"))


def test_checkpoint_sliced_into_start(tmp_path):
    """With an existing .pth, START carries the partition's slice of it."""
    full = build_partition("ViT", "CIFAR10", [0, 0])
    torch.save(full.state_dict(), os.path.join(str(tmp_path), "ViT_CIFAR10.pth"))
    control = InProcControl()
    server = Server(_cfg(tmp_path), control, checkpoint_dir=str(tmp_path))
    clients = FakeClients(control)
    for cid, stage in ((0, 1), (1, 2)):
        control.send("server", {"action": "REGISTER", "client_id": cid,
                                "layer_id": stage, "profile": {}, "cluster": None})
    _run(server, clients)
    s1_params = clients.started[0]["parameters"]
    assert s1_params is not None
    expect = set(build_partition("ViT", "CIFAR10", [0, 6]).state_dict().keys())
    assert set(s1_params.keys()) == expect
    for k in expect:
        assert torch.equal(s1_params[k], full.state_dict()[k])
