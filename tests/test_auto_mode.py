"""Auto-mode orchestration: KMeans clustering of label distributions, GMM
device selection, and profiled cut-point search drive the round end-to-end
(reference src/Server.py:300-382 semantics) on the loopback transport."""

import os

import torch

from split_learning_amd.config import load_config
from split_learning_amd.parallel.cluster import clustering_algorithm
from split_learning_amd.parallel.launch import run_loopback
from split_learning_amd.parallel.partition import partition
from split_learning_amd.parallel.selection import auto_threshold


def _profile(n_units=12, slow=False):
    exe = [5e5 if slow else 1e5] * n_units   # ns per unit
    sizes = [4096 * (n_units - i) for i in range(n_units)]
    return {"exe_time": exe, "size_data": sizes,
            "speed": 1e-9 if slow else 1e-7, "network": 1e-3}


def test_auto_mode_round(tmp_path):
    cfg = load_config(None, overrides={
        "server": {
            "global-round": 1, "clients": [2, 2], "model": "ViT",
            "data-name": "CIFAR10", "auto-mode": True, "validation": False,
            "parameters": {"load": True, "save": True},
            "data-distribution": {"num-sample": 40, "num-label": 10,
                                  "non-iid": True, "dirichlet": {"alpha": 0.3},
                                  "refresh": True},
            "cluster-selection": {"num-cluster": 2, "algorithm-cluster": "KMeans",
                                  "selection-mode": False},
        },
        "log_path": str(tmp_path), "debug_mode": False,
        "learning": {"batch-size": 8, "control-count": 2,
                     "learning-rate": 5e-4, "momentum": 0.5, "weight-decay": 0.01},
    })
    specs = [
        {"client_id": 0, "layer_id": 1, "cluster": None, "profile": _profile()},
        {"client_id": 1, "layer_id": 1, "cluster": None, "profile": _profile()},
        {"client_id": 2, "layer_id": 2, "cluster": 0, "profile": _profile()},
        {"client_id": 3, "layer_id": 2, "cluster": 1, "profile": _profile()},
    ]
    # run_loopback builds runtimes from specs; wire profiles through
    from split_learning_amd.parallel.client import ClientRuntime
    from split_learning_amd.parallel.control import InProcControl
    from split_learning_amd.parallel.data_plane import LoopbackData
    from split_learning_amd.parallel.policies import make_server
    import threading
    control = InProcControl()
    plane = LoopbackData()
    server = make_server(cfg, control, checkpoint_dir=str(tmp_path))
    threads = []
    for rec in specs:
        rt = ClientRuntime(rec["client_id"], rec["layer_id"], control, plane,
                           torch.device("cpu"), cluster=rec["cluster"],
                           profile=rec["profile"],
                           scheduler_cfg=cfg.get("scheduler"))
        t = threading.Thread(target=rt.run, daemon=True)
        threads.append((rt, t))
    for _rt, t in threads:
        t.start()
    for rt, _t in threads:
        rt.register()
    server.run()
    for _rt, t in threads:
        t.join(timeout=60)
    assert server.round == 0
    assert server.num_cluster == 2
    # auto cut computed from the profiles (not the manual config default)
    assert all(len(c) == 1 and 1 <= c[0] <= 12 for c in server.list_cut_layers)
    assert os.path.exists(os.path.join(str(tmp_path), "ViT_CIFAR10.pth"))


def test_cluster_unit():
    counts = [[10, 0, 0], [9, 1, 0], [0, 0, 10], [0, 1, 9]]
    labels, infor = clustering_algorithm(counts, 2)
    assert labels[0] == labels[1] and labels[2] == labels[3]
    assert labels[0] != labels[2]
    assert sorted(row[0] for row in infor) == [2, 2]


def test_selection_unit():
    perf = [1.0, 1.1, 0.9, 1.05, 0.001, 0.002]  # two slow stragglers
    thr = auto_threshold(perf)
    fast = [p for p in perf if p >= thr]
    assert len(fast) == 4 and all(p > 0.5 for p in fast)


def test_partition_unit():
    # heavy front layers + cheap tail: cut should avoid the first layer
    exe = [[10.0, 10.0, 1.0, 1.0]]
    net = [1.0]
    sizes = [100.0, 1.0, 50.0, 1.0]
    cut = partition(exe, net, exe, net, sizes)
    assert cut == [2]  # minimal comm at the cheap boundary
