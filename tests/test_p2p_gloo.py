"""Multi-process p2p transport tests on CPU (gloo backend, world_size 2/3) —
the same P2PData/StoreControl path that runs RCCL-over-xGMI on the GPU node,
proving the distributed round protocol end to end without a GPU."""

import os
import socket

import pytest
import torch
import torch.multiprocessing as mp


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _cfg(tmpdir, clients, cuts, num_sample=48, batch=16, fedavg="control"):
    from split_learning_amd.config import load_config
    return load_config(None, overrides={
        "transport": {"fedavg": fedavg},
        "server": {
            "global-round": 1, "clients": clients, "model": "VGG16",
            "data-name": "CIFAR10", "validation": False,
            "parameters": {"load": True, "save": True},
            "data-distribution": {"num-sample": num_sample, "num-label": 10,
                                  "non-iid": False, "dirichlet": {"alpha": 1},
                                  "refresh": True},
            "manual": {"cluster-mode": False, "no-cluster": {"cut-layers": cuts}},
        },
        "log_path": str(tmpdir), "debug_mode": False,
        "learning": {"batch-size": batch, "control-count": 3,
                     "learning-rate": 5e-4, "momentum": 0.5,
                     "weight-decay": 0.01},
    })


def _worker(rank, world, pg_port, ctl_port, tmpdir, clients, cuts,
            fedavg="control"):
    import torch.distributed as dist
    from split_learning_amd.parallel.launch import run_p2p_client
    dist.init_process_group("gloo", init_method=f"tcp://127.0.0.1:{pg_port}",
                            rank=rank, world_size=world)
    cfg = _cfg(tmpdir, clients, cuts, fedavg=fedavg)
    run_p2p_client(cfg, rank, world, torch.device("cpu"), "127.0.0.1", ctl_port,
                   checkpoint_dir=str(tmpdir))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_p2p_two_rank_round(tmp_path):
    world = 2
    pg_port, ctl_port = _free_port(), _free_port()
    mp.spawn(_worker, args=(world, pg_port, ctl_port, str(tmp_path), [1, 1], [7]),
             nprocs=world, join=True)
    ckpt = os.path.join(str(tmp_path), "VGG16_CIFAR10.pth")
    assert os.path.exists(ckpt)
    sd = torch.load(ckpt, weights_only=True)
    from split_learning_amd.models import get_model_class
    assert set(sd.keys()) == set(get_model_class("VGG16", "CIFAR10")().state_dict().keys())


@pytest.mark.timeout(900)
def test_p2p_three_stage_round(tmp_path):
    """3 ranks, cuts [7,14]: middle-stage relay over p2p."""
    world = 3
    pg_port, ctl_port = _free_port(), _free_port()
    mp.spawn(_worker, args=(world, pg_port, ctl_port, str(tmp_path), [1, 1, 1],
                            [7, 14]),
             nprocs=world, join=True)
    assert os.path.exists(os.path.join(str(tmp_path), "VGG16_CIFAR10.pth"))


@pytest.mark.timeout(900)
def test_p2p_rccl_allreduce_fedavg(tmp_path):
    """2+1 clients with the all-reduce FedAvg path: the stage-1 pair averages
    its parameters over a collective group; only the representative ships the
    dict; the saved checkpoint has the full key set."""
    world = 3
    pg_port, ctl_port = _free_port(), _free_port()
    mp.spawn(_worker, args=(world, pg_port, ctl_port, str(tmp_path), [2, 1], [7],
                            "rccl"),
             nprocs=world, join=True)
    ckpt = os.path.join(str(tmp_path), "VGG16_CIFAR10.pth")
    assert os.path.exists(ckpt)
    sd = torch.load(ckpt, weights_only=True)
    from split_learning_amd.models import get_model_class
    assert set(sd.keys()) == set(get_model_class("VGG16", "CIFAR10")().state_dict().keys())


def _noniid_worker(rank, w, pgp, ctlp, tmpdir):
    import torch.distributed as dist
    from split_learning_amd.parallel.launch import run_p2p_client
    dist.init_process_group("gloo", init_method=f"tcp://127.0.0.1:{pgp}",
                            rank=rank, world_size=w)
    cfg = _cfg(tmpdir, [2, 2], [7], num_sample=32, batch=8, fedavg="rccl")
    cfg["server"]["data-distribution"]["non-iid"] = True
    cfg["server"]["data-distribution"]["dirichlet"]["alpha"] = 0.5
    run_p2p_client(cfg, rank, w, torch.device("cpu"), "127.0.0.1", ctlp,
                   checkpoint_dir=tmpdir)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(900)
def test_p2p_noniid_allreduce_2plus2(tmp_path):
    """BASELINE config-3 shape scaled to CI: 2+2 clients, Dirichlet non-IID,
    all-reduce FedAvg, 4 gloo ranks."""
    world = 4
    pg_port, ctl_port = _free_port(), _free_port()
    mp.spawn(_noniid_worker, args=(world, pg_port, ctl_port, str(tmp_path)),
             nprocs=world, join=True)
    assert os.path.exists(os.path.join(str(tmp_path), "VGG16_CIFAR10.pth"))


def _bcast_worker(rank, world, pg_port, ctl_port, tmpdir):
    import torch.distributed as dist
    from split_learning_amd.parallel.launch import run_p2p_client
    dist.init_process_group("gloo", init_method=f"tcp://127.0.0.1:{pg_port}",
                            rank=rank, world_size=world)
    cfg = _cfg(tmpdir, [1, 1], [7])
    cfg["server"]["global-round"] = 2
    cfg["transport"]["params"] = "rccl"   # force the broadcast path (round 2+)
    run_p2p_client(cfg, rank, world, torch.device("cpu"), "127.0.0.1", ctl_port,
                   checkpoint_dir=str(tmpdir))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(900)
def test_p2p_param_broadcast_rounds(tmp_path):
    """Two rounds with transport.params=rccl: round 2's START parameters go
    out as ONE collective broadcast of the full model (bcast.py) instead of
    per-client control blobs; the round completes and the checkpoint holds
    the full key set with finite values."""
    world = 2
    pg_port, ctl_port = _free_port(), _free_port()
    mp.spawn(_bcast_worker, args=(world, pg_port, ctl_port, str(tmp_path)),
             nprocs=world, join=True)
    ckpt = os.path.join(str(tmp_path), "VGG16_CIFAR10.pth")
    assert os.path.exists(ckpt)
    sd = torch.load(ckpt, weights_only=True)
    from split_learning_amd.models import get_model_class
    assert set(sd.keys()) == set(
        get_model_class("VGG16", "CIFAR10")().state_dict().keys())
    assert all(torch.isfinite(v.float()).all() for v in sd.values())
