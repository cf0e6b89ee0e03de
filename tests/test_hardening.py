"""Round-2 hardening tests: checkpoint reload gating, FedAvg NaN/inf
semantics, per-client subset seeding, sequential-policy x RCCL-FedAvg
interaction, and stage-loop stall diagnostics."""

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


def _base_cfg(tmpdir, clients=(1, 1), cuts=(7,), fedavg="control",
              load=True, save=True, policy=None, rounds=1):
    from split_learning_amd.config import load_config
    cfg = load_config(None, overrides={
        "transport": {"fedavg": fedavg},
        "server": {
            "global-round": rounds, "clients": list(clients), "model": "VGG16",
            "data-name": "CIFAR10", "validation": False,
            "parameters": {"load": load, "save": save},
            "data-distribution": {"num-sample": 32, "num-label": 10,
                                  "non-iid": False, "dirichlet": {"alpha": 1},
                                  "refresh": True},
            "manual": {"cluster-mode": False,
                       "no-cluster": {"cut-layers": list(cuts)}},
        },
        "log_path": str(tmpdir), "debug_mode": False,
        "learning": {"batch-size": 8, "control-count": 3,
                     "learning-rate": 5e-4, "momentum": 0.5,
                     "weight-decay": 0.01},
    })
    if policy:
        cfg["scheduler"] = {"policy": policy, "recompute": True}
    return cfg


# ---------------------------------------------------------------------------
# checkpoint reload gating

def test_ckpt_reload_gating(tmp_path):
    """load=False blocks only the FIRST round's reload; later rounds reload on
    save=True alone (reference src/Server.py:230-232 reloads on save alone)."""
    from split_learning_amd.parallel.control import InProcControl
    from split_learning_amd.parallel.server import Server

    cfg = _base_cfg(tmp_path, load=False, save=True, rounds=3)
    srv = Server(cfg, InProcControl(), checkpoint_dir=str(tmp_path))
    sd = {"layer1.weight": torch.ones(2, 2)}
    torch.save(sd, srv.ckpt_path)

    # round 1 (round == global_round): load=False -> fresh init
    assert srv._load_ckpt() is None
    # later round: save=True alone triggers the reload
    srv.round = 2
    loaded = srv._load_ckpt()
    assert loaded is not None and torch.equal(loaded["layer1.weight"], sd["layer1.weight"])

    # load=True: first round also reloads
    cfg2 = _base_cfg(tmp_path, load=True, save=True, rounds=3)
    srv2 = Server(cfg2, InProcControl(), checkpoint_dir=str(tmp_path))
    assert srv2._load_ckpt() is not None

    # save=False, load=False: never reloads
    cfg3 = _base_cfg(tmp_path, load=False, save=False, rounds=3)
    srv3 = Server(cfg3, InProcControl(), checkpoint_dir=str(tmp_path))
    srv3.round = 1
    assert srv3._load_ckpt() is None


def test_save_only_round_progress(tmp_path):
    """{save: True, load: False}: training progress must carry across rounds —
    round 2's START parameters are the round-1 aggregate, not None (ADVICE.md:
    with the old gating every round restarted from random init)."""
    from split_learning_amd.parallel.launch import run_loopback

    cfg = _base_cfg(tmp_path, load=False, save=True, rounds=2)
    seen_params = []

    from split_learning_amd.parallel import server as server_mod
    orig = server_mod.Server._send_start

    def spy(self, rec, full_state, state_override=None, **kw):
        seen_params.append(full_state is not None or state_override is not None)
        return orig(self, rec, full_state, state_override, **kw)

    server_mod.Server._send_start = spy
    try:
        run_loopback(cfg, device="cpu", max_batches=2,
                     checkpoint_dir=str(tmp_path))
    finally:
        server_mod.Server._send_start = orig
    # 2 clients x 2 rounds = 4 STARTs: first round fresh (False), second loaded
    assert seen_params[:2] == [False, False]
    assert seen_params[2:] == [True, True]


# ---------------------------------------------------------------------------
# FedAvg inf/NaN semantics

def test_fedavg_nan_zeroed_inf_preserved():
    from split_learning_amd.parallel.fedavg import fedavg_state_dicts
    a = {"w": torch.tensor([1.0, float("nan"), float("inf"), -float("inf")])}
    b = {"w": torch.tensor([3.0, 2.0, 1.0, 1.0])}
    out = fedavg_state_dicts([a, b])
    assert out["w"][0] == 2.0
    assert out["w"][1] == 1.0          # NaN -> 0, then mean with 2.0
    assert torch.isinf(out["w"][2]) and out["w"][2] > 0   # +inf propagates
    assert torch.isinf(out["w"][3]) and out["w"][3] < 0   # -inf propagates


# ---------------------------------------------------------------------------
# per-client subset seeding

def test_subset_seed_diversity():
    from split_learning_amd.data.real import _subset_by_distribution
    x = torch.arange(1000).unsqueeze(1).float()
    y = torch.arange(1000) % 10
    dist_counts = [5] * 10
    xa, _ = _subset_by_distribution(x, y, dist_counts, seed=0)
    xb, _ = _subset_by_distribution(x, y, dist_counts, seed=1)
    xa2, _ = _subset_by_distribution(x, y, dist_counts, seed=0)
    assert torch.equal(xa, xa2)                       # deterministic per seed
    assert not torch.equal(xa, xb)                    # different across clients
    assert set(xa.flatten().tolist()) != set(xb.flatten().tolist())


# ---------------------------------------------------------------------------
# sequential policies force control-plane FedAvg (VERDICT weak #4)

def test_sequential_policies_force_control_fedavg(tmp_path):
    from split_learning_amd.parallel.control import InProcControl
    from split_learning_amd.parallel.policies import (ClusterFSLServer,
                                                      DCSLServer, FlexServer,
                                                      TwoLSServer,
                                                      VanillaServer)
    from split_learning_amd.parallel.server import Server

    cfg = _base_cfg(tmp_path, fedavg="rccl")
    assert Server(cfg, InProcControl()).RCCL_FEDAVG_OK is True
    assert FlexServer(cfg, InProcControl()).RCCL_FEDAVG_OK is True
    for klass in (VanillaServer, ClusterFSLServer, DCSLServer, TwoLSServer):
        assert klass(cfg, InProcControl()).RCCL_FEDAVG_OK is False, klass


def _vanilla_rccl_worker(rank, world, pg_port, ctl_port, tmpdir):
    import torch.distributed as dist
    from split_learning_amd.parallel.launch import run_p2p_client
    dist.init_process_group("gloo", init_method=f"tcp://127.0.0.1:{pg_port}",
                            rank=rank, world_size=world)
    cfg = _base_cfg(tmpdir, clients=(2, 1), cuts=(7,), fedavg="rccl",
                    policy="vanilla")
    run_p2p_client(cfg, rank, world, torch.device("cpu"), "127.0.0.1", ctl_port,
                   checkpoint_dir=str(tmpdir))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(900)
def test_vanilla_with_rccl_fedavg_config(tmp_path):
    """config asks for transport.fedavg=rccl but the vanilla policy is
    sequential: the server must force the control path (else the group
    all-reduce deadlocks and the relay receives None).  Round completes and
    the checkpoint holds real (finite) parameters."""
    world = 3
    pg_port, ctl_port = _free_port(), _free_port()
    mp.spawn(_vanilla_rccl_worker,
             args=(world, pg_port, ctl_port, str(tmp_path)),
             nprocs=world, join=True)
    ckpt = os.path.join(str(tmp_path), "VGG16_CIFAR10.pth")
    assert os.path.exists(ckpt)
    sd = torch.load(ckpt, weights_only=True)
    from split_learning_amd.models import get_model_class
    assert set(sd.keys()) == set(
        get_model_class("VGG16", "CIFAR10")().state_dict().keys())
    assert all(torch.isfinite(v.float()).all() for v in sd.values())


# ---------------------------------------------------------------------------
# stage-loop stall diagnostics (VERDICT weak #6 / next-round #9)

def _middle_ctx(plane, control, stall=0.6):
    from split_learning_amd.models import build_partition
    from split_learning_amd.parallel.schedulers import StageContext
    model = build_partition("VGG16", "CIFAR10", [7, 14])
    opt = torch.optim.SGD(model.parameters(), lr=1e-3)
    return StageContext(
        client_id=1, layer_id=2, n_stages=3, cluster=0, model=model,
        optimizer=opt, learning={"control-count": 3}, plane=plane,
        control=control, device=torch.device("cpu"), stall_timeout_s=stall)


@pytest.mark.timeout(120)
def test_middle_stage_stall_raises_named_error(tmp_path):
    """A gradient lost upstream must surface StageStallError naming the
    missing microbatch within stall_timeout_s, not spin forever."""
    from split_learning_amd.parallel.control import InProcControl
    from split_learning_amd.parallel.data_plane import LoopbackData
    from split_learning_amd.parallel.messages import ActivationMsg
    from split_learning_amd.parallel.schedulers import (StageStallError,
                                                        train_middle_stage)

    plane = LoopbackData()
    control = InProcControl()
    ctx = _middle_ctx(plane, control)
    # one activation arrives, its gradient never does
    plane.send_activation(1, 0, ActivationMsg(
        42, torch.randn(4, 64, 16, 16), torch.zeros(4, dtype=torch.int64), [0]))
    with pytest.raises(StageStallError) as ei:
        train_middle_stage(ctx)
    assert 42 in ei.value.missing_ids
    assert "middle" in str(ei.value)


@pytest.mark.timeout(120)
def test_middle_stage_pause_polled_with_inflight(tmp_path):
    """PAUSE arriving while a microbatch is in flight: the loop must notice it
    (poll-with-inflight) and, when the gradient never comes, still raise the
    stall error rather than hang."""
    from split_learning_amd.parallel.control import InProcControl
    from split_learning_amd.parallel.data_plane import LoopbackData
    from split_learning_amd.parallel.messages import ActivationMsg
    from split_learning_amd.parallel.schedulers import (StageStallError,
                                                        train_middle_stage)

    plane = LoopbackData()
    control = InProcControl()
    control.send("client_1", {"action": "PAUSE"})
    plane.send_activation(1, 0, ActivationMsg(
        7, torch.randn(4, 64, 16, 16), torch.zeros(4, dtype=torch.int64), [0]))
    ctx = _middle_ctx(plane, control)
    with pytest.raises(StageStallError):
        train_middle_stage(ctx)


@pytest.mark.timeout(120)
def test_first_stage_stall_raises(tmp_path):
    """Stage-1 waiting forever on a lost gradient raises within the bound."""
    from torch.utils.data import DataLoader, TensorDataset
    from split_learning_amd.models import build_partition
    from split_learning_amd.parallel.control import InProcControl
    from split_learning_amd.parallel.data_plane import LoopbackData
    from split_learning_amd.parallel.schedulers import (StageContext,
                                                        StageStallError,
                                                        train_first_stage)

    model = build_partition("VGG16", "CIFAR10", [0, 7])
    opt = torch.optim.SGD(model.parameters(), lr=1e-3)
    loader = DataLoader(TensorDataset(torch.randn(8, 3, 32, 32),
                                      torch.zeros(8, dtype=torch.int64)),
                        batch_size=4)
    ctx = StageContext(
        client_id=0, layer_id=1, n_stages=2, cluster=0, model=model,
        optimizer=opt, learning={"control-count": 2}, plane=LoopbackData(),
        control=InProcControl(), device=torch.device("cpu"),
        train_loader=loader, stall_timeout_s=0.6)
    with pytest.raises(StageStallError) as ei:
        train_first_stage(ctx)
    assert "first" in str(ei.value)


def _bcast_unit_worker(rank, world, port, tmpdir):
    import torch.distributed as dist
    from split_learning_amd.models import build_partition
    from split_learning_amd.parallel import bcast
    dist.init_process_group("gloo", init_method=f"tcp://127.0.0.1:{port}",
                            rank=rank, world_size=world)
    full = build_partition("VGG16", "CIFAR10", [0, 0]).state_dict() \
        if rank == 0 else None
    out = bcast.broadcast_full_state("VGG16", "CIFAR10", full,
                                     torch.device("cpu"))
    ref = build_partition("VGG16", "CIFAR10", [0, 0]).state_dict()
    assert set(out.keys()) == set(ref.keys())
    for k, v in out.items():
        assert v.shape == ref[k].shape and v.dtype == ref[k].dtype, k
    # round-trip exactness on rank 0 (fp32 path is bit-exact; int buffers ride
    # as fp32, exact below 2^24)
    if rank == 0:
        for k in full:
            assert torch.equal(out[k], full[k]), k
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_param_broadcast_roundtrip(tmp_path):
    """bcast.broadcast_full_state: every rank reconstructs the exact full
    state dict (keys, shapes, dtypes; bit-exact values on the source)."""
    port = _free_port()
    mp.spawn(_bcast_unit_worker, args=(2, port, str(tmp_path)), nprocs=2,
             join=True)


def test_opt_desc_python_matches_cpp():
    """The fused-step pointer table is built in Python (pinned host buffer +
    async copy, legal inside hipGraph capture — parallel/optim._build_desc);
    its layout must stay bit-identical to the C++ builder the kernels were
    written against (ops/csrc/optim.hip make_opt_desc: prefix[n+1], numel[n],
    param/grad/state1/state2 ptrs[n])."""
    import torch

    from split_learning_amd.ops import native
    from split_learning_amd.parallel.optim import _build_desc, _pinned_host

    p = [torch.randn(7), torch.randn(3, 5), torch.randn(20000)]
    g = [torch.randn_like(t) for t in p]
    b = [torch.zeros_like(t) for t in p]
    ref = native().make_opt_desc(p, g, b, b)
    host = _pinned_host(len(p))
    got, chunks = _build_desc(host, p, g, b, b, torch.device("cpu"))
    assert torch.equal(ref, got)
    # chunks = ceil(numel / 16384) summed; the 20000-element tensor needs 2
    assert chunks == 1 + 1 + 2


def test_opt_desc_rebuild_retires_host_buffer():
    """A captured graph's memcpy node re-reads its pinned source on every
    replay, so a descriptor REBUILD (outside capture) must take a fresh host
    buffer and retire the old one rather than overwrite it in place."""
    import torch

    from split_learning_amd.parallel import optim as O

    p = [torch.randn(100, requires_grad=True)]
    o = O.FusedSGD(p, lr=0.1, momentum=0.9)
    p[0].grad = torch.randn(100)
    live = list(zip(o.params, o.bufs))
    h0 = o._host
    o._desc(live)
    p[0].grad = torch.randn(100)  # new storage -> signature miss -> rebuild
    o._desc(live)
    assert o._host is not h0
    assert O._RETIRED_HOSTS and O._RETIRED_HOSTS[-1] is h0
