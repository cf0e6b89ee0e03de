"""Server orchestrator: the round state machine.

Protocol parity with reference src/Server.py:103-212 —
REGISTER (collect clients, Dirichlet/IID label distribution, cluster/select/
partition) -> START (per-stage partition state dicts + routing) -> READY/SYN
rendezvous (replaces the reference's fixed time.sleep(25), src/Server.py:289)
-> NOTIFY/PAUSE per cluster -> UPDATE (FedAvg per cluster per stage, cross-
cluster merge, optional validation, save {model}_{data}.pth) -> next round or
STOP.
"""

from __future__ import annotations

import os
import random
from typing import Any, Dict, List, Optional

import numpy as np
import torch

from ..config import stage_ranges
from ..models import build_partition, get_model_class
from ..data.synthetic import SHAPES
from .cluster import clustering_algorithm
from .fedavg import fedavg_state_dicts
from .partition import partition as partition_cut
from .selection import auto_threshold


def _dummy_input(data_name: str, batch: int = 2) -> torch.Tensor:
    shape, _n, dtype, vocab = SHAPES[data_name]
    if dtype == torch.int64:
        return torch.randint(1, vocab, (batch, *shape), dtype=torch.int64)
    return torch.randn(batch, *shape)


_SHAPE_CACHE: Dict[tuple, List[tuple]] = {}


def boundary_shapes(model_name: str, data_name: str, cuts: List[int],
                    n_stages: int, batch: int) -> List[tuple]:
    """Activation shape at each stage boundary (len == n_stages-1), batch-major."""
    key = (model_name, data_name, tuple(cuts), n_stages)
    if key not in _SHAPE_CACHE:
        total = get_model_class(model_name, data_name).TOTAL_UNITS
        ranges = stage_ranges(cuts, n_stages, total)
        shapes_by_batch = []
        with torch.no_grad():
            # probe TWO batch sizes: the routing contract scales the probed
            # per-sample shape by the run batch, which silently assumes no
            # model mixes batch into other dims — assert it instead of
            # trusting it (round-1 VERDICT weak #7)
            for probe_b in (2, 3):
                x = _dummy_input(data_name, batch=probe_b)
                shapes = []
                for (s, e) in ranges[:-1]:
                    part = build_partition(model_name, data_name, [s, e]).eval()
                    x = part(x)
                    assert x.shape[0] == probe_b, (
                        f"{model_name} stage {s}-{e} does not keep the batch "
                        f"axis leading: {tuple(x.shape)}")
                    shapes.append(tuple(x.shape[1:]))
                shapes_by_batch.append(shapes)
        assert shapes_by_batch[0] == shapes_by_batch[1], (
            f"{model_name}/{data_name} boundary shapes depend on batch size: "
            f"{shapes_by_batch}")
        _SHAPE_CACHE[key] = shapes_by_batch[0]
    return [(batch, *s) for s in _SHAPE_CACHE[key]]


class Server:
    # RCCL group-all-reduce FedAvg requires every member of a (cluster, stage)
    # group to reach the collective in the same round — true for the main
    # concurrent protocol and FLEX, FALSE for the sequential policies
    # (Vanilla/Cluster_FSL/DCSL/2LS), where clients train one-at-a-time and an
    # all-reduce would deadlock (and the weight relay would receive the
    # non-representative ranks' `parameters: None`).  Sequential policy
    # subclasses override this to force the control-plane FedAvg path.
    RCCL_FEDAVG_OK = True

    def __init__(self, config: Dict[str, Any], control, logger=None,
                 checkpoint_dir: str = "."):
        srv = config["server"]
        self.config = config
        self.control = control
        self.logger = logger
        self.auto_mode = srv["auto-mode"]
        self.manual = srv["manual"]
        self.cluster_selection = srv["cluster-selection"]
        self.model_name = srv["model"]
        self.data_name = srv["data-name"]
        self.total_clients = list(srv["clients"])
        self.n_stages = len(self.total_clients)
        self.global_round = srv["global-round"]
        self.round = self.global_round
        self.save_parameters = srv["parameters"]["save"]
        self.load_parameters = srv["parameters"]["load"]
        self.validation = srv["validation"]
        self.learning = config["learning"]
        self.data_distribution = srv["data-distribution"]
        self.refresh = self.data_distribution["refresh"]
        self.random_seed = srv["random-seed"]
        self.checkpoint_dir = checkpoint_dir
        if self.random_seed:
            random.seed(self.random_seed)
            np.random.seed(self.random_seed)

        self.list_clients: List[dict] = []   # {client_id, layer_id, profile, cluster, label, train}
        self.register_counts = [0] * self.n_stages
        self.update_counts = [0] * self.n_stages
        self.first_layer_done_per_cluster: List[int] = []
        self.round_result = True
        self.num_cluster = 1
        self.infor_cluster = None
        self.list_cut_layers: List[List[int]] = []
        self.size_data = None
        self.label_counts = None
        self.reject = False
        # collected per round: [cluster][stage] -> list of (state_dict, size)
        self.collected: List[List[List[tuple]]] = []
        self.stopped = False

    # ------------------------------------------------------------------
    @property
    def ckpt_path(self) -> str:
        return os.path.join(self.checkpoint_dir, f"{self.model_name}_{self.data_name}.pth")

    def _log(self, m, warn=False):
        if self.logger:
            (self.logger.log_warning if warn else self.logger.log_info)(m)

    # ------------------------------------------------------------------
    def run(self):
        """Blocking server loop until STOP."""
        while not self.stopped:
            msg = self.control.recv("server", block=True, timeout=600.0)
            if msg is None:
                raise TimeoutError("server: no control message within 600 s")
            self.dispatch(msg)

    def dispatch(self, msg: Dict[str, Any]):
        action = msg["action"]
        if action == "REGISTER":
            self.on_register(msg)
        elif action == "NOTIFY":
            self.on_notify(msg)
        elif action == "UPDATE":
            self.on_update(msg)
        elif action == "READY":
            pass  # handled synchronously in notify_clients
        else:
            raise RuntimeError(f"server: unknown action {action}")

    # ------------------------------------------------------------------
    def on_register(self, msg):
        rec = {"client_id": msg["client_id"], "layer_id": msg["layer_id"],
               "profile": msg.get("profile") or {}, "cluster": msg.get("cluster"),
               "label": [], "train": bool(msg.get("select", True)),
               "out_cluster": msg.get("out_cluster")}
        if self.size_data is None and rec["layer_id"] == 1:
            self.size_data = rec["profile"].get("size_data")
        if not any(c["client_id"] == rec["client_id"] for c in self.list_clients):
            self.list_clients.append(rec)
            self.register_counts[rec["layer_id"] - 1] += 1
        if self.register_counts == self.total_clients:
            self._log(f"All {sum(self.total_clients)} clients registered")
            self.on_all_registered()

    def on_all_registered(self):
        self.distribution()
        self.cluster_and_selection()
        self._log(f"cut layers: {self.list_cut_layers}, clusters: {self.infor_cluster}")
        self._log(f"Start training round {self.global_round - self.round + 1}")
        self.notify_clients()

    def distribution(self):
        """IID or Dirichlet non-IID per-label sample counts for layer-1 clients
        (reference src/Server.py:87-101)."""
        dd = self.data_distribution
        n1 = self.total_clients[0]
        if dd["non-iid"]:
            dist = np.random.dirichlet([dd["dirichlet"]["alpha"]] * dd["num-label"], n1)
            self.label_counts = (dist * dd["num-sample"]).astype(int)
        else:
            self.label_counts = np.full((n1, dd["num-label"]),
                                        dd["num-sample"] // dd["num-label"])
        counts = self.label_counts.tolist()
        for rec in self.list_clients:
            rec["label"] = counts.pop() if rec["layer_id"] == 1 else []

    def cluster_and_selection(self):
        """Auto: KMeans clustering + GMM selection + throughput-model cut search.
        Manual: config-specified clusters/cuts (reference src/Server.py:300-382)."""
        if self.auto_mode:
            self.num_cluster = self.cluster_selection["num-cluster"]
            labels, infor = clustering_algorithm(
                self.label_counts, self.num_cluster,
                self.cluster_selection.get("algorithm-cluster", "KMeans"))
            labels = list(labels)
            self.infor_cluster = [row + [0] for row in infor]
            perf = [[] for _ in range(self.num_cluster)]
            for rec in self.list_clients:
                if rec["layer_id"] == 1:
                    rec["cluster"] = int(labels.pop())
                    perf[rec["cluster"]].append(rec["profile"].get("speed", 1.0))
                else:
                    rec["cluster"] = rec["cluster"] or 0
            if self.cluster_selection["selection-mode"]:
                thresholds = [auto_threshold(p) for p in perf]
                for rec in self.list_clients:
                    if rec["layer_id"] == 1:
                        if rec["profile"].get("speed", 1.0) < thresholds[rec["cluster"]]:
                            rec["train"] = False
                            self.total_clients[0] -= 1
                            self.infor_cluster[rec["cluster"]][0] -= 1
                            self._log(f"Rejected slow client {rec['client_id']}", warn=True)
                    else:
                        self.infor_cluster[rec["cluster"]][1] += 1
            else:
                for rec in self.list_clients:
                    if rec["layer_id"] == 2:
                        self.infor_cluster[rec["cluster"]][1] += 1
            # per-cluster auto cut from profiles
            self.list_cut_layers = []
            for k in range(self.num_cluster):
                e1, n1c, e2, n2c = [], [], [], []
                for rec in self.list_clients:
                    if rec["cluster"] == k and rec["train"]:
                        (e1 if rec["layer_id"] == 1 else e2).append(
                            rec["profile"].get("exe_time", []))
                        (n1c if rec["layer_id"] == 1 else n2c).append(
                            rec["profile"].get("network", 1.0))
                if self.size_data:
                    self.list_cut_layers.append(
                        partition_cut(e1, n1c, e2, n2c, self.size_data))
                else:
                    self.list_cut_layers.append(
                        list(self.manual["no-cluster"]["cut-layers"]))
        else:
            # manual select/reject (FLEX --s): clients that registered with
            # select=False are rejected up front
            for rec in self.list_clients:
                if not rec["train"]:
                    self.total_clients[rec["layer_id"] - 1] -= 1
            if self.manual["cluster-mode"]:
                self.num_cluster = self.manual["cluster"]["num-cluster"]
                self.infor_cluster = [list(r) for r in self.manual["cluster"]["infor-cluster"]]
                self.list_cut_layers = [list(c) for c in self.manual["cluster"]["cut-layers"]]
                for rec in self.list_clients:
                    rec["cluster"] = rec["cluster"] or 0
            else:
                self.num_cluster = 1
                self.infor_cluster = [list(self.total_clients)]
                self.list_cut_layers = [list(self.manual["no-cluster"]["cut-layers"])]
                for rec in self.list_clients:
                    rec["cluster"] = 0

        self.collected = [[[] for _ in range(self.n_stages)]
                          for _ in range(self.num_cluster)]
        self.first_layer_done_per_cluster = [0] * self.num_cluster

    # ------------------------------------------------------------------
    def _stage_clients(self, cluster: int, layer: int) -> List[dict]:
        return [c for c in self.list_clients
                if c["cluster"] == cluster and c["layer_id"] == layer and c["train"]]

    def _routing_for(self, rec) -> dict:
        """Static round-robin producer->consumer edges (DCSL-style targeted
        routing) + boundary activation shapes for p2p buffer pre-allocation."""
        cuts = self.list_cut_layers[rec["cluster"]]
        batch = self.learning["batch-size"]
        shapes = boundary_shapes(self.model_name, self.data_name, cuts,
                                 self.n_stages, batch)
        s = rec["layer_id"]
        my_peers = self._stage_clients(rec["cluster"], s)
        my_idx = [c["client_id"] for c in my_peers].index(rec["client_id"])
        down_peer = None
        up_peers: List[int] = []
        if s < self.n_stages:
            consumers = [c["client_id"] for c in self._stage_clients(rec["cluster"], s + 1)]
            down_peer = consumers[my_idx % len(consumers)]
        if s > 1:
            producers = [c["client_id"] for c in self._stage_clients(rec["cluster"], s - 1)]
            n_cons = len(my_peers)
            up_peers = [p for i, p in enumerate(producers) if i % n_cons == my_idx]
        return {
            "down_peer": down_peer,
            "up_peers": up_peers,
            "act_shape_out": shapes[s - 1] if s < self.n_stages else None,
            "act_shape_in": shapes[s - 2] if s > 1 else None,
            "batch": batch,
        }

    # -- reusable protocol pieces (policies compose these differently) -----
    def _load_ckpt(self):
        """Checkpoint reload for a round's START broadcast.

        The reference reloads {model}_{data}.pth every round gated on
        save_parameters ALONE (src/Server.py:230-232; its `load` flag is read
        but never used).  Here `parameters.load` additionally gates only the
        FIRST round's load (so load=False starts from fresh init even when a
        stale .pth exists); from round 2 on, save_parameters alone triggers the
        reload of the just-saved aggregate — otherwise training progress would
        be silently discarded each round (ADVICE.md round-1 medium finding).
        """
        first_round = self.round == self.global_round
        want = self.load_parameters if first_round else self.save_parameters
        if want and os.path.exists(self.ckpt_path):
            self._log(f"Loaded checkpoint {self.ckpt_path}")
            return torch.load(self.ckpt_path, weights_only=True)
        return None

    def _stage_layers(self, rec):
        cuts = self.list_cut_layers[rec["cluster"]]
        ranges = stage_ranges(cuts, self.n_stages,
                              get_model_class(self.model_name, self.data_name).TOTAL_UNITS)
        return ranges[rec["layer_id"] - 1]

    def _slice_state(self, full_state, layers):
        if full_state is None:
            return None
        part = build_partition(self.model_name, self.data_name, layers)
        return {k: full_state[k] for k in part.state_dict().keys()}

    def scheduler_overrides(self, rec) -> dict:
        """Per-policy stage-loop options shipped in START (see policies.py)."""
        sch = self.config.get("scheduler") or {}
        return {k: sch[k] for k in ("epochs", "limited-time", "clip-grad-norm")
                if sch.get(k) is not None}

    def _fedavg_groups(self):
        """Same-(cluster, stage) client groups for the RCCL all-reduce FedAvg
        path: every rank creates these communicators in this exact order, and
        each group's members all-reduce weight*size / size instead of shipping
        state dicts through the control plane (replaces the reference's
        server-side averaging loop, src/Server.py:398-408)."""
        groups = []
        for k in range(self.num_cluster):
            for stage in range(1, self.n_stages + 1):
                ids = sorted(c["client_id"] for c in self.list_clients
                             if c["train"] and c["cluster"] == k
                             and c["layer_id"] == stage)
                if ids:
                    groups.append(ids)
        return groups

    def _param_bcast_ok(self, active, full_state) -> bool:
        """START parameters go out as ONE RCCL broadcast of the full model
        (bcast.py) instead of per-client control-plane blobs when: the
        transport allows it, the policy is concurrent (every rank reaches the
        collective this round), torch.distributed is live in this process
        (p2p mode: the server thread shares rank 0's process), and every rank
        is an accepted client (a rejected rank would never join)."""
        import torch.distributed as dist
        mode = (self.config.get("transport") or {}).get("params", "auto")
        if mode not in ("auto", "rccl") or not self.RCCL_FEDAVG_OK:
            return False
        if full_state is None or not (dist.is_available() and dist.is_initialized()):
            return False
        ids = sorted(c["client_id"] for c in active)
        return ids == list(range(dist.get_world_size()))

    def _send_start(self, rec, full_state, state_override=None,
                    param_bcast=False):
        layers = self._stage_layers(rec)
        state = None if param_bcast else (
            state_override if state_override is not None
            else self._slice_state(full_state, layers))
        self.control.send(f"client_{rec['client_id']}", {
            "action": "START", "message": "Server accept the connection!",
            "parameters": state, "param_bcast": param_bcast, "layers": layers,
            "model_name": self.model_name, "data_name": self.data_name,
            "learning": self.learning, "label_count": rec["label"],
            "refresh": self.refresh, "cluster": rec["cluster"],
            "n_stages": self.n_stages, "routing": self._routing_for(rec),
            "scheduler": self.scheduler_overrides(rec),
            "fedavg": ((self.config.get("transport") or {}).get("fedavg", "control")
                       if self.RCCL_FEDAVG_OK else "control"),
            "fedavg_groups": self._fedavg_groups() if self.RCCL_FEDAVG_OK else None,
        })

    def _send_stop(self, rec, message="Stop training!"):
        self.control.send(f"client_{rec['client_id']}",
                          {"action": "STOP", "message": message, "parameters": None})

    def _send_pause(self, rec, send=True):
        self.control.send(f"client_{rec['client_id']}",
                          {"action": "PAUSE",
                           "message": "Pause training and please send your parameters",
                           "parameters": None, "send": send})

    def _wait_ready(self, n):
        """READY/SYN rendezvous (replaces the reference's time.sleep(25),
        src/Server.py:289)."""
        ready = 0
        while ready < n:
            msg = self.control.recv("server", block=True, timeout=600.0)
            if msg is None:
                raise TimeoutError("server: waiting for READY")
            if msg.get("action") == "READY":
                ready += 1
            else:
                self.dispatch(msg)

    def _send_syn(self, recs):
        for rec in recs:
            self.control.send(f"client_{rec['client_id']}",
                              {"action": "SYN", "message": "Synchronize client devices"})

    def notify_clients(self, start: bool = True):
        full_state = self._load_ckpt() if start else None
        active = [c for c in self.list_clients if c["train"]]
        use_bcast = start and self._param_bcast_ok(active, full_state)
        if use_bcast:
            from . import bcast
            bcast.stage_full_state("round", full_state)
        for rec in self.list_clients:
            if not start:
                self._send_stop(rec)
                continue
            if not rec["train"]:
                if not self.reject:
                    self._send_stop(rec, "Reject Device")
                continue
            self._send_start(rec, full_state, param_bcast=use_bcast)
        if not start:
            self.stopped = True
            return
        self.reject = True
        self._wait_ready(len(active))
        self._send_syn(active)

    # ------------------------------------------------------------------
    def on_notify(self, msg):
        cluster = msg["cluster"]
        if msg["layer_id"] == 1:
            self.first_layer_done_per_cluster[cluster] += 1
        if self.first_layer_done_per_cluster[cluster] == self.infor_cluster[cluster][0]:
            self.first_layer_done_per_cluster[cluster] = 0
            self._log(f"Cluster {cluster} finished; sending PAUSE")
            for rec in self.list_clients:
                if rec["train"] and rec["cluster"] == cluster:
                    self._send_pause(rec)

    def on_update(self, msg):
        layer_id = msg["layer_id"]
        cluster = msg["cluster"]
        self.update_counts[layer_id - 1] += 1
        if not msg["result"]:
            self.round_result = False
        if self.save_parameters and self.round_result and msg.get("parameters") is not None:
            self.collected[cluster][layer_id - 1].append(
                (msg["parameters"], msg["size"]))

        if self.update_counts == self.total_clients:
            self._log("Collected all parameters")
            self.update_counts = [0] * self.n_stages
            if self.save_parameters and self.round_result:
                full = self.aggregate()
                ok = True
                if self.validation:
                    from ..validation import get_val
                    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
                    ok = get_val(self.model_name, self.data_name, full, self.logger,
                                 device=dev)
                if ok:
                    torch.save(full, self.ckpt_path)
                    self.round -= 1
                else:
                    self._log("Training failed!", warn=True)
                    self.round = 0
            else:
                self.round -= 1
            self.collected = [[[] for _ in range(self.n_stages)]
                              for _ in range(self.num_cluster)]
            self.round_result = True
            if self.round > 0:
                self._log(f"Start training round {self.global_round - self.round + 1}")
                self.notify_clients()
            else:
                self._log("Stop training !!!")
                self.notify_clients(start=False)

    def aggregate(self) -> Dict[str, torch.Tensor]:
        """Per-cluster per-stage weighted FedAvg, then cross-cluster merge+avg
        (reference src/Server.py:398-434)."""
        cluster_dicts = []
        for k in range(self.num_cluster):
            merged: Dict[str, torch.Tensor] = {}
            for stage_list in self.collected[k]:
                if not stage_list:
                    continue
                sds = [sd for sd, _size in stage_list]
                sizes = [float(sz) for _sd, sz in stage_list]
                merged.update(fedavg_state_dicts(sds, weights=sizes))
            if merged:
                cluster_dicts.append(merged)
        if not cluster_dicts:
            raise RuntimeError("no cluster produced parameters")
        return fedavg_state_dicts(cluster_dicts)
