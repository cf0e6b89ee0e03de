"""Variant scheduling policies as pluggable server orchestrators.

The reference ships five forks under other/ (SURVEY.md §2.3); here each is a
Server subclass that recomposes the same protocol pieces:

* VanillaServer (Vanilla_SL): SEQUENTIAL split learning — stage-1 edge devices
  train one at a time, weights relayed through the server to the next device;
  stage>=2 devices stay resident for the whole round; final FedAvg of collected
  parts (other/Vanilla_SL/src/Server.py:130-183).  Epoch loop + optional
  wall-clock time limit + clip-grad-norm ride the scheduler overrides.
* ClusterFSLServer (Cluster_FSL): clusters run one after another; within a
  cluster edge devices run in parallel then are FedAvg'd, and the averaged
  model seeds the next cluster (other/Cluster_FSL/src/Server.py).
* DCSLServer (DCSL): Cluster_FSL plus synchronous lock-step first layer and
  SDA concat-batching at the last stage (other/DCSL/src/Scheduler.py:115-221).
* FlexServer (FLEX): periodic aggregation — client-side FedAvg every t_c
  rounds, global merge+validate+save every t_g; PAUSE carries a `send` flag so
  clients skip parameter upload on non-aggregation rounds
  (other/FLEX/src/Server.py:135-208); first layer synchronous; manual
  select/reject via the client --s flag (handled in Server REGISTER).
* TwoLSServer (2LS): two-level clusters — out-clusters sequential in shuffled
  order; in-cluster results FedAvg'd then folded into the global model by
  FedAsync mixing (1-a)*global + a*new with a = 1/(1+arrival_rank)
  (other/2LS/src/Server.py:141-233).
"""

from __future__ import annotations

import os
import random
from typing import Dict, List, Optional

import torch

from .fedavg import fedavg_state_dicts
from .server import Server


class VanillaServer(Server):
    """Sequential SL: one stage-1 device at a time, weights relayed onward."""

    RCCL_FEDAVG_OK = False  # sequential: group all-reduce would deadlock

    def __init__(self, *a, **kw):
        super().__init__(*a, **kw)
        self._edges: List[dict] = []
        self._seq_idx = 0
        self._relay_state: Optional[Dict[str, torch.Tensor]] = None
        self._edge_parts: List[tuple] = []   # (state_dict, size) per edge
        self._tail_updates: List[dict] = []
        self._expect_tail = 0

    def scheduler_overrides(self, rec):
        out = super().scheduler_overrides(rec)
        sch = self.config.get("scheduler") or {}
        if sch.get("epochs"):
            out["epochs"] = sch["epochs"]
        return out

    def on_all_registered(self):
        self.distribution()
        self.cluster_and_selection()
        self.start_round()

    def start_round(self):
        self._log(f"[vanilla] round {self.global_round - self.round + 1}")
        self._edges = [c for c in self.list_clients
                       if c["layer_id"] == 1 and c["train"]]
        self._resident = [c for c in self.list_clients
                          if c["layer_id"] > 1 and c["train"]]
        self._seq_idx = 0
        self._edge_parts = []
        self._tail_updates = []
        full = self._load_ckpt()
        self._relay_state = (self._slice_state(full, self._stage_layers(self._edges[0]))
                             if full is not None else None)
        for rec in self._resident:
            self._send_start(rec, full)
        self._start_edge(self._edges[0], wait_extra=len(self._resident))
        self._send_syn(self._resident)

    def _start_edge(self, rec, wait_extra=0):
        self._send_start(rec, None, state_override=self._relay_state)
        self._wait_ready(1 + wait_extra)
        self._send_syn([rec])

    def on_notify(self, msg):
        # an edge device finished its pass
        if self._seq_idx < len(self._edges) - 1:
            self._send_pause(self._edges[self._seq_idx])
        else:
            # last edge: pause it and every resident device
            self._send_pause(self._edges[self._seq_idx])
            for rec in self._resident:
                self._send_pause(rec)
            self._expect_tail = len(self._resident) + 1

    def on_update(self, msg):
        if msg["layer_id"] == 1 and self._seq_idx < len(self._edges) - 1:
            self._edge_parts.append((msg["parameters"], msg["size"]))
            if msg["parameters"] is not None:       # relay to the next device
                self._relay_state = msg["parameters"]  # (None: keep last state)
            self._seq_idx += 1
            self._start_edge(self._edges[self._seq_idx])
            return
        # tail collection: last edge + resident devices
        self._tail_updates.append(msg)
        if msg["layer_id"] == 1:
            self._edge_parts.append((msg["parameters"], msg["size"]))
        if not msg["result"]:
            self.round_result = False
        if len(self._tail_updates) == self._expect_tail:
            self._finish_round()

    def _finish_round(self):
        if self.save_parameters and self.round_result:
            merged: Dict[str, torch.Tensor] = {}
            sds = [sd for sd, _ in self._edge_parts if sd]
            sizes = [float(sz) for sd, sz in self._edge_parts if sd]
            if sds:
                merged.update(fedavg_state_dicts(sds, weights=sizes))
            by_stage: Dict[int, List[tuple]] = {}
            for m in self._tail_updates:
                if m["layer_id"] > 1 and m.get("parameters"):
                    by_stage.setdefault(m["layer_id"], []).append(
                        (m["parameters"], float(m["size"] or 1)))
            for stage_list in by_stage.values():
                merged.update(fedavg_state_dicts([sd for sd, _ in stage_list],
                                                 weights=[w for _, w in stage_list]))
            ok = True
            if self.validation:
                from ..validation import get_val
                ok = get_val(self.model_name, self.data_name, merged, self.logger)
            if ok:
                torch.save(merged, self.ckpt_path)
                self.round -= 1
            else:
                self._log("Training failed!", warn=True)
                self.round = 0
        else:
            self.round -= 1
        self.round_result = True
        if self.round > 0:
            self.start_round()
        else:
            self.notify_clients(start=False)


class ClusterFSLServer(Server):
    """Sequential clusters; averaged model seeds the next cluster."""

    RCCL_FEDAVG_OK = False  # sequential clusters: later clusters' ranks would
    SCHED_EXTRA: dict = {}  # never reach the group collective (see Server)

    def __init__(self, *a, **kw):
        super().__init__(*a, **kw)
        self._order: List[int] = []
        self._ci = 0
        self._carry: Optional[Dict[str, torch.Tensor]] = None
        self._cluster_updates: List[dict] = []
        self._cluster_avgs: List[Dict[str, torch.Tensor]] = []

    def scheduler_overrides(self, rec):
        out = super().scheduler_overrides(rec)
        out.update(self.SCHED_EXTRA)
        return out

    def _cluster_order(self) -> List[int]:
        return list(range(self.num_cluster))

    def on_all_registered(self):
        self.distribution()
        self.cluster_and_selection()
        self.start_round()

    def _members(self, k):
        return [c for c in self.list_clients if c["train"] and c["cluster"] == k]

    def start_round(self):
        self._log(f"[{type(self).__name__}] round {self.global_round - self.round + 1}")
        self._order = self._cluster_order()
        self._ci = 0
        self._cluster_avgs = []
        self._carry = self._load_ckpt()
        self._start_cluster()

    def _start_cluster(self):
        k = self._order[self._ci]
        members = self._members(k)
        self._cluster_updates = []
        self._notified = 0
        self._expect = len(members)
        for rec in members:
            state = self._slice_state(self._carry, self._stage_layers(rec))
            self._send_start(rec, None, state_override=state)
        self._wait_ready(len(members))
        self._send_syn(members)

    def on_notify(self, msg):
        k = self._order[self._ci]
        if msg["layer_id"] == 1:
            self._notified += 1
        n1 = sum(1 for m in self._members(k) if m["layer_id"] == 1)
        if self._notified == n1:
            self._notified = 0
            for rec in self._members(k):
                self._send_pause(rec)

    def on_update(self, msg):
        self._cluster_updates.append(msg)
        if not msg["result"]:
            self.round_result = False
        if len(self._cluster_updates) < self._expect:
            return
        # cluster complete: per-stage weighted FedAvg -> carry to next cluster
        merged: Dict[str, torch.Tensor] = {}
        by_stage: Dict[int, List[tuple]] = {}
        for m in self._cluster_updates:
            if m.get("parameters"):
                by_stage.setdefault(m["layer_id"], []).append(
                    (m["parameters"], float(m["size"] or 1)))
        for stage_list in by_stage.values():
            merged.update(fedavg_state_dicts([sd for sd, _ in stage_list],
                                             weights=[w for _, w in stage_list]))
        self._fold_cluster(merged)
        self._ci += 1
        if self._ci < len(self._order):
            self._start_cluster()
        else:
            self._finish_round()

    def _fold_cluster(self, merged):
        self._cluster_avgs.append(merged)
        self._carry = merged  # seeds the next cluster

    def _round_final_state(self):
        return fedavg_state_dicts(self._cluster_avgs)

    def _finish_round(self):
        if self.save_parameters and self.round_result:
            final = self._round_final_state()
            ok = True
            if self.validation:
                from ..validation import get_val
                ok = get_val(self.model_name, self.data_name, final, self.logger)
            if ok:
                torch.save(final, self.ckpt_path)
                self.round -= 1
            else:
                self._log("Training failed!", warn=True)
                self.round = 0
        else:
            self.round -= 1
        self.round_result = True
        if self.round > 0:
            self.start_round()
        else:
            self.notify_clients(start=False)


class DCSLServer(ClusterFSLServer):
    """Cluster_FSL + synchronous lock-step first layer + SDA concat-batching
    (sda_size = size of the largest cluster, other/DCSL/src/Server.py:138)."""

    def scheduler_overrides(self, rec):
        out = super().scheduler_overrides(rec)
        out["sync-first"] = True
        sch = self.config.get("scheduler") or {}
        out["epochs"] = sch.get("local-round", 1)
        if rec["layer_id"] == self.n_stages:
            sda = max(row[0] for row in self.infor_cluster)
            out["sda-size"] = max(1, int(sda))
        return out


class FlexServer(Server):
    """Periodic aggregation: client FedAvg every t_c rounds, global save every
    t_g; non-aggregation rounds skip the parameter upload (send=False)."""

    def __init__(self, *a, **kw):
        super().__init__(*a, **kw)
        sch = self.config.get("scheduler") or {}
        self.t_c = max(1, int(sch.get("t-c", 1)))
        self.t_g = max(1, int(sch.get("t-g", 1)))
        self._last_full: Optional[Dict[str, torch.Tensor]] = None

    def scheduler_overrides(self, rec):
        out = super().scheduler_overrides(rec)
        out["sync-first"] = True  # FLEX first layer is fully synchronous
        return out

    def _round_no(self) -> int:
        return self.global_round - self.round + 1

    def _agg_round(self) -> bool:
        r = self._round_no()
        return (r % self.t_c == 0) or (self.round == 1)

    def on_notify(self, msg):
        cluster = msg["cluster"]
        if msg["layer_id"] == 1:
            self.first_layer_done_per_cluster[cluster] += 1
        if self.first_layer_done_per_cluster[cluster] == self.infor_cluster[cluster][0]:
            self.first_layer_done_per_cluster[cluster] = 0
            send = self._agg_round()
            for rec in self.list_clients:
                if rec["train"] and rec["cluster"] == cluster:
                    self._send_pause(rec, send=send)

    def on_update(self, msg):
        layer_id = msg["layer_id"]
        cluster = msg["cluster"]
        self.update_counts[layer_id - 1] += 1
        if not msg["result"]:
            self.round_result = False
        if msg.get("parameters") is not None and self.round_result:
            self.collected[cluster][layer_id - 1].append(
                (msg["parameters"], msg["size"]))
        if self.update_counts != self.total_clients:
            return
        self.update_counts = [0] * self.n_stages
        r = self._round_no()
        if self._agg_round() and self.round_result:
            full = self.aggregate()
            self._last_full = full
            if (r % self.t_g == 0) or self.round == 1:
                ok = True
                if self.validation:
                    from ..validation import get_val
                    ok = get_val(self.model_name, self.data_name, full, self.logger)
                if ok:
                    torch.save(full, self.ckpt_path)
        self.round -= 1
        self.collected = [[[] for _ in range(self.n_stages)]
                          for _ in range(self.num_cluster)]
        self.round_result = True
        if self.round > 0:
            self._log(f"[flex] round {self._round_no()}")
            self.notify_clients()
        else:
            self.notify_clients(start=False)


class TwoLSServer(ClusterFSLServer):
    """Two-level clusters: out-clusters sequential (shuffled), in-cluster
    FedAvg folded into the global model by FedAsync alpha-mixing."""

    def __init__(self, *a, **kw):
        super().__init__(*a, **kw)
        self._global: Optional[Dict[str, torch.Tensor]] = None
        self._arrival = 0

    def _cluster_order(self):
        # out-clusters in shuffled order (other/2LS/src/Server.py:141-143);
        # clients' out_cluster defaults to their cluster id when unset
        outs = sorted({c.get("out_cluster") if c.get("out_cluster") is not None
                       else c["cluster"]
                       for c in self.list_clients if c["train"]})
        random.shuffle(outs)
        return outs

    def _members(self, k):
        return [c for c in self.list_clients if c["train"] and
                ((c.get("out_cluster") if c.get("out_cluster") is not None
                  else c["cluster"]) == k)]

    def start_round(self):
        self._arrival = 0
        self._global = self._load_ckpt()
        super().start_round()

    def _fold_cluster(self, merged):
        # FedAsync exponential mixing: alpha = 1/(1 + arrival_rank)
        alpha = 1.0 / (1.0 + self._arrival)
        self._arrival += 1
        if self._global is None:
            self._global = merged
        else:
            out = {}
            for k in merged:
                if k in self._global:
                    g = self._global[k].float()
                    m = merged[k].float()
                    v = (1 - alpha) * g + alpha * m
                    out[k] = v.to(self._global[k].dtype)
                else:
                    out[k] = merged[k]
            for k in self._global:
                out.setdefault(k, self._global[k])
            self._global = out
        self._carry = self._global
        self._cluster_avgs.append(merged)

    def _round_final_state(self):
        return self._global


POLICIES = {
    "main": Server,
    "vanilla": VanillaServer,
    "cluster_fsl": ClusterFSLServer,
    "dcsl": DCSLServer,
    "flex": FlexServer,
    "2ls": TwoLSServer,
}


def make_server(config, control, logger=None, checkpoint_dir="."):
    policy = (config.get("scheduler") or {}).get("policy", "main")
    klass = POLICIES.get(policy)
    if klass is None:
        raise ValueError(f"unknown scheduler policy {policy!r}")
    return klass(config, control, logger=logger, checkpoint_dir=checkpoint_dir)
