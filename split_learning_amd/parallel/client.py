"""Client runtime: the per-GPU worker that registers, builds its partition,
trains its stage, and ships parameter updates.

Generalises reference src/RpcClient.py:16-146 + src/train/* into one runtime
with pluggable stage loops (schedulers.py) and transports.  One instance runs
per GPU process (RCCL mode) or per thread (loopback mode).
"""

from __future__ import annotations

import copy
import threading
from typing import Any, Dict, Optional

import torch

from ..data import data_loader
from ..models import build_partition
from ..models.lora import apply_lora, merge_and_unload
from .optim import make_optimizer
from .schedulers import StageContext, run_stage


class Inbox:
    """Control-plane receive wrapper with pushback (so stage loops can poll for
    PAUSE without losing other message types)."""

    def __init__(self, control, key: str):
        self.control = control
        self.key = key
        self._stash = []

    def recv(self, _key=None, block=True, timeout=None):
        if self._stash:
            return self._stash.pop(0)
        return self.control.recv(self.key, block=block, timeout=timeout)

    def push_back(self, msg):
        self._stash.append(msg)


class ClientRuntime:
    def __init__(self, client_id: int, layer_id: int, control, plane,
                 device: torch.device, cluster: Optional[int] = None,
                 profile: Optional[dict] = None, logger=None,
                 scheduler_cfg: Optional[dict] = None,
                 plane_factory=None):
        self.client_id = client_id
        self.layer_id = layer_id
        self.control = control
        self.plane = plane
        self.plane_factory = plane_factory  # called with routing dict (p2p mode)
        self.device = device
        self.cluster = cluster
        self.profile = profile or {"speed": 1.0, "exe_time": [], "size_data": [],
                                   "network": 1.0}
        self.logger = logger
        self.scheduler_cfg = scheduler_cfg or {}
        self.inbox = Inbox(control, f"client_{client_id}")

        self.model = None
        self.model_name = None
        self.data_name = None
        self.learning: Dict[str, Any] = {}
        self.train_loader = None
        self.label_count = None
        self.n_stages = None
        self.optimizer = None
        self._routing = None
        self.select = True           # FLEX manual select flag (client --s)
        self.out_cluster = None      # 2LS two-level clusters
        self._fedavg_mode = "control"
        self._group_spec = None      # cached (cluster,stage) group layout
        self._my_group = None
        self._my_group_ids = None

    # ------------------------------------------------------------------
    def register(self):
        self.control.send("server", {
            "action": "REGISTER", "client_id": self.client_id,
            "layer_id": self.layer_id, "profile": self.profile,
            "cluster": self.cluster, "select": self.select,
            "out_cluster": self.out_cluster,
        })

    CONTROL_TIMEOUT_S = 600.0

    def run(self, max_batches: Optional[int] = None, on_step=None):
        """Main loop: handle START/SYN/STOP until the server stops us.

        On a GPU, each client thread gets its OWN HIP stream so colocated
        stages (loopback mode: several clients sharing one device) genuinely
        overlap — the loopback data plane orders cross-stream tensor hand-offs
        with events."""
        if self.device.type == "cuda":
            with torch.cuda.stream(torch.cuda.Stream(device=self.device)):
                return self._run_loop(max_batches, on_step)
        return self._run_loop(max_batches, on_step)

    def _run_loop(self, max_batches=None, on_step=None):
        while True:
            msg = self.inbox.recv(block=True, timeout=self.CONTROL_TIMEOUT_S)
            if msg is None:
                raise TimeoutError(
                    f"client {self.client_id}: no control message within "
                    f"{self.CONTROL_TIMEOUT_S}s — server lost?")
            action = msg.get("action")
            if action == "START":
                self._handle_start(msg)
                self.control.send("server", {"action": "READY",
                                             "client_id": self.client_id,
                                             "layer_id": self.layer_id})
            elif action == "SYN":
                self._handle_syn(max_batches=max_batches, on_step=on_step)
            elif action == "STOP":
                if hasattr(self.plane, "poison_shutdown"):
                    self.plane.poison_shutdown()
                return
            elif action == "PAUSE":
                # late PAUSE outside a stage loop: ignore
                continue
            else:
                raise RuntimeError(f"client {self.client_id}: unexpected {action}")

    # ------------------------------------------------------------------
    def _handle_start(self, msg):
        state_dict = msg.get("parameters")
        self.model_name = msg["model_name"]
        self.data_name = msg["data_name"]
        self.learning = msg["learning"]
        layers = msg["layers"]
        if msg.get("param_bcast"):
            # parameters arrive as ONE RCCL broadcast of the full model over
            # xGMI (bcast.py); rank 0's client feeds the server-staged state
            # into the collective, every rank slices its partition locally
            from . import bcast
            import torch.distributed as dist
            staged = bcast.take_staged("round") if dist.get_rank() == 0 else None
            full = bcast.broadcast_full_state(self.model_name, self.data_name,
                                              staged, self.device)
            part = build_partition(self.model_name, self.data_name, layers)
            state_dict = {k: full[k] for k in part.state_dict().keys()}
        self.cluster = msg.get("cluster", self.cluster) or 0
        self.n_stages = msg["n_stages"]
        refresh = msg.get("refresh", True)
        if self.label_count is None:
            self.label_count = msg.get("label_count")

        self.model = build_partition(self.model_name, self.data_name, layers)
        if state_dict:
            self.model.load_state_dict(state_dict)
        if self.model_name == "BERT":
            # LoRA wrap matching reference peft config (src/RpcClient.py:61-66)
            trainable_extra = ()
            if self.layer_id == self.n_stages:
                trainable_extra = (f"layer{self.model.TOTAL_UNITS}.classifier",)
            apply_lora(self.model, r=8, alpha=16, dropout=0.1,
                       target_modules=("query", "key", "value", "dense"),
                       trainable_extra=trainable_extra)
        self.model.to(self.device)
        self.optimizer = make_optimizer(self.model_name, self.model.parameters(),
                                        self.learning)

        if self.layer_id == 1 and (self.train_loader is None or refresh):
            self.train_loader = data_loader(self.data_name,
                                            self.learning["batch-size"],
                                            self.label_count, train=True,
                                            seed=self.client_id)

        # per-round scheduler policy overrides travel in START (epochs,
        # sync-first, sda-size, time limit — the variant policies set these)
        overrides = msg.get("scheduler") or {}
        self.scheduler_cfg = {**self.scheduler_cfg, **overrides}

        routing = msg.get("routing")
        if routing is not None and self.plane_factory is not None:
            if routing != self._routing:
                self.plane = self.plane_factory(routing)
                self._routing = routing

        # RCCL all-reduce FedAvg: build the (cluster, stage) communicators —
        # collective, so every rank replays the same new_group sequence
        self._fedavg_mode = msg.get("fedavg", "control")
        groups = msg.get("fedavg_groups")
        if self._fedavg_mode == "rccl" and groups is not None:
            import torch.distributed as dist
            if dist.is_available() and dist.is_initialized():
                spec = tuple(tuple(g) for g in groups)
                if spec != self._group_spec:
                    self._group_spec = spec
                    self._my_group = None
                    self._my_group_ids = None
                    for ids in groups:
                        g = dist.new_group(list(ids))
                        if self.client_id in ids:
                            self._my_group = g
                            self._my_group_ids = list(ids)
            else:
                self._fedavg_mode = "control"

    def _handle_syn(self, max_batches=None, on_step=None):
        sch = self.scheduler_cfg
        ctx = StageContext(
            client_id=self.client_id, layer_id=self.layer_id,
            n_stages=self.n_stages, cluster=self.cluster, model=self.model,
            optimizer=self.optimizer, learning=self.learning, plane=self.plane,
            control=self.inbox, device=self.device,
            train_loader=self.train_loader,
            recompute=bool(sch.get("recompute", True)),
            max_batches=max_batches,
            time_limit_s=sch.get("limited-time"),
            clip_grad_norm=sch.get("clip-grad-norm"),
            epochs=int(sch.get("epochs", 1)),
            sync_first=bool(sch.get("sync-first", False)),
            sda_size=int(sch.get("sda-size", 1)),
            on_step=on_step,
            # reference prints the loss each batch (src/train/VGG16.py:168);
            # we keep that behind debug_mode since .item() syncs the device
            log_loss=((lambda loss: self.logger.log_debug(
                f"client {self.client_id} loss: {float(loss.detach()):.6f}"))
                if (self.logger is not None and getattr(self.logger, "debug_mode", False)
                    and self.layer_id == self.n_stages) else None),
        )
        result, size = run_stage(ctx)
        pause_msg = ctx.pause_msg

        if self.layer_id == 1:
            # reference: first stage notifies, then blocks for PAUSE
            self.control.send("server", {"action": "NOTIFY",
                                         "client_id": self.client_id,
                                         "layer_id": self.layer_id,
                                         "cluster": self.cluster})
            deferred = []
            while True:
                msg = self.inbox.recv(block=True)
                if msg.get("action") == "PAUSE":
                    pause_msg = msg
                    break
                deferred.append(msg)
            for m in deferred:
                self.inbox.push_back(m)

        # FLEX periodic aggregation: PAUSE may carry send=False, in which case
        # the parameter upload is skipped this round (other/FLEX/src/Server.py:
        # 138-143, src/RpcClient.py:103-120)
        send_params = True
        if pause_msg is not None and pause_msg.get("send") is False:
            send_params = False

        sd = None
        if send_params:
            model = self.model
            if self.model_name == "BERT":
                model = merge_and_unload(model)
                self.model = model
            if self._fedavg_mode == "rccl" and self._my_group is not None:
                # group all-reduce weighted average (xGMI); only the group
                # representative ships the (already averaged) dict to the server
                from .fedavg import allreduce_fedavg_
                allreduce_fedavg_(model, float(max(size, 1)), group=self._my_group)
                if self.client_id != min(self._my_group_ids):
                    sd = None
                else:
                    sd = {k: v.detach().to("cpu")
                          for k, v in model.state_dict().items()}
            else:
                sd = copy.deepcopy(model.state_dict())
                sd = {k: v.detach().to("cpu") for k, v in sd.items()}
        self.control.send("server", {
            "action": "UPDATE", "client_id": self.client_id,
            "layer_id": self.layer_id, "cluster": self.cluster,
            "result": result, "size": size, "parameters": sd,
            "message": "Sent parameters to Server",
        })


def run_client_thread(runtime: ClientRuntime, max_batches=None, on_step=None):
    t = threading.Thread(target=runtime.run,
                         kwargs={"max_batches": max_batches, "on_step": on_step},
                         daemon=True)
    t.start()
    return t
