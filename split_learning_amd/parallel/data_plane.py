"""Data plane: cut-layer activation / gradient tensor movement between stages.

* LoopbackData — in-process queues passing tensors by reference (CPU tests and
  single-GPU deployments where several stages share one device: zero-copy).
* P2PData — torch.distributed point-to-point transfers, one process per GPU.
  On ROCm the "nccl" backend IS RCCL, so sends ride xGMI links directly;
  the same class runs on "gloo" for multi-process CPU tests.  This replaces
  the reference's pickled-numpy RabbitMQ queues (src/train/VGG16.py:20-53):
  no host round-trip — tensors leave and arrive GPU-resident.

Routing: the reference's shared per-cluster AMQP queue gives multi-consumer
work-stealing for free; p2p is pairwise, so the server assigns static
round-robin producer->consumer edges (the DCSL variant's targeted routing,
other/DCSL/src/Scheduler.py:110-115, is the template) and gradient edges are
the reverse.  Each message is a (header, labels, payload) tensor triple with
static shapes so receivers can pre-post irecvs.
"""

from __future__ import annotations

import collections
import queue
import threading
from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

from .messages import ActivationMsg, GradientMsg

MAX_TRACE = 8  # max pipeline depth for the trace stack


class LoopbackData:
    """Single-process data plane; queue per activation edge / gradient target."""

    def __init__(self):
        self._act: Dict[Tuple[int, int], queue.Queue] = {}
        self._grad: Dict[Tuple[int, int], queue.Queue] = {}
        self._lock = threading.Lock()

    def _q(self, table, key):
        with self._lock:
            if key not in table:
                table[key] = queue.Queue()
            return table[key]

    # activation queue is shared per (producer_stage, cluster) like the
    # reference's intermediate_queue_{layer}_{cluster}
    def send_activation(self, stage: int, cluster: int, msg: ActivationMsg,
                        dst_client: Optional[int] = None):
        self._q(self._act, (stage, cluster)).put(msg)

    def recv_activation(self, prev_stage: int, cluster: int, my_client: int,
                        block=False, timeout=None) -> Optional[ActivationMsg]:
        try:
            return self._q(self._act, (prev_stage, cluster)).get(block=block,
                                                                 timeout=timeout)
        except queue.Empty:
            return None

    def send_gradient(self, stage: int, to_client: int, msg: GradientMsg):
        self._q(self._grad, (stage, to_client)).put(msg)

    def recv_gradient(self, stage: int, client: int, block=False,
                      timeout=None) -> Optional[GradientMsg]:
        try:
            return self._q(self._grad, (stage, client)).get(block=block,
                                                            timeout=timeout)
        except queue.Empty:
            return None

    def flush(self):
        pass


class _Edge:
    """One directed p2p channel with static shapes; keeps an irecv pipelined."""

    def __init__(self, peer: int, shape, batch: int, device, tag_base: int,
                 dtype=torch.float32):
        self.peer = peer
        self.device = device
        self.header = torch.zeros(3 + MAX_TRACE, dtype=torch.int64, device=device)
        self.labels = torch.zeros(batch, dtype=torch.int64, device=device)
        self.payload = torch.zeros(*shape, dtype=dtype, device=device)
        self.works = None

    def post_recv(self):
        w1 = dist.irecv(self.header, src=self.peer)
        w2 = dist.irecv(self.labels, src=self.peer)
        w3 = dist.irecv(self.payload, src=self.peer)
        self.works = (w1, w2, w3)

    def poll(self):
        """Return (header, labels, payload) clones if a message landed, else None."""
        if self.works is None:
            self.post_recv()
        if not self.works[0].is_completed():
            return None
        for w in self.works:
            w.wait()
        out = (self.header.clone(), self.labels.clone(), self.payload.clone())
        self.works = None
        self.post_recv()
        return out

    def wait(self):
        if self.works is None:
            self.post_recv()
        for w in self.works:
            w.wait()
        out = (self.header.clone(), self.labels.clone(), self.payload.clone())
        self.works = None
        self.post_recv()
        return out


class P2PData:
    """torch.distributed p2p data plane (RCCL on GPU, gloo on CPU).

    Construction needs the routing plan the server computed:
      * down_peer: rank to send activations to (None for last stage)
      * up_peers: ranks this stage receives activations from
      * act_shape_out / act_shape_in: payload shapes (batch-major)
      * grad peers mirror activation edges in reverse.
    """

    def __init__(self, my_rank: int, device: torch.device, batch: int,
                 down_peer: Optional[int], up_peers: List[int],
                 act_shape_out, act_shape_in, grad_from_down: bool):
        self.rank = my_rank
        self.device = device
        self.batch = batch
        self.down_peer = down_peer
        self.up_peers = list(up_peers)
        self._pending_sends = collections.deque()
        # recv edges for activations (from each upstream peer)
        self._act_edges = {p: _Edge(p, act_shape_in, batch, device, 0)
                           for p in self.up_peers} if act_shape_in else {}
        # recv edge for gradients (from the downstream peer, same shape as out act)
        self._grad_edge = (_Edge(down_peer, act_shape_out, batch, device, 1)
                           if (grad_from_down and down_peer is not None) else None)
        self._rr = 0

    # -- helpers -----------------------------------------------------------
    def _reap(self):
        while self._pending_sends and all(w.is_completed()
                                          for w in self._pending_sends[0][0]):
            self._pending_sends.popleft()

    def _send_triple(self, dst: int, header, labels, payload):
        w1 = dist.isend(header, dst=dst)
        w2 = dist.isend(labels, dst=dst)
        w3 = dist.isend(payload, dst=dst)
        self._pending_sends.append(((w1, w2, w3), (header, labels, payload)))
        self._reap()

    def _pack_header(self, data_id: int, trace: List[int]) -> torch.Tensor:
        h = torch.zeros(3 + MAX_TRACE, dtype=torch.int64)
        h[0] = data_id
        h[1] = 1
        h[2] = len(trace)
        for i, t in enumerate(trace):
            h[3 + i] = t
        return h.to(self.device, non_blocking=True)

    @staticmethod
    def _unpack_header(h: torch.Tensor):
        hc = h.cpu()
        data_id = int(hc[0])
        tlen = int(hc[2])
        trace = [int(hc[3 + i]) for i in range(tlen)]
        return data_id, trace

    # -- activation / gradient API (mirrors LoopbackData) ------------------
    def send_activation(self, stage: int, cluster: int, msg: ActivationMsg,
                        dst_client: Optional[int] = None):
        dst = self.down_peer if dst_client is None else dst_client
        header = self._pack_header(msg.data_id, msg.trace)
        labels = (msg.labels if msg.labels is not None
                  else torch.zeros(self.batch, dtype=torch.int64))
        labels = labels.to(self.device, non_blocking=True)
        payload = msg.data.to(self.device).contiguous()
        self._send_triple(dst, header, labels, payload)

    def recv_activation(self, prev_stage: int, cluster: int, my_client: int,
                        block=False, timeout=None) -> Optional[ActivationMsg]:
        n = len(self.up_peers)
        if n == 0:
            return None
        for i in range(n):
            peer = self.up_peers[(self._rr + i) % n]
            got = self._act_edges[peer].poll()
            if got is not None:
                self._rr = (self._rr + i + 1) % n
                return self._to_act(got)
        if block:
            # single-peer blocking wait; multi-peer keeps polling
            if n == 1:
                return self._to_act(self._act_edges[self.up_peers[0]].wait())
            while True:
                for peer in self.up_peers:
                    got = self._act_edges[peer].poll()
                    if got is not None:
                        return self._to_act(got)
        return None

    def _to_act(self, triple) -> ActivationMsg:
        header, labels, payload = triple
        data_id, trace = self._unpack_header(header)
        return ActivationMsg(data_id, payload, labels, trace)

    def send_gradient(self, stage: int, to_client: int, msg: GradientMsg):
        header = self._pack_header(msg.data_id, msg.trace)
        labels = torch.zeros(self.batch, dtype=torch.int64, device=self.device)
        self._send_triple(to_client, header, labels, msg.data.contiguous())

    def recv_gradient(self, stage: int, client: int, block=False,
                      timeout=None) -> Optional[GradientMsg]:
        if self._grad_edge is None:
            return None
        got = self._grad_edge.wait() if block else self._grad_edge.poll()
        if got is None:
            return None
        header, _, payload = got
        data_id, trace = self._unpack_header(header)
        return GradientMsg(data_id, payload, trace)

    def flush(self):
        for works, _bufs in self._pending_sends:
            for w in works:
                w.wait()
        self._pending_sends.clear()
