"""Data plane: cut-layer activation / gradient tensor movement between stages.

* LoopbackData — in-process queues passing tensors by reference (CPU tests and
  single-GPU deployments where several stages share one device: zero-copy).
* P2PData — torch.distributed point-to-point transfers, one process per GPU.
  On ROCm the "nccl" backend IS RCCL, so sends ride xGMI links directly;
  the same class runs on "gloo" for multi-process CPU tests.  This replaces
  the reference's pickled-numpy RabbitMQ queues (src/train/VGG16.py:20-53):
  no host round-trip — tensors leave and arrive GPU-resident.

Deadlock discipline (RCCL kernels spin until the peer joins, and all ops of
one communicator execute in enqueue order on its internal stream):
  1. forward (activation) and backward (gradient) traffic use SEPARATE
     process groups -> separate RCCL comms/streams, so cross-direction
     enqueue-order cycles cannot form;
  2. consumers PRE-POST a ring of `depth` irecvs per edge (depth >= the
     pipeline's control-count), so producers can run ahead;
  3. at STOP, producers send one poison message per owned edge so outstanding
     pre-posted irecvs complete before process-group teardown.

Routing: the reference's shared per-cluster AMQP queue gives multi-consumer
work-stealing for free; p2p is pairwise, so the server assigns static
round-robin producer->consumer edges (the DCSL variant's targeted routing,
other/DCSL/src/Scheduler.py:110-115, is the template) and gradient edges are
the reverse.  Each message is a (header, labels, payload) tensor triple with
static shapes so receivers can pre-post.
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
    """Single-process data plane; queue per activation edge / gradient target.

    Tensors pass by reference (zero-copy).  When clients run on per-thread HIP
    streams of ONE GPU (colocated stages overlap like the N=1 bench), each
    message carries a CUDA event recorded on the sender's stream; the receiver
    waits on it before touching the tensor — cross-stream hand-off is
    event-ordered, never implicit."""

    def __init__(self):
        self._act: Dict[Tuple[int, int], queue.Queue] = {}
        self._grad: Dict[Tuple[int, int], queue.Queue] = {}
        self._lock = threading.Lock()

    def _q(self, table, key):
        with self._lock:
            if key not in table:
                table[key] = queue.Queue()
            return table[key]

    @staticmethod
    def _wrap(msg):
        if msg.data is not None and msg.data.is_cuda:
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream())
            return (msg, ev)
        return (msg, None)

    @staticmethod
    def _unwrap(item):
        msg, ev = item
        if ev is not None:
            cur = torch.cuda.current_stream()
            cur.wait_event(ev)
            # the consumer stream now uses storage the sender's stream
            # allocated: tell the caching allocator so the block is not
            # recycled by the sender before these kernels finish
            if msg.data is not None and msg.data.is_cuda:
                msg.data.record_stream(cur)
            labels = getattr(msg, "labels", None)
            if labels is not None and labels.is_cuda:
                labels.record_stream(cur)
        return msg

    def send_activation(self, stage: int, cluster: int, msg: ActivationMsg,
                        dst_client: Optional[int] = None):
        self._q(self._act, (stage, cluster)).put(self._wrap(msg))

    def recv_activation(self, prev_stage: int, cluster: int, my_client: int,
                        block=False, timeout=None) -> Optional[ActivationMsg]:
        try:
            item = self._q(self._act, (prev_stage, cluster)).get(block=block,
                                                                 timeout=timeout)
        except queue.Empty:
            return None
        return self._unwrap(item)

    def send_gradient(self, stage: int, to_client: int, msg: GradientMsg):
        self._q(self._grad, (stage, to_client)).put(self._wrap(msg))

    def recv_gradient(self, stage: int, client: int, block=False,
                      timeout=None) -> Optional[GradientMsg]:
        try:
            item = self._q(self._grad, (stage, client)).get(block=block,
                                                            timeout=timeout)
        except queue.Empty:
            return None
        return self._unwrap(item)

    def flush(self):
        pass


class RecvRing:
    """Pre-posted irecv ring for one (peer, shape) channel.

    Message k is posted/sent with tags (3k, 3k+1, 3k+2) for its
    header/labels/payload: gloo matches concurrent p2p ops by tag (same-tag
    outstanding recvs match nondeterministically), while RCCL ignores tags but
    matches strictly FIFO per pair — so sequence tags make BOTH backends
    deterministic."""

    def __init__(self, peer: int, shape, batch: int, device, group, depth: int,
                 dtype=torch.float32):
        self.peer = peer
        self.group = group
        self.seq = 0
        self.depth = depth
        # gloo's Work.is_completed() never reflects async recv completion
        # (observed: buffers filled, flag stays False), so on gloo a helper
        # thread blocking-waits each message and feeds a python queue; on
        # nccl/RCCL is_completed() is CUDA-event-backed and polling works.
        self.threaded = dist.get_backend(group) == "gloo" if group is not None \
            else dist.get_backend() == "gloo"
        self.slots = []
        for _ in range(depth):
            self.slots.append({
                "header": torch.zeros(3 + MAX_TRACE, dtype=torch.int64, device=device),
                "labels": torch.zeros(batch, dtype=torch.int64, device=device),
                "payload": torch.zeros(*shape, dtype=dtype, device=device),
                "works": None,
            })
        self.order = collections.deque()
        for i in range(depth):
            self._post(i)
        if self.threaded:
            self._q: queue.Queue = queue.Queue()
            self._stop = False
            self._thread = threading.Thread(target=self._pump, daemon=True)
            self._thread.start()

    def _post(self, i):
        s = self.slots[i]
        t = self.seq * 3
        self.seq += 1
        w1 = dist.irecv(s["header"], src=self.peer, group=self.group, tag=t)
        w2 = dist.irecv(s["labels"], src=self.peer, group=self.group, tag=t + 1)
        w3 = dist.irecv(s["payload"], src=self.peer, group=self.group, tag=t + 2)
        s["works"] = (w1, w2, w3)
        self.order.append(i)

    def _take(self):
        """Blocking-wait the oldest posted message; no repost."""
        i = self.order[0]
        s = self.slots[i]
        for w in s["works"]:
            w.wait()
        out = (s["header"].clone(), s["labels"].clone(), s["payload"].clone())
        self.order.popleft()
        return i, out

    def _consume(self):
        i, out = self._take()
        if int(out[0][0]) != -1:  # poison messages are never reposted
            self._post(i)
        return out

    def _pump(self):
        while not self._stop:
            triple = self._consume()
            if int(triple[0][0]) == -1:
                self._q.put(None)
                return
            self._q.put(triple)

    def poll(self):
        if self.threaded:
            try:
                return self._q.get_nowait()
            except queue.Empty:
                return None
        i = self.order[0]
        if not all(w.is_completed() for w in self.slots[i]["works"]):
            return None
        return self._consume()

    def wait(self):
        if self.threaded:
            return self._q.get(block=True)
        return self._consume()

    def drain(self):
        """Consume every outstanding posted recv (the peer sends exactly
        `depth` poison messages at shutdown, so this terminates)."""
        if self.threaded:
            self._stop = True
            self._thread.join(timeout=60.0)
        while self.order:
            self._take()


class P2PData:
    """torch.distributed p2p data plane (RCCL on GPU, gloo on CPU)."""

    def __init__(self, my_rank: int, device: torch.device, batch: int,
                 down_peer: Optional[int], up_peers: List[int],
                 act_shape_out, act_shape_in, grad_from_down: bool,
                 group_fwd=None, group_bwd=None, depth: int = 4):
        self.rank = my_rank
        self.device = device
        self.batch = batch
        self.down_peer = down_peer
        self.up_peers = list(up_peers)
        self.group_fwd = group_fwd
        self.group_bwd = group_bwd
        self._pending_sends = collections.deque()
        self._send_seq: Dict[tuple, int] = {}
        self._act_rings = {p: RecvRing(p, act_shape_in, batch, device, group_fwd,
                                       depth)
                           for p in self.up_peers} if act_shape_in else {}
        self._grad_ring = (RecvRing(down_peer, act_shape_out, batch, device,
                                    group_bwd, depth)
                           if (grad_from_down and down_peer is not None) else None)
        self._rr = 0

    # -- send helpers -------------------------------------------------------
    def _reap(self):
        while self._pending_sends and all(w.is_completed()
                                          for w in self._pending_sends[0][0]):
            self._pending_sends.popleft()

    def _send_triple(self, dst, header, labels, payload, group):
        key = (dst, id(group))
        seq = self._send_seq.get(key, 0)
        self._send_seq[key] = seq + 1
        t = seq * 3
        w1 = dist.isend(header, dst=dst, group=group, tag=t)
        w2 = dist.isend(labels, dst=dst, group=group, tag=t + 1)
        w3 = dist.isend(payload, dst=dst, group=group, tag=t + 2)
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
        return int(hc[0]), [int(hc[3 + i]) for i in range(int(hc[2]))]

    # -- activation / gradient API (mirrors LoopbackData) -------------------
    def send_activation(self, stage: int, cluster: int, msg: ActivationMsg,
                        dst_client: Optional[int] = None):
        dst = self.down_peer if dst_client is None else dst_client
        header = self._pack_header(msg.data_id, msg.trace)
        labels = (msg.labels if msg.labels is not None
                  else torch.zeros(self.batch, dtype=torch.int64))
        labels = labels.to(self.device, non_blocking=True)
        payload = msg.data.to(self.device).contiguous()
        self._send_triple(dst, header, labels, payload, self.group_fwd)

    def recv_activation(self, prev_stage: int, cluster: int, my_client: int,
                        block=False, timeout=None) -> Optional[ActivationMsg]:
        n = len(self.up_peers)
        if n == 0:
            return None
        import time as _time
        t0 = _time.monotonic() if (block and timeout) else None
        empty_sweeps = 0
        while True:
            for i in range(n):
                peer = self.up_peers[(self._rr + i) % n]
                got = self._act_rings[peer].poll()
                if got is not None:
                    self._rr = (self._rr + i + 1) % n
                    header, labels, payload = got
                    data_id, trace = self._unpack_header(header)
                    return ActivationMsg(data_id, payload, labels, trace)
            if not block:
                return None
            if t0 is not None and _time.monotonic() - t0 > timeout:
                raise TimeoutError(
                    f"P2PData.recv_activation: rank {self.rank} waited "
                    f"{timeout:.0f}s on peers {self.up_peers} — producer lost?")
            # spin-then-yield: don't burn the host core while RCCL completes
            empty_sweeps += 1
            if empty_sweeps >= 512:
                _time.sleep(0.0002)

    def send_gradient(self, stage: int, to_client: int, msg: GradientMsg):
        header = self._pack_header(msg.data_id, msg.trace)
        labels = torch.zeros(self.batch, dtype=torch.int64, device=self.device)
        self._send_triple(to_client, header, labels, msg.data.contiguous(),
                          self.group_bwd)

    def recv_gradient(self, stage: int, client: int, block=False,
                      timeout=None) -> Optional[GradientMsg]:
        if self._grad_ring is None:
            return None
        if block and timeout:
            import time as _time
            t0 = _time.monotonic()
            got = None
            empty = 0
            while got is None:
                got = self._grad_ring.poll()
                if got is None:
                    if _time.monotonic() - t0 > timeout:
                        raise TimeoutError(
                            f"P2PData.recv_gradient: rank {self.rank} waited "
                            f"{timeout:.0f}s on peer {self.down_peer} — "
                            f"consumer lost?")
                    empty += 1
                    if empty >= 512:
                        _time.sleep(0.0002)
        else:
            got = self._grad_ring.wait() if block else self._grad_ring.poll()
        if got is None:
            return None
        header, _, payload = got
        data_id, trace = self._unpack_header(header)
        return GradientMsg(data_id, payload, trace)

    # -- shutdown -----------------------------------------------------------
    def poison_shutdown(self):
        """Satisfy every pre-posted irecv: each sender pushes `depth` dummy
        messages per owned edge, then receivers drain."""
        if self.down_peer is not None and self._grad_ring is not None:
            depth = len(self._grad_ring.slots)
        else:
            depth = 4
        if self.down_peer is not None:
            # I produce activations for down_peer's act ring
            for _ in range(depth):
                self.send_activation(0, 0, ActivationMsg(
                    -1, torch.zeros(*self._grad_ring.slots[0]["payload"].shape,
                                    device=self.device)
                    if self._grad_ring else torch.zeros(1, device=self.device),
                    None, []))
        for peer in self.up_peers:
            # I produce gradients for peer's grad ring
            ring = self._act_rings[peer]
            for _ in range(len(ring.slots)):
                self._send_triple(
                    peer, self._pack_header(-1, []),
                    torch.zeros(self.batch, dtype=torch.int64, device=self.device),
                    torch.zeros_like(ring.slots[0]["payload"]), self.group_bwd)
        self.flush()
        for ring in self._act_rings.values():
            ring.drain()
        if self._grad_ring is not None:
            self._grad_ring.drain()

    def flush(self):
        for works, _bufs in self._pending_sends:
            for w in works:
                w.wait()
        self._pending_sends.clear()
