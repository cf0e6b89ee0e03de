"""Control plane: small server<->client messages (REGISTER/START/SYN/...).

Two implementations:

* InProcControl — queue.Queue per destination, for single-process (loopback)
  deployments and CPU tests.
* StoreControl — torch.distributed.TCPStore message queues, for one-process-
  per-GPU deployments (and for separate server.py / client.py processes).
  Replaces the reference's RabbitMQ rpc_queue / reply_{id} queues
  (src/Server.py:61,387-396) with atomic counters + pickled blobs; the
  fixed `time.sleep(25)` SYN barrier (src/Server.py:289) becomes a real
  ready-acknowledgement rendezvous.
"""

from __future__ import annotations

import pickle
import queue
import threading
import time
from datetime import timedelta
from typing import Any, Dict, Optional


class ControlPlane:
    def send(self, dst: str, msg: Dict[str, Any]) -> None:
        raise NotImplementedError

    def recv(self, dst: str, block: bool = True, timeout: Optional[float] = None):
        """Pop next message addressed to `dst`; None if non-blocking and empty."""
        raise NotImplementedError


class InProcControl(ControlPlane):
    def __init__(self):
        self._queues: Dict[str, queue.Queue] = {}
        self._lock = threading.Lock()

    def _q(self, dst: str) -> queue.Queue:
        with self._lock:
            if dst not in self._queues:
                self._queues[dst] = queue.Queue()
            return self._queues[dst]

    def send(self, dst, msg):
        self._q(dst).put(msg)

    def recv(self, dst, block=True, timeout=None):
        try:
            return self._q(dst).get(block=block, timeout=timeout)
        except queue.Empty:
            return None


class StoreControl(ControlPlane):
    """Message queues over a torch.distributed TCPStore.

    Per destination `d`: counter key `mqc_<d>` (store.add) and blob keys
    `mq_<d>_<i>`.  `recv` polls the counter and store.get()s the next blob.
    """

    POLL_S = 0.002

    def __init__(self, store):
        self.store = store
        self._read: Dict[str, int] = {}
        # one TCPStore object = one socket; serialise all ops on it (the
        # underlying connection is not safe under concurrent threads)
        self._lock = threading.Lock()

    @staticmethod
    def create(master_addr: str, port: int, is_server: bool, timeout_s: float = 300.0):
        import torch.distributed as dist
        store = dist.TCPStore(master_addr, port, is_master=is_server,
                              timeout=timedelta(seconds=timeout_s),
                              wait_for_workers=False)
        return StoreControl(store)

    # TCPStore rejects payloads > 8 MiB; chunk large blobs (UPDATE messages
    # carry full partition state dicts).
    CHUNK = 4 * 1024 * 1024

    def send(self, dst, msg):
        blob = pickle.dumps(msg, protocol=pickle.HIGHEST_PROTOCOL)
        with self._lock:
            seq = self.store.add(f"mqc_w_{dst}", 1) - 1  # 0-based slot
            if len(blob) <= self.CHUNK:
                self.store.set(f"mq_{dst}_{seq}", blob)
            else:
                n = (len(blob) + self.CHUNK - 1) // self.CHUNK
                for i in range(n):
                    self.store.set(f"mq_{dst}_{seq}_c{i}",
                                   blob[i * self.CHUNK:(i + 1) * self.CHUNK])
                self.store.set(f"mq_{dst}_{seq}", b"__CHUNKED__:%d" % n)

    def recv(self, dst, block=True, timeout=None):
        deadline = None if timeout is None else time.monotonic() + timeout
        while True:
            nxt = self._read.get(dst, 0)
            with self._lock:
                avail = self.store.add(f"mqc_w_{dst}", 0)
                if avail > nxt:
                    blob = self.store.get(f"mq_{dst}_{nxt}")
                    self.store.delete_key(f"mq_{dst}_{nxt}")
                    if blob.startswith(b"__CHUNKED__:"):
                        n = int(blob.split(b":", 1)[1])
                        parts = []
                        for i in range(n):
                            k = f"mq_{dst}_{nxt}_c{i}"
                            parts.append(bytes(self.store.get(k)))
                            self.store.delete_key(k)
                        blob = b"".join(parts)
                    self._read[dst] = nxt + 1
                    return pickle.loads(bytes(blob))
            if not block:
                return None
            if deadline is not None and time.monotonic() > deadline:
                return None
            time.sleep(self.POLL_S)
