"""RCCL parameter broadcast for START.

Replaces the per-client chunked TCPStore state-dict blobs with ONE
ncclBroadcast of the flattened full model over xGMI (SURVEY §2.4 "START
param broadcast as ncclBroadcast" row; reference ships per-client pickled
dicts through RabbitMQ, src/Server.py:262-272).

The server thread runs inside rank 0's process: it STAGES the full state
dict here before sending START, and rank 0's client feeds it into the
collective.  Every rank derives the flat layout deterministically from the
model class (sorted keys, known shapes), so no metadata crosses the wire —
one dense [total_numel] fp32 broadcast at xGMI bandwidth.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.distributed as dist

_STAGED: Dict[str, dict] = {}


def stage_full_state(tag: str, full_state) -> None:
    _STAGED[tag] = full_state


def take_staged(tag: str):
    return _STAGED.pop(tag, None)


def flat_spec(model_name: str, data_name: str):
    from ..models import build_partition
    sd = build_partition(model_name, data_name, [0, 0]).state_dict()
    return [(k, sd[k].shape, sd[k].dtype) for k in sorted(sd.keys())]


def broadcast_full_state(model_name: str, data_name: str,
                         full_state: Optional[dict], device,
                         src: int = 0) -> Dict[str, torch.Tensor]:
    """Collective: every rank calls this; rank `src` supplies full_state.

    Integer buffers (num_batches_tracked) ride as fp32 — exact below 2^24.
    Returns the full state dict on every rank (CPU tensors, original dtypes).
    """
    spec = flat_spec(model_name, data_name)
    total = sum(int(torch.Size(s).numel()) for _, s, _ in spec)
    dev = device if (device.type == "cuda" and dist.get_backend() == "nccl") \
        else torch.device("cpu")
    flat = torch.empty(total, dtype=torch.float32, device=dev)
    if dist.get_rank() == src:
        assert full_state is not None, "src rank has no staged state"
        off = 0
        for k, shape, _dt in spec:
            t = full_state[k].detach().float().reshape(-1)
            flat[off:off + t.numel()].copy_(t)
            off += t.numel()
    dist.broadcast(flat, src=src)
    flat = flat.cpu()
    out: Dict[str, torch.Tensor] = {}
    off = 0
    for k, shape, dt in spec:
        n = int(torch.Size(shape).numel())
        t = flat[off:off + n].reshape(shape)
        out[k] = t.to(dt) if dt.is_floating_point else t.round().to(dt)
        off += n
    return out
