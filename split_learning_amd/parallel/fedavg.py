"""FedAvg parameter aggregation.

CPU path: weighted state-dict average with NaN zero-fill and integer-dtype
round-back — exact semantics of reference src/Utils.py:35-66.

GPU path (MI355X-native): same-stage client GPUs all-reduce
sum(weight_i * size_i) and sum(size_i) over an RCCL communicator group and
divide — replacing the reference's ship-everything-to-server loop
(src/Server.py:398-408) with xGMI-bandwidth collectives.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist


def fedavg_state_dicts(state_dicts: List[Dict[str, torch.Tensor]],
                       weights: Optional[List[float]] = None) -> Dict[str, torch.Tensor]:
    num = len(state_dicts)
    if weights is None:
        weights = [1.0] * num
    total_w = float(sum(weights))
    all_keys = set().union(*(sd.keys() for sd in state_dicts))
    out: Dict[str, torch.Tensor] = {}
    for key in sorted(all_keys):
        acc = None
        for sd, w in zip(state_dicts, weights):
            if key not in sd:
                continue
            t = sd[key].float()
            # NaN -> 0 ONLY (reference src/Utils.py:51-52 zero-fills just NaNs;
            # +/-inf propagates there, so don't clamp it here either)
            if torch.isnan(t).any():
                t = torch.where(torch.isnan(t), torch.zeros_like(t), t)
            t = t * w
            acc = t if acc is None else acc + t
        avg = acc / total_w
        orig = next(sd[key] for sd in state_dicts if key in sd)
        if orig.dtype in (torch.int8, torch.int16, torch.int32, torch.int64, torch.bool):
            avg = avg.round().to(orig.dtype)
        else:
            avg = avg.to(orig.dtype)
        out[key] = avg
    return out


def allreduce_fedavg_(model: torch.nn.Module, my_size: float,
                      group=None) -> Dict[str, torch.Tensor]:
    """In-place RCCL weighted average of a stage group's parameters+buffers.

    Every rank in `group` calls this with its sample count; parameters are
    flattened into one bucket, scaled by size, all-reduced (one large
    collective per stage group — xGMI ring is per-link bound, so few large
    ops beat many small ones), divided by the reduced total size, and copied
    back.  Returns the averaged state dict (every rank gets it).
    """
    sd = model.state_dict()
    keys = sorted(sd.keys())
    float_keys = [k for k in keys if sd[k].is_floating_point()]
    other_keys = [k for k in keys if not sd[k].is_floating_point()]
    device = next(iter(sd.values())).device

    size_t = torch.tensor([my_size], dtype=torch.float32, device=device)
    dist.all_reduce(size_t, group=group)
    total = size_t.item()

    if float_keys:
        flat = torch.cat([sd[k].detach().float().reshape(-1) for k in float_keys])
        # NaN -> 0 only; leave +/-inf to propagate (reference src/Utils.py:51-52)
        torch.where(torch.isnan(flat), torch.zeros_like(flat), flat, out=flat)
        flat.mul_(my_size)
        dist.all_reduce(flat, group=group)
        flat.div_(total)
        off = 0
        with torch.no_grad():
            for k in float_keys:
                n = sd[k].numel()
                sd[k].copy_(flat[off:off + n].view_as(sd[k]))
                off += n
    # integer buffers (num_batches_tracked): weighted-average + round, matching
    # fedavg_state_dicts' dtype handling
    for k in other_keys:
        t = sd[k].detach().float() * my_size
        dist.all_reduce(t, group=group)
        with torch.no_grad():
            sd[k].copy_((t / total).round().to(sd[k].dtype))
    return sd
