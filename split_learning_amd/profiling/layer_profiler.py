"""Device profiler: per-unit forward times + activation sizes -> profiling.json.

Reference parity (profiling.py:22-120): forward hooks time each numbered unit
and record its output byte size; 30 warm-up passes then a measured pass; the
recorded exe_time keeps the reference's x3 safety factor (profiling.py:73);
"speed" = batch / total-time; "network" = bytes/ns bandwidth estimate.  The
output profiling.json feeds REGISTER -> the server's auto cut-point search
(src/Server.py:115-117,354-362, src/Partition.py).

MI355X-native timing: hipEvents around each unit on GPU (synchronised once per
pass), wall clock on CPU.  The network probe measures real transport bandwidth:
RCCL p2p over xGMI when a process group is up, else device-copy bandwidth.
"""

from __future__ import annotations

import json
import time
from typing import List, Optional

import torch

from ..data.synthetic import SHAPES
from ..models import build_partition


def _dummy_batch(data_name: str, batch: int) -> torch.Tensor:
    shape, _n, dtype, vocab = SHAPES[data_name]
    if dtype == torch.int64:
        return torch.randint(1, vocab, (batch, *shape), dtype=torch.int64)
    return torch.randn(batch, *shape)


def profile_model(model_name: str, data_name: str, batch: int = 4,
                  device: Optional[torch.device] = None, warmup: int = 30):
    """Returns (exe_time ns per unit, size_data bytes per unit, speed)."""
    device = device or torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    model = build_partition(model_name, data_name, [0, 0]).to(device).eval()
    x = _dummy_batch(data_name, batch).to(device)
    use_events = device.type == "cuda"

    units = model.active_units()
    with torch.no_grad():
        for _ in range(warmup):
            model(x)
        if use_events:
            torch.cuda.synchronize()

        exe_time: List[float] = []
        size_data: List[int] = []
        h = x
        t_total0 = time.perf_counter_ns()
        for i in units:
            if use_events:
                ev0 = torch.cuda.Event(enable_timing=True)
                ev1 = torch.cuda.Event(enable_timing=True)
                ev0.record()
            else:
                t0 = time.perf_counter_ns()
            h = _forward_unit(model, i, h)
            if use_events:
                ev1.record()
                torch.cuda.synchronize()
                dt_ns = ev0.elapsed_time(ev1) * 1e6
            else:
                dt_ns = time.perf_counter_ns() - t0
            exe_time.append(float(dt_ns) * 3.0)  # reference x3 factor
            size_data.append(int(h.numel() * h.element_size()))
        t_total = time.perf_counter_ns() - t_total0
    speed = batch / max(t_total, 1)
    return exe_time, size_data, speed


def _forward_unit(model, i, h):
    """Single-unit forward via a temporary partition view."""
    saved_start, saved_end = model.start_layer, model.end_layer
    model.start_layer, model.end_layer = i - 1, i
    try:
        return model(h)
    finally:
        model.start_layer, model.end_layer = saved_start, saved_end


def network_probe(device: Optional[torch.device] = None, sizes_mb=range(1, 10),
                  reps: int = 10) -> float:
    """Bandwidth estimate in bytes/ns.  RCCL p2p over xGMI if a >1-rank process
    group is initialised (timed ping between rank pairs); else device copy."""
    import torch.distributed as dist
    device = device or torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    total_bytes = 0
    total_ns = 0
    if dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1:
        rank = dist.get_rank()
        peer = rank ^ 1
        if peer >= dist.get_world_size():
            peer = (rank + 1) % dist.get_world_size()
        for mb in sizes_mb:
            buf = torch.zeros(mb * 1024 * 1024 // 4, device=device)
            t0 = time.perf_counter_ns()
            for _ in range(reps):
                if rank < peer:
                    dist.send(buf, dst=peer)
                    dist.recv(buf, src=peer)
                else:
                    dist.recv(buf, src=peer)
                    dist.send(buf, dst=peer)
            if device.type == "cuda":
                torch.cuda.synchronize()
            total_ns += time.perf_counter_ns() - t0
            total_bytes += 2 * reps * buf.numel() * 4
    else:
        for mb in sizes_mb:
            src = torch.zeros(mb * 1024 * 1024 // 4, device=device)
            dst = torch.empty_like(src)
            t0 = time.perf_counter_ns()
            for _ in range(reps):
                dst.copy_(src)
            if device.type == "cuda":
                torch.cuda.synchronize()
            total_ns += time.perf_counter_ns() - t0
            total_bytes += reps * src.numel() * 4
    return total_bytes / max(total_ns, 1)


def write_profiling_json(path: str, model_name: str, data_name: str, batch: int,
                         device: Optional[torch.device] = None) -> dict:
    exe_time, size_data, speed = profile_model(model_name, data_name, batch, device)
    network = network_probe(device)
    prof = {"exe_time": exe_time, "size_data": size_data, "speed": speed,
            "network": network}
    with open(path, "w") as f:
        json.dump(prof, f)
    return prof
