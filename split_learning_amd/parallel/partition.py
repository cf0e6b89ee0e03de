"""Auto cut-point search: pick the cut maximising min(stage1, stage2) aggregate
throughput from profiled per-layer exe times and network bandwidths
(reference src/Partition.py:2-21, fed by profiling.json via REGISTER)."""

from __future__ import annotations

from typing import List


def partition(exe_time_layer_1: List[List[float]], net_layer_1: List[float],
              exe_time_layer_2: List[List[float]], net_layer_2: List[float],
              size_data: List[float]) -> List[int]:
    best_speed = 0.0
    best_cut = 0
    for cut in range(len(size_data)):
        size = size_data[cut]
        speed1 = sum(1.0 / (sum(exe[:cut + 1]) + size / comm)
                     for exe, comm in zip(exe_time_layer_1, net_layer_1))
        speed2 = sum(1.0 / (sum(exe[cut + 1:]) + size / comm)
                     for exe, comm in zip(exe_time_layer_2, net_layer_2))
        speed = min(speed1, speed2)
        if speed > best_speed:
            best_cut = cut + 1
            best_speed = speed
    return [best_cut]
