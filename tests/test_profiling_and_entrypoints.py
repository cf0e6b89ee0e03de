"""Profiler subsystem + entrypoint plumbing tests (CPU)."""

import json
import subprocess
import sys

import torch

from split_learning_amd.models import get_model_class
from split_learning_amd.profiling import profile_model, write_profiling_json
from split_learning_amd.parallel.partition import partition


def test_profile_model_vgg16():
    exe, sizes, speed = profile_model("VGG16", "CIFAR10", batch=2, warmup=1)
    n_units = get_model_class("VGG16", "CIFAR10").TOTAL_UNITS
    assert len(exe) == n_units and len(sizes) == n_units
    assert all(t > 0 for t in exe)
    # cut=7 boundary is [B,64,16,16] fp32
    assert sizes[6] == 2 * 64 * 16 * 16 * 4
    assert speed > 0


def test_profiling_json_feeds_partition(tmp_path):
    p = tmp_path / "profiling.json"
    prof = write_profiling_json(str(p), "VGG16", "CIFAR10", 2)
    loaded = json.load(open(p))
    assert set(loaded) == {"exe_time", "size_data", "speed", "network"}
    # auto cut-point search consumes exactly this shape of data
    cut = partition([prof["exe_time"]], [prof["network"]],
                    [prof["exe_time"]], [prof["network"]], prof["size_data"])
    assert len(cut) == 1 and 1 <= cut[0] <= 52


def test_profiling_cli(tmp_path):
    out = tmp_path / "profiling.json"
    r = subprocess.run([sys.executable, "profiling.py", "--model", "ViT",
                       "--size", "2", "--out", str(out)],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    prof = json.load(open(out))
    assert len(prof["exe_time"]) == 12
