"""Driver-contract tests for bench.py.

The round driver runs `python bench.py --gpus N --steps K --warmup W` (N=1
directly; N>1 via `python -m torch.distributed.run --nnodes=1
--nproc-per-node N --master-addr 127.0.0.1 ... bench.py ...`) and parses ONE
JSON line from rank 0.  These tests execute those exact invocations on CPU
(gloo fallback) and validate the contract fields, so a schema or rendezvous
regression is caught here rather than at round end on the GPU box."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _json_line(out: str) -> dict:
    for line in out.splitlines():
        line = line.strip()
        if line.startswith("{") and '"metric"' in line:
            return json.loads(line)
    raise AssertionError(f"no bench JSON line in output:\n{out[-2000:]}")


def _check(rec: dict, n: int):
    assert REQUIRED <= set(rec), f"missing fields: {REQUIRED - set(rec)}"
    assert rec["metric"] == "images/sec (whole node) VGG16/CIFAR10 cut=7"
    assert rec["n_gpus"] == n
    assert rec["higher_is_better"] is True
    assert rec["scaling"] == "weak"
    assert rec["dtype"] == "fp32"
    assert rec["value"] > 0 and rec["ms_per_step"] > 0
    cfg = rec["config"]
    assert cfg["model"] == "VGG16_CIFAR10"
    assert cfg["cut_layer"] == 7
    assert cfg["global_batch"] == 32 * max(n // 2, 1)
    assert f"dp{max(n // 2, 1)}" in cfg["parallelism"]


@pytest.mark.timeout(600)
def test_bench_n1_cpu_contract():
    p = subprocess.run([sys.executable, "bench.py", "--steps", "2",
                        "--warmup", "1"],
                       capture_output=True, text=True, cwd=REPO, timeout=550)
    assert p.returncode == 0, p.stderr[-2000:]
    _check(_json_line(p.stdout), 1)


@pytest.mark.timeout(900)
def test_bench_n2_torchrun_cpu_contract():
    """The driver's exact N=2 launch shape (gloo fallback on CPU)."""
    p = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29417", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, cwd=REPO, timeout=850)
    assert p.returncode == 0, p.stderr[-2000:]
    _check(_json_line(p.stdout), 2)


@pytest.mark.timeout(900)
def test_bench_n4_torchrun_cpu_contract():
    """4-rank rendezvous shape (2 pipelines) on gloo: the same launch the
    driver uses for the SCALE curve, so a multi-pipeline regression (rank
    mapping, eager comm warmup ordering, poison-drain) is caught on CPU."""
    p = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", "29423", "bench.py", "--gpus", "4",
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, cwd=REPO, timeout=850)
    assert p.returncode == 0, p.stderr[-2000:]
    _check(_json_line(p.stdout), 4)
