"""In-tree build of the _sl_kernels HIP extension for gfx950.

Drives hipcc directly (no hipify, no CUDA shims): every .hip source is native
CDNA4 HIP, compiled with --offload-arch=gfx950, linked against libtorch, and
placed at split_learning_amd/ops/_sl_kernels.so so the built artifact travels
with repo snapshots to GPU boxes.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from concurrent.futures import ThreadPoolExecutor

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")
OUT_SO = os.path.join(HERE, "_sl_kernels.so")
BUILD_DIR = os.path.join(HERE, "_build")

SOURCES = [
    "gemm_f32.hip",
    "conv2d.hip",
    "wino.hip",
    "norm.hip",
    "elementwise.hip",
    "loss.hip",
    "attention.hip",
    "optim.hip",
    "bindings.cpp",
]

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _flags():
    import torch
    import torch.utils.cpp_extension as cpp_ext

    torch_lib = os.path.join(os.path.dirname(torch.__file__), "lib")
    includes = cpp_ext.include_paths(device_type="cuda")  # torch+ROCm include set
    includes.append(sysconfig.get_paths()["include"])
    abi = int(getattr(torch._C, "_GLIBCXX_USE_CXX11_ABI", True))
    cxxflags = [
        "-O3", "-std=c++17", "-fPIC",
        f"--offload-arch={ARCH}",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_EXTENSION_NAME=_sl_kernels",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DUSE_ROCM=1",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
        "-Wno-deprecated-declarations",
        "-fno-gpu-rdc",
    ] + [f"-I{p}" for p in includes]
    if os.environ.get("SLK_MFMA32"):  # 32x32x2 MFMA core variant (A/B experiments)
        cxxflags.append("-DSLK_MFMA32")
    ldflags = [
        "-shared",
        f"-L{torch_lib}", "-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10",
        "-ltorch_hip", "-lc10_hip",
        "-L/opt/rocm/lib", "-lamdhip64",
        f"-Wl,-rpath,{torch_lib}",
    ]
    return cxxflags, ldflags


def _run(cmd):
    proc = subprocess.run(cmd, capture_output=True, text=True)
    if proc.returncode != 0:
        raise RuntimeError(
            f"command failed ({proc.returncode}): {' '.join(cmd)}\n"
            f"stdout:\n{proc.stdout[-4000:]}\nstderr:\n{proc.stderr[-8000:]}")
    return proc


def build(verbose: bool = True, force: bool = False) -> str:
    os.makedirs(BUILD_DIR, exist_ok=True)
    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    cxxflags, ldflags = _flags()

    objs = []
    jobs = []
    for src in SOURCES:
        sp = os.path.join(CSRC, src)
        op = os.path.join(BUILD_DIR, src.replace("/", "_") + ".o")
        objs.append(op)
        deps = [sp] + [os.path.join(CSRC, h) for h in ("common.h", "tile_gemm.h")]
        if (not force and os.path.exists(op)
                and all(os.path.getmtime(op) >= os.path.getmtime(d) for d in deps)):
            continue
        cmd = [hipcc, "-c", sp, "-o", op] + cxxflags
        if src.endswith(".cpp"):
            cmd.append("-x")
            cmd.append("c++")  # host-only TU
        jobs.append(cmd)

    if jobs:
        if verbose:
            print(f"[build] compiling {len(jobs)} TU(s) with {hipcc} for {ARCH}")
        with ThreadPoolExecutor(max_workers=min(8, len(jobs))) as ex:
            list(ex.map(_run, jobs))

    if force or not os.path.exists(OUT_SO) or any(
            os.path.getmtime(o) > os.path.getmtime(OUT_SO) for o in objs):
        if verbose:
            print(f"[build] linking {OUT_SO}")
        _run([hipcc] + objs + ["-o", OUT_SO] + ldflags)
    return OUT_SO


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print("ok:", OUT_SO)
