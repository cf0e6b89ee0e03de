"""Native CDNA4 (gfx950) kernel extension loader.

The extension is built IN-TREE as ``split_learning_amd/ops/_sl_kernels*.so`` by
``__graft_entry__.build()`` (hipcc via torch.utils.cpp_extension,
PYTORCH_ROCM_ARCH=gfx950) so the built artifact travels with repo snapshots.

Policy (fail-loud): on a CUDA/HIP tensor the native kernels are REQUIRED —
if the extension is missing we raise instead of silently falling back to
eager PyTorch.  CPU tensors use the plain torch implementations (the module
layer handles that dispatch) so CPU-only tests run anywhere.
"""

from __future__ import annotations

import importlib
import os

_EXT = None
_TRIED = False
_ERR: Exception | None = None


def _try_load():
    global _EXT, _TRIED, _ERR
    if _TRIED:
        return _EXT
    _TRIED = True
    try:
        _EXT = importlib.import_module("split_learning_amd.ops._sl_kernels")
    except Exception as e:  # noqa: BLE001
        _EXT = None
        _ERR = e
    return _EXT


def native_available() -> bool:
    return _try_load() is not None


def native():
    """Return the native module or raise loudly (GPU path must not silently fall back)."""
    ext = _try_load()
    if ext is None:
        raise RuntimeError(
            "split_learning_amd native HIP extension (_sl_kernels) is not built/loadable "
            "but a GPU tensor reached a native op. Build it with "
            "`python -c \"import __graft_entry__; __graft_entry__.build()\"` from the repo "
            f"root. Original import error: {_ERR!r}"
        )
    return ext
