"""Loader for the in-tree HIP extension (zaremba_amd/_hip*.so).

The extension is built for gfx950 by ``setup.py build_ext --inplace`` (or
``__graft_entry__.build()``) and the resulting .so lives inside the package
directory so it travels with the repo snapshot to GPU boxes.

Policy: on a CUDA/ROCm device the HIP kernels are THE compute path. If the
extension is missing while a GPU is visible, ops raise rather than silently
falling back to eager PyTorch (the eager path is for CPU and for explicit
--engine eager debugging only).
"""

from __future__ import annotations

import importlib
import os

_ext = None
_load_error: Exception | None = None


def _try_load():
    global _ext, _load_error
    if _ext is not None or _load_error is not None:
        return
    try:
        import torch  # noqa: F401  (the extension links against torch libs)

        _ext = importlib.import_module("zaremba_amd._hip")
    except Exception as e:  # pragma: no cover - exercised only without the .so
        _load_error = e


def available() -> bool:
    _try_load()
    return _ext is not None


def ext():
    """Return the extension module, raising loudly if it is absent."""
    _try_load()
    if _ext is None:
        raise RuntimeError(
            "zaremba_amd HIP extension (_hip.so) is not available: "
            f"{_load_error}\nBuild it with `python setup.py build_ext --inplace` "
            "(PYTORCH_ROCM_ARCH=gfx950). The eager fallback is not used on GPU "
            "unless explicitly requested with engine='eager'."
        )
    return _ext


def force_eager_env() -> bool:
    return os.environ.get("ZAREMBA_AMD_FORCE_EAGER", "0") == "1"
