"""HIP/CDNA4 op loading for DeepConsensus-AMD.

On a GPU machine the HIP extension is REQUIRED for the device hot path: ops
fail loudly if it is missing rather than silently falling back to eager
PyTorch. CPU paths always use the torch reference implementations.
"""
from __future__ import annotations

import os

_ext = None
_ext_error: str | None = None


def _try_load():
    global _ext, _ext_error
    if _ext is not None or _ext_error is not None:
        return
    build_dir = os.path.join(os.path.dirname(__file__), "_build")
    so_path = os.path.join(build_dir, "dc_hip_kernels.so")
    try:
        if os.path.exists(so_path):
            import importlib.util

            spec = importlib.util.spec_from_file_location(
                "dc_hip_kernels", so_path
            )
            import torch  # noqa: F401  (torch symbols must be loaded first)

            mod = importlib.util.module_from_spec(spec)
            spec.loader.exec_module(mod)
            _ext = mod
        else:
            from deepconsensus_amd.ops import build as _build

            _ext = _build.build()
    except Exception as e:  # pragma: no cover
        _ext_error = f"{type(e).__name__}: {e}"


def get_ext(required: bool = False):
    """Returns the HIP extension module, or None (raises when required)."""
    _try_load()
    if _ext is None and required:
        raise RuntimeError(
            "DeepConsensus-AMD HIP extension (dc_hip_kernels) is not "
            "available on this GPU machine — the native kernel path is "
            "mandatory on-device. Build it with "
            "`python -m deepconsensus_amd.ops.build`. "
            f"Last load error: {_ext_error}"
        )
    return _ext


def have_ext() -> bool:
    _try_load()
    return _ext is not None
