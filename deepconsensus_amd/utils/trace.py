"""rocTX trace ranges for the device hot path.

Parity with the reference's `tf.profiler.experimental.Trace` wrapping of
every train/eval step (model_train_custom_loop.py:248,277) and the
north-star requirement that K2..K14 be visible as named ranges in
rocprof timelines: `range(name)` pushes an NVTX/rocTX range (PyTorch's
torch.cuda.nvtx routes to roctx on ROCm builds), so
`rocprofv3 --marker-trace` shows labeled spans around each kernel group.

Ranges are compiled to no-ops unless DC_TRACE=1 (or enable() is called):
roctxRangePush costs a few hundred ns per call, which the serving inner
loop (tens of thousands of windows/s) should not pay by default.
"""
from __future__ import annotations

import contextlib
import os

_enabled = os.environ.get("DC_TRACE", "0") == "1"
_nvtx = None


def enable(on: bool = True) -> None:
    global _enabled
    _enabled = on


def _get_nvtx():
    global _nvtx
    if _nvtx is None:
        try:
            import torch

            if torch.cuda.is_available():
                _nvtx = torch.cuda.nvtx
            else:
                _nvtx = False
        except Exception:  # pragma: no cover
            _nvtx = False
    return _nvtx


@contextlib.contextmanager
def range(name: str):
    """Context manager: rocTX range around a code region (no-op unless
    tracing is enabled and a GPU is present)."""
    nvtx = _get_nvtx() if _enabled else False
    if not nvtx:
        yield
        return
    nvtx.range_push(name)
    try:
        yield
    finally:
        nvtx.range_pop()


def mark(name: str) -> None:
    nvtx = _get_nvtx() if _enabled else False
    if nvtx:
        nvtx.mark(name)
