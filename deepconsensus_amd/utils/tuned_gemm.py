"""hipBLASLt GEMM algorithm selection via PyTorch TunableOp.

The default hipBLASLt heuristics pick poor algorithms for the training
wgrad shapes (tall-skinny, K = batch*len): the out-projection wgrad ran
at ~4% of peak. A one-off on-device tuning pass
(PYTORCH_TUNABLEOP_TUNING=1, scripts/train_bench.py) produced
models/tunableop_gfx950.csv; loading it at startup is worth ~7% of the
training step (106.0 -> 99.2 ms at batch 4096; journal). Tuning stays
OFF at runtime — unknown shapes fall back to the normal heuristics.

DC_TUNED_GEMM=0 disables. Re-tune after a ROCm/hipBLASLt upgrade with:
  PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
  PYTORCH_TUNABLEOP_FILENAME=out.csv python scripts/train_bench.py ...
"""
from __future__ import annotations

import os

_ENABLED = None


def enable_tuned_gemms() -> bool:
    """Idempotently load the in-repo TunableOp results (GPU only)."""
    global _ENABLED
    if _ENABLED is not None:
        return _ENABLED
    _ENABLED = False
    if os.environ.get("DC_TUNED_GEMM", "1") == "0":
        return False
    try:
        import torch

        if not torch.cuda.is_available():
            return False
        csv = os.path.join(
            os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            "models",
            "tunableop_gfx950.csv",
        )
        if not os.path.exists(csv):
            return False
        t = torch.cuda.tunable
        t.enable(True)
        t.tuning_enable(False)
        t.read_file(csv)
        _ENABLED = True
    except Exception:  # pragma: no cover - best effort
        _ENABLED = False
    return _ENABLED
