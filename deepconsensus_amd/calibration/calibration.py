"""Base-quality calibration (piecewise linear transform).

Parity: deepconsensus/quality_calibration/calibration_lib.py:35-99 — a
"threshold,w,b" string parses into a transform applying q*w+b to qualities
above the threshold (or everywhere when threshold == 0). v1.2 ships
dc_calibration="0,1.197654,-0.99781" in params.json.
"""
from __future__ import annotations

import dataclasses

import numpy as np


@dataclasses.dataclass
class QualityCalibrationValues:
    enabled: bool
    threshold: float
    w: float
    b: float


def parse_calibration_string(calibration: str) -> QualityCalibrationValues:
    """Parses "threshold,w,b" (or "skip") into calibration values."""
    if calibration == "skip":
        return QualityCalibrationValues(
            enabled=False, threshold=0.0, w=1.0, b=0.0
        )
    parsed = calibration.split(",")
    if len(parsed) != 3:
        raise ValueError(
            "Malformed calibration string. Expected 3 values (or 'skip' to "
            f"perform no quality calibration): {calibration!r}"
        )
    return QualityCalibrationValues(
        enabled=True,
        threshold=float(parsed[0]),
        w=float(parsed[1]),
        b=float(parsed[2]),
    )


def calibrate_quality_scores(
    quality_scores: np.ndarray,
    calibration_values: QualityCalibrationValues,
) -> np.ndarray:
    """Applies q*w+b above the threshold (everywhere when threshold == 0)."""
    if calibration_values.threshold == 0:
        return quality_scores * calibration_values.w + calibration_values.b
    w_values = np.where(
        quality_scores > calibration_values.threshold,
        calibration_values.w,
        1.0,
    )
    b_values = np.where(
        quality_scores > calibration_values.threshold,
        calibration_values.b,
        0.0,
    )
    return quality_scores * w_values + b_values
