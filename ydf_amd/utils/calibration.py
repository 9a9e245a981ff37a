"""Isotonic (PAV) probability calibration.

Capability analogue of the reference's smoothed PAV calibration
(utils/smoothed_pav_calibration_{fit,inference}): fits a monotone step
function from scores to probabilities with the pool-adjacent-violators
algorithm, with optional interpolation between step midpoints.
"""
from __future__ import annotations

import dataclasses
from typing import Optional

import numpy as np


@dataclasses.dataclass
class PavCalibration:
    """Fitted isotonic calibration: thresholds (ascending score bin
    upper-bounds' midpoints) and calibrated values."""

    scores: np.ndarray   # representative score per step (ascending)
    values: np.ndarray   # calibrated probability per step (non-decreasing)

    def apply(self, s: np.ndarray, interpolate: bool = True) -> np.ndarray:
        s = np.asarray(s, dtype=np.float64)
        if len(self.scores) == 0:
            return np.full_like(s, 0.5)
        if not interpolate:
            idx = np.clip(np.searchsorted(self.scores, s), 0,
                          len(self.values) - 1)
            return self.values[idx]
        return np.interp(s, self.scores, self.values)

    def to_json(self) -> dict:
        return {"scores": self.scores.tolist(),
                "values": self.values.tolist()}

    @classmethod
    def from_json(cls, d: dict) -> "PavCalibration":
        return cls(scores=np.asarray(d["scores"], dtype=np.float64),
                   values=np.asarray(d["values"], dtype=np.float64))


def fit_pav(scores: np.ndarray, labels: np.ndarray,
            weights: Optional[np.ndarray] = None) -> PavCalibration:
    """Pool-adjacent-violators over (score, 0/1 label) pairs."""
    s = np.asarray(scores, dtype=np.float64)
    y = np.asarray(labels, dtype=np.float64)
    w = (np.ones_like(s) if weights is None
         else np.asarray(weights, dtype=np.float64))
    order = np.argsort(s, kind="mergesort")
    s, y, w = s[order], y[order], w[order]
    # blocks as (value_sum, weight, score_sum) stacks merged on violation
    vals, wts, sco = [], [], []
    for i in range(len(s)):
        vals.append(y[i] * w[i])
        wts.append(w[i])
        sco.append(s[i] * w[i])
        while len(vals) > 1 and (vals[-2] / wts[-2]) >= (vals[-1] / wts[-1]):
            vals[-2] += vals[-1]
            wts[-2] += wts[-1]
            sco[-2] += sco[-1]
            vals.pop(); wts.pop(); sco.pop()
    vals = np.asarray(vals)
    wts = np.asarray(wts)
    sco = np.asarray(sco)
    return PavCalibration(scores=sco / wts, values=vals / wts)
