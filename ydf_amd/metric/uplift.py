"""Uplift metrics: AUUC and Qini (reference metric/uplift.cc:150-219,
"Optimal personalized treatment learning models" notation)."""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np


def auuc_qini(outcomes: np.ndarray, treatments: np.ndarray,
              predicted_uplift: np.ndarray,
              weights: Optional[np.ndarray] = None
              ) -> Tuple[float, float]:
    """Area under the uplift curve + Qini (auuc - max_lift/2), examples
    sorted by decreasing predicted uplift, ties grouped."""
    n = len(outcomes)
    if n == 0:
        return 0.0, 0.0
    w = np.ones(n) if weights is None else np.asarray(weights, np.float64)
    order = np.argsort(-predicted_uplift, kind="mergesort")
    y = np.asarray(outcomes, np.float64)[order]
    t = (np.asarray(treatments)[order] > 0.5)
    w = w[order]
    p = np.asarray(predicted_uplift, np.float64)[order]
    sum_t = (w * t).sum()
    sum_c = (w * ~t).sum()
    sum_w = w.sum()
    if sum_t == 0 or sum_c == 0:
        return 0.0, 0.0
    acc_t = np.cumsum(w * y * t)
    acc_c = np.cumsum(w * y * ~t)
    acc_w = np.cumsum(w)
    # group ties: evaluate the curve only at the last index of each
    # distinct predicted value
    last = np.ones(n, dtype=bool)
    last[:-1] = p[:-1] != p[1:]
    idx = np.nonzero(last)[0]
    net_lift = acc_t[idx] / sum_t - acc_c[idx] / sum_c
    aw = acc_w[idx] / sum_w
    prev_lift = np.concatenate([[0.0], net_lift[:-1]])
    prev_w = np.concatenate([[0.0], aw[:-1]])
    auuc = float(((aw - prev_w) * (net_lift + prev_lift) / 2.0).sum())
    qini = auuc - float(net_lift[-1]) / 2.0
    return auuc, qini
