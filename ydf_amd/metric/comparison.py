"""Statistical model comparison (capability analogue of the reference
metric/comparison.{h,cc}: pairwise significance tests between two models
on one evaluation dataset)."""
from __future__ import annotations

import dataclasses
from typing import Dict, Optional

import numpy as np


@dataclasses.dataclass
class ModelComparison:
    """Result of compare_models: per-metric values and the p-value of the
    null hypothesis "model_2 is not better than model_1"."""

    metrics_1: Dict[str, float]
    metrics_2: Dict[str, float]
    p_value: float
    test: str

    def __str__(self) -> str:
        lines = [f"test: {self.test} (p={self.p_value:.4g})"]
        for k in self.metrics_1:
            lines.append(f"{k}: {self.metrics_1[k]:.6g} -> "
                         f"{self.metrics_2.get(k, float('nan')):.6g}")
        return "\n".join(lines)


def _labels_for(model, data) -> np.ndarray:
    from ydf_amd.dataset.dataset import _to_column_dict
    from ydf_amd.dataset.dataspec import Semantic

    cols = _to_column_dict(data)
    lname = model.dataspec.label
    lspec = model.dataspec.label_column
    if lspec.semantic == Semantic.CATEGORICAL:
        lookup = {item: i for i, item in enumerate(lspec.vocab)}
        return np.fromiter(
            (lookup.get(s, 0) - 1 for s in cols[lname].astype(str)),
            dtype=np.float32, count=len(cols[lname]))
    return np.asarray(cols[lname], dtype=np.float32)


def compare_models(model_1, model_2, data,
                   device: Optional[str] = "cpu") -> ModelComparison:
    """Pairwise comparison (reference metric/comparison.h:PairwiseCompare):
    classification -> McNemar test on per-example correctness;
    regression -> paired t-test on squared errors. Small p-value means
    model_2 significantly beats model_1."""
    from scipy import stats

    from ydf_amd.dataset.dataspec import Task

    task = model_1.task()
    y = _labels_for(model_1, data)
    p1 = model_1.predict(data, device=device)
    p2 = model_2.predict(data, device=device)
    ev1 = model_1.evaluate(data, device=device)
    ev2 = model_2.evaluate(data, device=device)
    if task == Task.CLASSIFICATION:
        c1 = (p1 >= 0.5) == (y > 0.5) if p1.ndim == 1 \
            else p1.argmax(1) == y.astype(np.int64)
        c2 = (p2 >= 0.5) == (y > 0.5) if p2.ndim == 1 \
            else p2.argmax(1) == y.astype(np.int64)
        b = int((c1 & ~c2).sum())   # 1 right, 2 wrong
        c = int((~c1 & c2).sum())   # 2 right, 1 wrong
        if b + c == 0:
            p_value = 1.0
        else:
            # one-sided mid-p McNemar via binomial(c; b+c, 0.5)
            p_value = float(stats.binom.sf(c - 1, b + c, 0.5))
        m1 = {"accuracy": ev1.accuracy, "auc": ev1.auc or float("nan")}
        m2 = {"accuracy": ev2.accuracy, "auc": ev2.auc or float("nan")}
        return ModelComparison(m1, m2, p_value, "one-sided McNemar")
    # regression: paired one-sided t-test on squared errors
    e1 = (p1 - y) ** 2
    e2 = (p2 - y) ** 2
    t, p_two = stats.ttest_rel(e1, e2)
    p_value = float(p_two / 2.0) if t > 0 else float(1.0 - p_two / 2.0)
    m1 = {"rmse": ev1.rmse}
    m2 = {"rmse": ev2.rmse}
    return ModelComparison(m1, m2, p_value, "paired one-sided t-test")
