"""User-provided secondary metrics (PYDF custom_metric.py analogue).

The metric's evaluation_func receives (labels, predictions-without-
activation, weights) and returns a float; values are recorded per
validation interval in the model's training logs.
"""
from __future__ import annotations

import dataclasses
from typing import Callable


@dataclasses.dataclass(frozen=True)
class AbstractCustomMetric:
    name: str
    evaluation_func: Callable = None

    def __post_init__(self):
        if not isinstance(self.name, str):
            raise ValueError(
                f"custom metric name must be a string, got "
                f"{type(self.name)}")


@dataclasses.dataclass(frozen=True)
class RegressionMetric(AbstractCustomMetric):
    """Secondary metric for regression: f(labels, preds, weights)."""


@dataclasses.dataclass(frozen=True)
class BinaryClassificationMetric(AbstractCustomMetric):
    """Secondary metric for binary classification: predictions are
    margins (apply a sigmoid for probabilities)."""


@dataclasses.dataclass(frozen=True)
class MultiClassificationMetric(AbstractCustomMetric):
    """Secondary metric for multi-class: predictions are per-class
    margins [C, N]."""
