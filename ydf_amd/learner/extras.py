"""Smaller learner utilities: multitasker + backward feature selection.

Capability analogues of the reference's learner/multitasker/ (trains N
learners over a shared dataset) and PYDF's BackwardSelectionFeatureSelector
(ydf/learner/feature_selector.py).
"""
from __future__ import annotations

import copy
import dataclasses
from typing import Dict, List, Optional

import numpy as np


@dataclasses.dataclass
class MultitaskItem:
    label: str
    task: object  # ydf.Task


class MultitaskerLearner:
    """Trains one model per task on the same dataset (reference
    learner/multitasker/multitasker.h)."""

    def __init__(self, tasks: List[MultitaskItem], learner_factory=None,
                 **learner_kwargs):
        import ydf_amd as ydf

        self.tasks = tasks
        self.factory = learner_factory or ydf.GradientBoostedTreesLearner
        self.kwargs = learner_kwargs

    def train(self, data) -> "MultitaskerModel":
        models = {}
        for item in self.tasks:
            learner = self.factory(label=item.label, task=item.task,
                                   **self.kwargs)
            models[item.label] = learner.train(data)
        return MultitaskerModel(models)


class MultitaskerModel:
    def __init__(self, models: Dict[str, object]):
        self.models = models

    def predict(self, data) -> Dict[str, np.ndarray]:
        return {k: m.predict(data) for k, m in self.models.items()}

    def evaluate(self, data) -> Dict[str, object]:
        return {k: m.evaluate(data) for k, m in self.models.items()}


@dataclasses.dataclass
class FeatureSelectorLogs:
    iterations: List[dict]
    selected_features: List[str]


class BackwardSelectionFeatureSelector:
    """Backward feature elimination driven by validation quality (mirrors
    ydf.BackwardSelectionFeatureSelector)."""

    def __init__(self, removal_count: int = 1, min_features: int = 1,
                 objective_metric: str = "loss"):
        self.removal_count = removal_count
        self.min_features = min_features
        self.objective_metric = objective_metric

    def run(self, learner, data, valid) -> "FeatureSelectorLogs":
        """Iteratively removes the least important features while validation
        quality does not degrade; returns the selection trace."""
        from ydf_amd.dataset.dataset import _to_column_dict

        cols = _to_column_dict(data)
        features = [c for c in cols if c != learner.label]
        best_score = None
        logs = []
        best_features = list(features)
        while len(features) >= max(self.min_features, 1):
            lrn = copy.copy(learner)
            lrn.features = list(features)
            model = lrn.train(data)
            ev = model.evaluate(valid)
            score = getattr(ev, self.objective_metric, None)
            if score is None:
                score = ev.loss
            lower_better = self.objective_metric in ("loss", "rmse", "mae")
            better = (best_score is None
                      or (score <= best_score if lower_better
                          else score >= best_score))
            logs.append({"features": list(features), "score": float(score)})
            if better:
                best_score = score
                best_features = list(features)
            if len(features) <= self.min_features:
                break
            vi = model.variable_importances()
            ranked = vi.get("SUM_SCORE") or vi.get("NUM_NODES") or []
            ranked_names = [n for _, n in ranked]
            unused = [f for f in features if f not in ranked_names]
            drop = (unused + list(reversed(ranked_names)))[
                : self.removal_count]
            if not drop:
                break
            features = [f for f in features if f not in drop]
        return FeatureSelectorLogs(iterations=logs,
                                   selected_features=best_features)
