"""Name -> factory registries (reference utils/registration.h
REGISTRATION_REGISTER_CLASS / ClassPool): learners, models and serving
engines are discoverable by name."""
from __future__ import annotations

from typing import Callable, Dict


class Registry:
    def __init__(self, kind: str):
        self.kind = kind
        self._items: Dict[str, Callable] = {}

    def register(self, name: str, factory: Callable = None):
        if factory is not None:
            self._items[name] = factory
            return factory

        def deco(f):
            self._items[name] = f
            return f
        return deco

    def get(self, name: str) -> Callable:
        if name not in self._items:
            raise KeyError(
                f"no {self.kind} registered under {name!r}; "
                f"available: {sorted(self._items)}")
        return self._items[name]

    def names(self):
        return sorted(self._items)


learner_registry = Registry("learner")
model_registry = Registry("model")


def _bootstrap():
    from ydf_amd.deep import (MultiLayerPerceptronLearner,
                              TabularTransformerLearner)
    from ydf_amd.learner.specialized_learners import (
        CartLearner, GradientBoostedTreesLearner, IsolationForestLearner,
        RandomForestLearner)
    from ydf_amd.model.specialized import MODEL_CLASSES

    learner_registry.register("GRADIENT_BOOSTED_TREES",
                              GradientBoostedTreesLearner)
    learner_registry.register("RANDOM_FOREST", RandomForestLearner)
    learner_registry.register("CART", CartLearner)
    learner_registry.register("ISOLATION_FOREST", IsolationForestLearner)
    learner_registry.register("MULTI_LAYER_PERCEPTRON",
                              MultiLayerPerceptronLearner)
    learner_registry.register("TABULAR_TRANSFORMER",
                              TabularTransformerLearner)
    for name, cls in MODEL_CLASSES.items():
        model_registry.register(name, cls)


def get_learner(name: str):
    """ydf.get_learner analogue (reference GetLearner registry entry)."""
    if not learner_registry._items:
        _bootstrap()
    return learner_registry.get(name)
