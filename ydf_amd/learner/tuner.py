"""Hyper-parameter tuning (capability analogue of the reference
HyperParameterOptimizerLearner + random-search optimizer,
learner/hyperparameters_optimizer/hyperparameters_optimizer.h:46,
optimizers/random.cc; Python surface of ydf.RandomSearchTuner)."""
from __future__ import annotations

import dataclasses
from typing import Dict, List, Optional

import numpy as np


@dataclasses.dataclass
class TrialLog:
    hyperparameters: Dict
    score: float  # higher is better (negative validation loss)


@dataclasses.dataclass
class OptimizerLogs:
    trials: List[TrialLog]

    @property
    def best_trial(self) -> TrialLog:
        return max(self.trials, key=lambda t: t.score)


class RandomSearchTuner:
    """Random search over declared hyper-parameter choices."""

    def __init__(self, num_trials: int = 50, automatic_search_space:
                 bool = False, seed: int = 1234):
        self.num_trials = num_trials
        self.seed = seed
        self._choices: Dict[str, list] = {}
        if automatic_search_space:
            # predefined space (reference PredefinedHyperParameterSpace,
            # gradient_boosted_trees.cc hyperparameter templates)
            self.choice("shrinkage", [0.02, 0.05, 0.1, 0.15])
            self.choice("max_depth", [3, 4, 6, 8])
            self.choice("subsample", [0.6, 0.8, 1.0])
            self.choice("l2_regularization", [0.0, 0.1, 1.0])
            self.choice("num_candidate_attributes_ratio", [0.5, 0.9, 1.0])
            self.choice("split_axis", ["AXIS_ALIGNED", "SPARSE_OBLIQUE"])

    def choice(self, name: str, values: list, merge: bool = False):
        if merge and name in self._choices:
            self._choices[name].extend(values)
        else:
            self._choices[name] = list(values)
        return self

    def sample(self, rng: np.random.RandomState) -> Dict:
        return {k: v[rng.randint(len(v))]
                for k, v in self._choices.items()}


class VizierTuner(RandomSearchTuner):
    """Placeholder keeping API parity: falls back to random search (the
    reference's Vizier backend is Google-internal)."""
