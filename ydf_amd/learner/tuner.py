"""Hyper-parameter tuning (capability analogue of the reference
HyperParameterOptimizerLearner + random-search optimizer,
learner/hyperparameters_optimizer/hyperparameters_optimizer.h:46,
optimizers/random.cc; Python surface of ydf.RandomSearchTuner)."""
from __future__ import annotations

import dataclasses
import enum
from typing import Dict, List, Optional

import numpy as np


class OptimizeMetric(enum.Enum):
    """Tuning objectives (mirrors PYDF tuner.OptimizeMetric)."""

    LOSS = "loss"
    ACCURACY = "accuracy"
    AUC = "auc"
    PR_AUC = "pr-auc"
    RMSE = "rmse"
    MAE = "mae"
    MSE = "mse"
    NDCG_5 = "ndcg@5"
    MRR_10 = "mrr@10"
    QINI = "qini"


# OptimizeMetric -> (Evaluation attribute, direction: +1 = maximize)
_METRIC_DIR = {
    OptimizeMetric.ACCURACY: ("accuracy", 1.0),
    OptimizeMetric.AUC: ("auc", 1.0),
    OptimizeMetric.PR_AUC: ("pr_auc", 1.0),
    OptimizeMetric.RMSE: ("rmse", -1.0),
    OptimizeMetric.MAE: ("mae", -1.0),
    OptimizeMetric.MSE: ("loss", -1.0),  # regression loss = mse
    OptimizeMetric.NDCG_5: ("ndcg", 1.0),
    OptimizeMetric.MRR_10: ("mrr", 1.0),
    OptimizeMetric.QINI: ("qini", 1.0),
}


def trial_score(model, eval_data, optimize_metric) -> float:
    """Higher-is-better trial objective. LOSS (default) reads the
    model's validation loss; other metrics evaluate on `eval_data`
    (pass the validation set to `train(valid=...)` for unbiased model
    selection — falling back to the training data is logged)."""
    if optimize_metric in (None, OptimizeMetric.LOSS):
        vloss = None
        if model.training_logs:
            vloss = model.training_logs[-1].get("valid_loss")
        return -(vloss if vloss is not None else float("inf"))
    name, sign = _METRIC_DIR[OptimizeMetric(optimize_metric)]
    v = getattr(model.evaluate(eval_data), name)
    return sign * float(v) if v is not None and np.isfinite(v) \
        else float("-inf")


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
    """Random search over declared hyper-parameter choices.

    parallel_trials > 1 evaluates trials concurrently, one process per
    trial slot (the MI355X mapping of the reference's distributed HPO:
    hyperparameters_optimizer.h:46 + learner/generic_worker/ train one
    model per remote worker). On a multi-GPU box each slot pins a GPU
    round-robin; on CPU the slots are plain processes. Trial sampling
    happens up front from the seed, so results are independent of the
    execution order."""

    def __init__(self, num_trials: int = 50, automatic_search_space:
                 bool = False, seed: int = 1234,
                 parallel_trials: int = 1,
                 optimize_metric=None):
        self.num_trials = num_trials
        self.seed = seed
        self.parallel_trials = parallel_trials
        self.optimize_metric = optimize_metric
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


def _trial_worker(payload):
    """One tuning trial in a worker process (spawn-safe top-level).

    payload: (pickled learner copy, hp_override, data, slot_idx).
    Returns (score, serialized model bytes)."""
    import torch

    from ydf_amd.model.model_lib import serialize_model

    learner, hp, data, valid, metric, slot = payload
    if torch.cuda.is_available():
        # one trial per GPU, round-robin over slots
        learner.device = f"cuda:{slot % torch.cuda.device_count()}"
    learner.hyperparameters = dict(learner.hyperparameters)
    learner.hyperparameters.update(
        {k: v for k, v in hp.items() if k in learner.hyperparameters})
    if learner.hyperparameters.get("validation_ratio", 0) == 0:
        learner.hyperparameters["validation_ratio"] = 0.1
    model = learner.train(data)
    score = trial_score(model, valid if valid is not None else data,
                        metric)
    return score, serialize_model(model)


def run_parallel_trials(learner, tuner, data, valid=None):
    """Evaluates all trials in a process pool (parallel_trials slots).

    Falls back to sequential trials when worker processes cannot be
    spawned (multiprocessing "spawn" must re-import __main__, which an
    interactive session / heredoc script does not have)."""
    import multiprocessing as mp
    import sys

    from ydf_amd.dataset.dataset import _to_column_dict
    from ydf_amd.model.model_lib import deserialize_model

    import copy

    main_file = getattr(sys.modules.get("__main__"), "__file__", None)
    if not main_file or not __import__("os").path.exists(main_file):
        from ydf_amd.utils.log import info

        info("parallel_trials: __main__ is not an importable file "
             "(interactive session); running trials sequentially")
        seq = copy.copy(tuner)
        seq.parallel_trials = 1
        seq_learner = copy.copy(learner)
        seq_learner.tuner = seq
        return seq_learner.train(data)

    rng = np.random.RandomState(tuner.seed)
    samples = [tuner.sample(rng) for _ in range(tuner.num_trials)]
    cols = _to_column_dict(data)
    vcols = _to_column_dict(valid) if valid is not None else None
    metric = getattr(tuner, "optimize_metric", None)
    ln = copy.copy(learner)
    ln.tuner = None  # the worker runs a plain (non-tuning) train
    payloads = [(ln, hp, cols, vcols, metric, i)
                for i, hp in enumerate(samples)]
    ctx = mp.get_context("spawn")
    with ctx.Pool(processes=tuner.parallel_trials) as pool:
        results = pool.map(_trial_worker, payloads)
    logs = [TrialLog(hyperparameters=hp, score=score)
            for hp, (score, _) in zip(samples, results)]
    best_i = int(np.argmax([r[0] for r in results]))
    model = deserialize_model(results[best_i][1])
    model.tuner_logs = OptimizerLogs(trials=logs)
    return model


# PYDF exposes an AbstractTuner base; RandomSearchTuner is the only
# optimizer backend here (the reference's other backend, Vizier, is
# Google-internal — VizierTuner falls back to random search).
AbstractTuner = RandomSearchTuner
