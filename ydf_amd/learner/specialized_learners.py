"""Concrete learners: GradientBoostedTrees, RandomForest, Cart,
IsolationForest.

API mirrors PYDF (port/python/ydf/learner/specialized_learners_pre_generated
.py signatures; hyperparameter names + defaults from SURVEY.md Appendix A,
i.e. the reference proto defaults). The training hot paths run through the
HIP/gfx950 kernels in ydf_amd/ops on GPU or their C++ CPU twins.
"""
from __future__ import annotations

import math
from typing import Optional, Sequence, Union

import numpy as np
import torch

from ydf_amd.dataset.dataset import VerticalDataset
from ydf_amd.dataset.dataspec import Column, Task
from ydf_amd.learner import trainer as trainer_lib
from ydf_amd.learner.generic_learner import GenericLearner
from ydf_amd.model.forest import FlatForest, build_flat_forest
from ydf_amd.model.specialized import (GradientBoostedTreesModel,
                                       IsolationForestModel,
                                       RandomForestModel)
from ydf_amd.utils.log import info

# Predefined hyperparameter templates (reference
# GetPredefinedHyperParameterTemplates: gradient_boosted_trees.cc /
# random_forest.cc). growing_strategy=BEST_FIRST_GLOBAL entries are
# dropped: trees grow level-wise here (documented deviation).
HYPERPARAMETER_TEMPLATES = {
    "GBT": {
        "better_default@v1": {},
        "benchmark_rank1@v1": {
            "split_axis": "SPARSE_OBLIQUE",
            "sparse_oblique_normalization": "MIN_MAX",
            "sparse_oblique_num_projections_exponent": 1.0,
        },
    },
    "RF": {
        "better_default@v1": {"winner_take_all": True},
    },
}


def _apply_template(hp: dict, family: str, name) -> None:
    if not name:
        return
    table = HYPERPARAMETER_TEMPLATES[family]
    if name not in table and f"{name}@v1" in table:
        name = f"{name}@v1"
    if name not in table:
        raise ValueError(
            f"unknown hyperparameter template {name!r}; available: "
            f"{sorted(table)}")
    hp.update(table[name])
    info(f"applied hyperparameter template {name}")


class GradientBoostedTreesLearner(GenericLearner):
    """GBT learner (reference learner/gradient_boosted_trees/; Python
    surface of ydf.GradientBoostedTreesLearner).

    Notes vs the reference: numerical features are always trained on the
    256-bin quantile-discretized representation (the reference's
    force_numerical_discretization path); split gain always uses the
    second-order formulation (reference use_hessian_gain)."""

    def __init__(self, label: str, task: Task = Task.CLASSIFICATION,
                 features: Optional[Sequence[Union[str, Column]]] = None,
                 ranking_group: Optional[str] = None,
                 label_event_observed: Optional[str] = None,
                 label_entry_age: Optional[str] = None,
                 ndcg_truncation: int = 5,
                 num_trees: int = 300, max_depth: int = 6,
                 growing_strategy: str = "LOCAL",
                 max_num_nodes: int = 31,
                 shrinkage: float = 0.1, subsample: float = 1.0,
                 sampling_method: str = "RANDOM",
                 goss_alpha: float = 0.2, goss_beta: float = 0.1,
                 selective_gradient_boosting_ratio: float = 0.01,
                 min_examples: int = 5, l2_regularization: float = 0.0,
                 l1_regularization: float = 0.0,
                 min_sum_hessian_in_leaf: float = 1e-3,
                 validation_ratio: float = 0.1,
                 validation_interval_in_trees: int = 1,
                 lambda_loss: float = 1.0,
                 total_max_num_nodes: int = -1,
                 early_stopping: str = "LOSS_INCREASE",
                 early_stopping_num_trees_look_ahead: int = 30,
                 early_stopping_initial_iteration: int = 10,
                 num_candidate_attributes_ratio: float = -1.0,
                 num_candidate_attributes: int = -1,
                 in_split_min_examples_check: bool = True,
                 sorting_strategy: str = "PRESORT",
                 keep_non_leaf_label_distribution: bool = True,
                 honest: bool = False,
                 honest_ratio_leaf_examples: float = 0.5,
                 honest_fixed_separation: bool = False,
                 mhld_oblique_sample_attributes=None,
                 compute_permutation_variable_importance: bool = False,
                 uplift_split_score: str = "KULLBACK_LEIBLER",
                 uplift_min_examples_in_treatment: int = 5,
                 use_hessian_gain: bool = True,
                 apply_link_function: bool = True,
                 l2_categorical_regularization: float = 1.0,
                 focal_loss_alpha: float = 0.5,
                 focal_loss_gamma: float = 2.0,
                 forest_extraction: str = "MART",
                 dart_dropout: float = 0.01,
                 adapt_subsample_for_maximum_training_duration:
                 bool = False,
                 split_axis: str = "AXIS_ALIGNED",
                 sparse_oblique_num_projections_exponent: float = 2.0,
                 sparse_oblique_max_num_projections: int = 6000,
                 sparse_oblique_projection_density_factor: float = 2.0,
                 sparse_oblique_normalization: str = "NONE",
                 sparse_oblique_weights: str = "BINARY",
                 sparse_oblique_max_num_features: int = -1,
                 mhld_oblique_max_num_attributes: int = 4,
                 loss: str = "DEFAULT",
                 discretize_numerical_columns: bool = True,
                 working_dir: Optional[str] = None,
                 resume_training: bool = False,
                 resume_training_snapshot_interval_seconds: float = 1800.0,
                 maximum_training_duration_seconds: float = -1.0,
                 hyperparameter_template: Optional[str] = None,
                 custom_metrics=None,
                 random_seed: int = 123456, **kwargs):
        super().__init__(label=label, task=task, features=features,
                         random_seed=random_seed, **kwargs)
        self.custom_metrics = custom_metrics
        self.ranking_group = ranking_group
        self.label_event_observed = label_event_observed
        self.label_entry_age = label_entry_age
        self.ndcg_truncation = ndcg_truncation
        # reference DEFAULT trains exact numerical splits; the MI355X
        # hot path always bins (discretize=True, documented deviation).
        # False selects the CPU exact-split oracle (learner/exact.py).
        self.discretize_numerical_columns = discretize_numerical_columns
        self.hyperparameters = dict(
            num_trees=num_trees, max_depth=max_depth,
            growing_strategy=growing_strategy, max_num_nodes=max_num_nodes,
            shrinkage=shrinkage,
            subsample=subsample, sampling_method=sampling_method,
            goss_alpha=goss_alpha, goss_beta=goss_beta,
            selective_gradient_boosting_ratio=(
                selective_gradient_boosting_ratio),
            min_examples=min_examples,
            l2_regularization=l2_regularization,
            l1_regularization=l1_regularization,
            min_sum_hessian_in_leaf=min_sum_hessian_in_leaf,
            validation_ratio=validation_ratio,
            validation_interval_in_trees=validation_interval_in_trees,
            lambda_loss=lambda_loss,
            total_max_num_nodes=total_max_num_nodes,
            early_stopping=early_stopping,
            early_stopping_num_trees_look_ahead=(
                early_stopping_num_trees_look_ahead),
            early_stopping_initial_iteration=early_stopping_initial_iteration,
            num_candidate_attributes_ratio=num_candidate_attributes_ratio,
            num_candidate_attributes=num_candidate_attributes,
            in_split_min_examples_check=in_split_min_examples_check,
            sorting_strategy=sorting_strategy,
            keep_non_leaf_label_distribution=(
                keep_non_leaf_label_distribution),
            honest=honest,
            honest_ratio_leaf_examples=honest_ratio_leaf_examples,
            honest_fixed_separation=honest_fixed_separation,
            mhld_oblique_sample_attributes=mhld_oblique_sample_attributes,
            compute_permutation_variable_importance=(
                compute_permutation_variable_importance),
            uplift_split_score=uplift_split_score,
            uplift_min_examples_in_treatment=(
                uplift_min_examples_in_treatment),
            use_hessian_gain=use_hessian_gain,
            apply_link_function=apply_link_function,
            l2_categorical_regularization=l2_categorical_regularization,
            focal_loss_alpha=focal_loss_alpha,
            focal_loss_gamma=focal_loss_gamma,
            forest_extraction=forest_extraction,
            dart_dropout=dart_dropout,
            adapt_subsample_for_maximum_training_duration=(
                adapt_subsample_for_maximum_training_duration),
            split_axis=split_axis,
            sparse_oblique_num_projections_exponent=(
                sparse_oblique_num_projections_exponent),
            sparse_oblique_max_num_projections=(
                sparse_oblique_max_num_projections),
            sparse_oblique_projection_density_factor=(
                sparse_oblique_projection_density_factor),
            sparse_oblique_normalization=sparse_oblique_normalization,
            sparse_oblique_weights=sparse_oblique_weights,
            sparse_oblique_max_num_features=(
                sparse_oblique_max_num_features),
            mhld_oblique_max_num_attributes=(
                mhld_oblique_max_num_attributes),
            loss=loss,
            working_dir=working_dir, resume_training=resume_training,
            resume_training_snapshot_interval_seconds=(
                resume_training_snapshot_interval_seconds),
            maximum_training_duration_seconds=(
                maximum_training_duration_seconds),
        )
        _apply_template(self.hyperparameters, "GBT",
                        hyperparameter_template)

    def _train_exact(self, data) -> GradientBoostedTreesModel:
        """Exact (non-binned) numerical split training on CPU — the
        reference's DEFAULT split semantics (splitter_scanner.h:1290,
        preprocessing.h:106). Serves as the correctness oracle that
        bounds the quality effect of the 256-bin hot path."""
        from ydf_amd.learner.exact import (exact_trees_to_forest,
                                           train_gbt_exact)

        hp = self.hyperparameters
        if self._task not in (Task.CLASSIFICATION, Task.REGRESSION):
            raise NotImplementedError(
                "discretize_numerical_columns=False supports "
                "classification and regression")
        if hp.get("split_axis", "AXIS_ALIGNED") != "AXIS_ALIGNED" \
                or hp.get("forest_extraction") == "DART" \
                or hp.get("sampling_method") not in (None, "RANDOM") \
                or hp.get("subsample", 1.0) < 1.0:
            raise NotImplementedError(
                "exact-split mode supports axis-aligned MART without "
                "subsampling")
        device = torch.device("cpu")
        ds, bins, labels, bnd, cat_flags, weights, mono = self._prepare(
            data, device)
        if labels is None:
            raise ValueError(f"label column {self.label!r} missing")
        if weights is not None or mono is not None:
            raise NotImplementedError(
                "exact-split mode: weights/monotonic not supported")
        classes = self._label_classes(ds)
        if classes and len(classes) != 2:
            raise NotImplementedError(
                "exact-split mode supports binary classification")
        loss_named = hp.get("loss")
        if isinstance(loss_named, str) and loss_named not in (
                "DEFAULT", "SQUARED_ERROR", "BINOMIAL_LOG_LIKELIHOOD"):
            raise NotImplementedError(
                f"exact-split mode does not support loss {loss_named}")
        X = np.ascontiguousarray(ds.X)
        if np.isnan(X).any():
            raise NotImplementedError(
                "exact-split mode: impute missing values first "
                "(GLOBAL_IMPUTATION runs at encode time by default)")
        y = labels.cpu().numpy().astype(np.float64)
        loss = 1 if self._task == Task.CLASSIFICATION else 2
        cf = cat_flags.cpu().numpy().astype(bool) \
            if cat_flags is not None else None
        info(f"exact-split training ({hp['num_trees']} trees, no "
             f"validation/early-stopping in this mode)")
        trees, init = train_gbt_exact(
            X, y, cf, loss, hp["num_trees"], hp["shrinkage"],
            hp["max_depth"], hp["min_examples"],
            hp["min_sum_hessian_in_leaf"],
            hp.get("l1_regularization", 0.0), hp["l2_regularization"],
            cat_smooth=hp["l2_categorical_regularization"])
        forest = exact_trees_to_forest(trees, hp["shrinkage"])
        activation = "identity"
        if self._task == Task.CLASSIFICATION and \
                hp["apply_link_function"]:
            activation = "sigmoid"
        return GradientBoostedTreesModel(
            forest=forest, dataspec=ds.dataspec, task=self._task,
            label_classes=classes, init_predictions=[init],
            num_trees_per_iter=1, activation=activation,
            metadata={"loss": loss, "exact_splits": True,
                      "missing_value_policy": "GLOBAL_IMPUTATION"})

    def _train_streaming(self, cache) -> GradientBoostedTreesModel:
        """Out-of-core GBT over an on-disk binned DatasetCache
        (learner/streaming.py; reference ShardedSamplingTrain,
        gradient_boosted_trees.cc:655). Chunks stream through the
        device; the dataset can exceed HBM."""
        from ydf_amd.dataset.dataspec import Semantic
        from ydf_amd.learner.streaming import train_gbt_streaming
        from ydf_amd.model.forest import padded_boundaries

        hp = self.hyperparameters
        if self._task not in (Task.CLASSIFICATION, Task.REGRESSION):
            raise NotImplementedError(
                "streaming training supports classification/regression")
        if hp.get("subsample", 1.0) < 1.0 or \
                hp.get("split_axis", "AXIS_ALIGNED") != "AXIS_ALIGNED" \
                or hp.get("forest_extraction") == "DART":
            raise NotImplementedError(
                "streaming training: axis-aligned MART without "
                "subsampling")
        device = self._resolve_device()
        lspec = cache.dataspec.label_column
        classes = None
        if self._task == Task.CLASSIFICATION:
            if lspec.semantic != Semantic.CATEGORICAL:
                raise ValueError("classification label must be "
                                 "categorical")
            classes = list(lspec.vocab[1:])
            if len(classes) != 2:
                raise NotImplementedError(
                    "streaming training supports binary classification")
        loss = trainer_lib.LOSS_BINOMIAL \
            if self._task == Task.CLASSIFICATION \
            else trainer_lib.LOSS_SQUARED_ERROR
        cfg = trainer_lib.TrainerConfig(
            loss=loss, num_trees=hp["num_trees"],
            max_depth=hp["max_depth"], shrinkage=hp["shrinkage"],
            lambda_l2=hp["l2_regularization"],
            lambda_l1=hp.get("l1_regularization", 0.0),
            min_examples=hp["min_examples"],
            min_hessian=hp["min_sum_hessian_in_leaf"],
            cat_smooth=hp["l2_categorical_regularization"],
            seed=self.random_seed)
        trees, init = train_gbt_streaming(cache, cfg, device, log=info)
        bnd = padded_boundaries(cache.dataspec.feature_columns)
        cat_feats = np.asarray(
            [c.semantic == Semantic.CATEGORICAL
             for c in cache.dataspec.feature_columns], dtype=bool)
        flat = build_flat_forest(trees, bnd,
                                 leaf_scale=hp["shrinkage"],
                                 cat_feats=cat_feats)
        activation = "identity"
        if self._task == Task.CLASSIFICATION and \
                hp["apply_link_function"]:
            activation = "sigmoid"
        return GradientBoostedTreesModel(
            forest=flat, dataspec=cache.dataspec, task=self._task,
            label_classes=classes, init_predictions=[init],
            num_trees_per_iter=1, activation=activation,
            metadata={"loss": int(loss), "streaming_cache": True,
                      "missing_value_policy": "GLOBAL_IMPUTATION"})

    def _train_mhld(self, data) -> GradientBoostedTreesModel:
        """MHLD oblique splits (reference oblique.h:33 + oblique.cc
        FindBestConditionMHLDObliqueTemplate): greedy LDA-projection
        subsets per node. Runs on the CPU exact path (per-node LDA
        solves; classification only, like the reference)."""
        from ydf_amd.learner.exact import (MhldSplitter,
                                           mhld_trees_to_forest)

        hp = self.hyperparameters
        if self._task != Task.CLASSIFICATION:
            raise NotImplementedError(
                "MHLD_OBLIQUE is classification-only "
                "(reference oblique.cc:690); use SPARSE_OBLIQUE")
        ds, bins, labels, bnd, cat_flags, weights, mono = self._prepare(
            data, torch.device("cpu"))
        classes = self._label_classes(ds)
        if classes is None or len(classes) != 2:
            raise NotImplementedError(
                "MHLD_OBLIQUE supports binary classification")
        if weights is not None or mono is not None:
            raise NotImplementedError("MHLD_OBLIQUE: weights/monotonic "
                                      "not supported")
        X = np.ascontiguousarray(ds.X)
        y = labels.cpu().numpy().astype(np.float64)
        cf = cat_flags.cpu().numpy().astype(bool) \
            if cat_flags is not None else None
        p = np.clip(y.mean(), 1e-6, 1 - 1e-6)
        init = float(np.log(p / (1 - p)))
        preds = np.full(len(y), init)
        sp = MhldSplitter(
            X, cf, y01=y > 0.5,
            max_attributes=hp.get("mhld_oblique_max_num_attributes", 4),
            max_depth=hp["max_depth"], min_examples=hp["min_examples"],
            min_hessian=hp["min_sum_hessian_in_leaf"],
            lambda_l2=hp["l2_regularization"])
        trees = []
        for _ in range(hp["num_trees"]):
            pr = 1.0 / (1.0 + np.exp(-preds))
            g = pr - y
            h = np.maximum(pr * (1.0 - pr), 1e-16)
            t = sp.grow_tree(g, h)
            trees.append(t)
            preds += hp["shrinkage"] * t.thr[sp._last_node_of_row]
        flat = mhld_trees_to_forest(trees, hp["shrinkage"])
        return GradientBoostedTreesModel(
            forest=flat, dataspec=ds.dataspec, task=self._task,
            label_classes=classes, init_predictions=[init],
            num_trees_per_iter=1,
            activation="sigmoid" if hp["apply_link_function"]
            else "identity",
            metadata={"loss": 1, "mhld_oblique": True,
                      "missing_value_policy": "GLOBAL_IMPUTATION"})

    def train(self, data, valid=None, verbose=None
              ) -> GradientBoostedTreesModel:
        from ydf_amd.dataset.cache import DatasetCache

        if isinstance(data, DatasetCache):
            return self._train_streaming(data)
        if getattr(self, "feature_selector", None) is not None:
            return self._train_with_feature_selection(data, valid)
        if self.tuner is not None:
            return self._train_with_tuner(data, valid=valid)
        if self.hyperparameters.get("split_axis") == "MHLD_OBLIQUE":
            return self._train_mhld(data)
        if not self.discretize_numerical_columns:
            return self._train_exact(data)
        hp = self.hyperparameters
        device = self._resolve_device()
        group_ids = None
        surv_events = surv_entry = None
        if self._task == Task.SURVIVAL_ANALYSIS:
            # Cox proportional hazards (reference loss_imp_cox.cc):
            # label = time, label_event_observed = indicator column
            if self.label_event_observed is None:
                raise ValueError("task=SURVIVAL_ANALYSIS needs "
                                 "label_event_observed=")
            from ydf_amd.dataset.dataset import _to_column_dict

            cols = _to_column_dict(data)
            surv_events = np.asarray(
                cols.pop(self.label_event_observed)).astype(bool)
            if self.label_entry_age is not None:
                surv_entry = np.asarray(
                    cols.pop(self.label_entry_age), dtype=np.float64)
            data = cols
            if self.features is None:
                self.features = [k for k in cols if k != self.label]
        if self._task == Task.RANKING:
            if self.ranking_group is None:
                raise ValueError("task=RANKING needs ranking_group=")
            # sort rows by group so queries are contiguous, drop the group
            # column from the features
            from ydf_amd.dataset.dataset import _to_column_dict

            cols = _to_column_dict(data)
            gvals = np.asarray(cols[self.ranking_group])
            order = np.argsort(gvals, kind="stable")
            data = {k: np.asarray(v)[order] for k, v in cols.items()}
            group_ids = data.pop(self.ranking_group)
            if self.features is None:
                self.features = [k for k in data if k != self.label]
        ds, bins, labels, bnd, cat_flags, weights, mono = self._prepare(
            data, device)
        if labels is None:
            raise ValueError(f"label column {self.label!r} missing")
        obl = self._oblique_cfg(bins.shape[0], cat_flags)
        raw_t = valid_raw_t = None
        if obl:
            raw_t = torch.from_numpy(
                np.ascontiguousarray(ds.X)).to(device)
        classes = self._label_classes(ds)
        n_classes = len(classes) if classes else 2
        custom_loss = None
        if hp.get("loss") is not None and hp.get("loss") != "DEFAULT" \
                and not isinstance(hp.get("loss"), str):
            custom_loss = hp["loss"]
        if self._task == Task.CLASSIFICATION:
            named = hp.get("loss") if isinstance(hp.get("loss"), str) \
                else "DEFAULT"
            if named == "BINARY_FOCAL_LOSS":
                if n_classes > 2:
                    raise ValueError(
                        "BINARY_FOCAL_LOSS needs a binary label")
                loss = trainer_lib.LOSS_FOCAL
            else:
                loss = (trainer_lib.LOSS_MULTINOMIAL if n_classes > 2
                        else trainer_lib.LOSS_BINOMIAL)
        elif self._task == Task.REGRESSION:
            named = hp.get("loss") if isinstance(hp.get("loss"), str) \
                else "DEFAULT"
            loss = {"DEFAULT": trainer_lib.LOSS_SQUARED_ERROR,
                    "SQUARED_ERROR": trainer_lib.LOSS_SQUARED_ERROR,
                    "POISSON": trainer_lib.LOSS_POISSON,
                    "MEAN_AVERAGE_ERROR": trainer_lib.LOSS_MAE,
                    }.get(named, trainer_lib.LOSS_SQUARED_ERROR)
        elif self._task == Task.RANKING:
            named = hp.get("loss") if isinstance(hp.get("loss"), str) \
                else "DEFAULT"
            loss = (trainer_lib.LOSS_XE_NDCG
                    if named in ("XE_NDCG", "XE_NDCG_MART")
                    else trainer_lib.LOSS_LAMBDA_MART_NDCG)
        elif self._task == Task.SURVIVAL_ANALYSIS:
            loss = trainer_lib.LOSS_COX
        else:
            raise NotImplementedError(
                f"GBT task {self._task} not yet supported")

        # validation split (reference validation_set_ratio,
        # gradient_boosted_trees.cc:1243)
        valid_bins = valid_labels = None
        ranking = valid_ranking = None
        vr = hp["validation_ratio"]
        if self._task == Task.RANKING:
            from ydf_amd.learner.ranking import RankingLambdas

            # group-aware validation split: whole queries go to validation
            if vr > 0.0 and hp["early_stopping"] != "NONE":
                uniq, starts = np.unique(group_ids, return_index=True)
                rng = np.random.RandomState(self.random_seed)
                vmask_g = rng.random_sample(len(uniq)) < vr
                gidx = np.searchsorted(np.sort(starts),
                                       np.arange(len(group_ids)),
                                       side="right") - 1
                vmask = vmask_g[gidx]
                vi = torch.from_numpy(np.nonzero(vmask)[0]).to(device)
                ti = torch.from_numpy(np.nonzero(~vmask)[0]).to(device)
                valid_bins = bins[:, vi].contiguous()
                valid_labels = labels[vi].contiguous()
                bins = bins[:, ti].contiguous()
                labels = labels[ti].contiguous()
                if raw_t is not None:
                    valid_raw_t = raw_t[:, vi].contiguous()
                    raw_t = raw_t[:, ti].contiguous()
                valid_ranking = RankingLambdas(
                    group_ids[vmask], group_ids[vmask] * 0
                    + valid_labels.cpu().numpy(), device,
                    truncation=self.ndcg_truncation)
                group_ids = group_ids[~vmask]
            ranking = RankingLambdas(group_ids, labels.cpu().numpy(),
                                     device,
                                     truncation=self.ndcg_truncation,
                                     sigma=hp.get("lambda_loss", 1.0))
        elif valid is not None:
            if self._task == Task.SURVIVAL_ANALYSIS:
                raise NotImplementedError(
                    "survival: use validation_ratio (user valid= needs "
                    "event columns threaded through; ROADMAP)")
            valid_bins, valid_labels, vds = self._prepare_valid(
                valid, ds, device)
            if raw_t is not None:
                valid_raw_t = torch.from_numpy(
                    np.ascontiguousarray(vds.X)).to(device)
        elif vr > 0.0 and hp["early_stopping"] != "NONE":
            N = bins.shape[1]
            rng = np.random.RandomState(self.random_seed)
            perm = rng.permutation(N)
            n_valid = max(1, int(N * vr)) if N > 10 else 0
            if n_valid:
                vi = torch.from_numpy(perm[:n_valid].copy()).to(device)
                ti = torch.from_numpy(perm[n_valid:].copy()).to(device)
                valid_bins = bins[:, vi].contiguous()
                valid_labels = labels[vi].contiguous()
                bins = bins[:, ti].contiguous()
                labels = labels[ti].contiguous()
                if weights is not None:
                    weights = weights[ti].contiguous()
                if raw_t is not None:
                    valid_raw_t = raw_t[:, vi].contiguous()
                    raw_t = raw_t[:, ti].contiguous()
                if surv_events is not None:
                    sv, st = perm[:n_valid], perm[n_valid:]
                    surv_events_v = surv_events[sv]
                    surv_events = surv_events[st]
                    if surv_entry is not None:
                        surv_entry_v = surv_entry[sv]
                        surv_entry = surv_entry[st]

        cox = valid_cox = None
        if self._task == Task.SURVIVAL_ANALYSIS:
            from ydf_amd.learner.survival import CoxData

            cox = CoxData(labels.cpu().numpy(), surv_events, device,
                          surv_entry)
            if valid_bins is not None:
                valid_cox = CoxData(
                    valid_labels.cpu().numpy(), surv_events_v, device,
                    surv_entry_v if surv_entry is not None else None)

        F = bins.shape[0]
        if not hp.get("in_split_min_examples_check", True):
            raise NotImplementedError(
                "in_split_min_examples_check=False is not implemented "
                "(min_examples is always enforced at split search)")
        if hp.get("sorting_strategy", "PRESORT") not in (
                "IN_NODE", "PRESORT", "FORCE_PRESORT", "AUTO", "LAYER"):
            raise ValueError(
                f"unknown sorting_strategy {hp['sorting_strategy']!r}")
        # sorting_strategy is a CPU-splitter performance hint in the
        # reference; the 256-bin GPU histogram path has no per-node sort
        if hp.get("mhld_oblique_sample_attributes"):
            raise NotImplementedError(
                "mhld_oblique_sample_attributes is not implemented")
        if hp.get("honest") and hp.get(
                "growing_strategy") == "BEST_FIRST_GLOBAL":
            raise NotImplementedError(
                "honest trees with BEST_FIRST_GLOBAL growth are not "
                "supported")
        ncand = 0
        if hp.get("num_candidate_attributes", -1) > 0:
            ncand = min(F, int(hp["num_candidate_attributes"]))
        elif hp["num_candidate_attributes_ratio"] > 0:
            ncand = max(1, int(round(hp["num_candidate_attributes_ratio"]
                                     * F)))
        cfg = trainer_lib.TrainerConfig(
            loss=loss, num_trees=hp["num_trees"], max_depth=hp["max_depth"],
            shrinkage=hp["shrinkage"], lambda_l2=hp["l2_regularization"],
            lambda_l1=hp.get("l1_regularization", 0.0),
            min_examples=hp["min_examples"],
            min_hessian=hp["min_sum_hessian_in_leaf"],
            subsample=hp["subsample"],
            sampling_method=hp.get("sampling_method", "RANDOM"),
            goss_alpha=hp.get("goss_alpha", 0.2),
            goss_beta=hp.get("goss_beta", 0.1),
            selgb_ratio=hp.get("selective_gradient_boosting_ratio", 0.01),
            n_classes=n_classes,
            seed=self.random_seed, num_candidate_features=ncand,
            honest=hp.get("honest", False),
            honest_ratio=hp.get("honest_ratio_leaf_examples", 0.5),
            honest_fixed_separation=hp.get("honest_fixed_separation",
                                           False),
            early_stopping=(hp["early_stopping"] != "NONE"
                            and valid_bins is not None),
            early_stopping_num_trees_look_ahead=(
                hp["early_stopping_num_trees_look_ahead"]),
            early_stopping_initial_iteration=(
                hp["early_stopping_initial_iteration"]),
            cat_smooth=hp["l2_categorical_regularization"],
            lambda_loss=hp.get("lambda_loss", 1.0),
            validation_interval=hp.get("validation_interval_in_trees", 1),
            total_max_num_nodes=hp.get("total_max_num_nodes", -1),
            growing_strategy=hp.get("growing_strategy", "LOCAL"),
            max_num_nodes=hp.get("max_num_nodes", 31),
            focal_gamma=hp.get("focal_loss_gamma", 2.0),
            focal_alpha=hp.get("focal_loss_alpha", 0.5),
            na_mode=(getattr(self, "missing_value_policy",
                             "GLOBAL_IMPUTATION") == "LOCAL_IMPUTATION"),
            dart_dropout=(hp.get("dart_dropout", 0.01)
                          if hp.get("forest_extraction") == "DART" else 0.0),
            adapt_sample_for_duration=hp.get(
                "adapt_subsample_for_maximum_training_duration", False),
            **obl,
        )
        t = trainer_lib.ForestTrainer(bins, labels, cfg,
                                      valid_bins=valid_bins,
                                      valid_labels=valid_labels,
                                      cat_flags=cat_flags, weights=weights,
                                      mono=mono, raw=raw_t,
                                      valid_raw=valid_raw_t)
        C = n_classes if loss == trainer_lib.LOSS_MULTINOMIAL else 1
        activation = "identity"
        if custom_loss is not None:
            activation = custom_loss.activation.value \
                if hp["apply_link_function"] else "identity"
        elif hp["apply_link_function"]:
            if loss in (trainer_lib.LOSS_BINOMIAL, trainer_lib.LOSS_FOCAL):
                activation = "sigmoid"
            elif loss == trainer_lib.LOSS_MULTINOMIAL:
                activation = "softmax"
            elif loss == trainer_lib.LOSS_POISSON:
                activation = "exp"
        cat_feats = self._cat_feature_flags(ds)
        names = [c.name for c in ds.dataspec.feature_columns]

        def make_model(flat, init_preds, gains):
            return GradientBoostedTreesModel(
                forest=flat, dataspec=ds.dataspec, task=self._task,
                label_classes=classes, init_predictions=init_preds,
                num_trees_per_iter=C, activation=activation,
                metadata={"feature_gains": gains,
                          # custom python losses have no reference Loss
                          # enum value; export refuses to mislabel them
                          "loss": (int(loss) if custom_loss is None
                                   else "custom"),
                          "missing_value_policy": getattr(
                              self, "missing_value_policy",
                              "GLOBAL_IMPUTATION"),
                          "ranking_group": self.ranking_group,
                          "label_event_observed":
                              self.label_event_observed,
                          "label_entry_age": self.label_entry_age,
                          "ndcg_truncation": self.ndcg_truncation})

        # checkpoint/resume (reference try_resume_training +
        # snapshot interval, abstract_learner.proto:52-56)
        import os as _os

        partial = None
        start_it = 0
        resume_margins = resume_valid_margins = None
        snap_dir = None
        snapshot_cb = None
        if hp.get("working_dir"):
            snap_dir = _os.path.join(hp["working_dir"], "snapshot")
            if hp.get("resume_training") and _os.path.exists(
                    _os.path.join(snap_dir, "done")):
                from ydf_amd.model.model_lib import load_model as _load

                partial = _load(snap_dir)
                start_it = partial.num_trees() // max(C, 1)
                info(f"resuming training from snapshot at iteration "
                     f"{start_it}")

            def _snapshot(trees_so_far, iteration, init_preds_s):
                import shutil

                from ydf_amd.parallel.dist import is_main

                if not is_main():
                    # data-parallel ranks grow identical trees (seeded
                    # masks + summed histograms): rank 0's snapshot is
                    # the job's snapshot; others must not race on the dir
                    return
                flat_s = self._finalize_forest(build_flat_forest(
                    trees_so_far, bnd, leaf_scale=hp["shrinkage"],
                    cat_feats=cat_feats))
                if partial is not None:
                    from ydf_amd.model.forest import concat_forests

                    flat_s = concat_forests(partial.forest, flat_s)
                ms = make_model(flat_s, init_preds_s,
                                self._feature_gains(trees_so_far, names))
                tmp = snap_dir + ".tmp"
                if _os.path.exists(tmp):
                    shutil.rmtree(tmp)
                ms.save(tmp, format="npz")  # snapshots: fast container
                if _os.path.exists(snap_dir):
                    shutil.rmtree(snap_dir)
                _os.replace(tmp, snap_dir)
                info(f"snapshot written at iteration {iteration}")

            snapshot_cb = _snapshot

        if partial is not None:
            import torch as _t

            # exact resume: margins of the partial model over the raw
            # training matrix, re-split with the same permutation
            X_all = ds.X
            perm_margins = partial.predict_margin(
                _t.from_numpy(np.ascontiguousarray(X_all)).to(device))
            # re-apply the train/valid permutation
            if valid_bins is not None:
                rngp = np.random.RandomState(self.random_seed)
                permp = rngp.permutation(X_all.shape[1])
                n_validp = max(1, int(X_all.shape[1]
                                      * hp["validation_ratio"]))                     if X_all.shape[1] > 10 else 0
                vi_idx = _t.from_numpy(permp[:n_validp].copy()).to(device)
                ti_idx = _t.from_numpy(permp[n_validp:].copy()).to(device)
                resume_margins = perm_margins[:, ti_idx].contiguous()
                resume_valid_margins = perm_margins[:, vi_idx].contiguous()
            else:
                resume_margins = perm_margins.contiguous()

        trees, init_preds, logs = trainer_lib.train_gbt(
            t, log=info, start_iteration=start_it,
            resume_margins=resume_margins,
            resume_valid_margins=resume_valid_margins,
            custom_loss=custom_loss, ranking=ranking,
            valid_ranking=valid_ranking, cox=cox, valid_cox=valid_cox,
            snapshot_cb=snapshot_cb,
            snapshot_interval_seconds=hp.get(
                "resume_training_snapshot_interval_seconds", 1800.0),
            max_duration_seconds=hp.get(
                "maximum_training_duration_seconds", -1.0),
            custom_metrics=self.custom_metrics)
        if hp.get("forest_extraction") == "DART":
            # HostTree.scale holds each tree's final absolute leaf scale
            for tr in trees:
                tr.leaf_value = tr.leaf_value * tr.scale
            flat = self._finalize_forest(build_flat_forest(
                trees, bnd, leaf_scale=1.0, cat_feats=cat_feats))
        else:
            flat = self._finalize_forest(build_flat_forest(
                trees, bnd, leaf_scale=hp["shrinkage"],
                cat_feats=cat_feats))
        gains = self._feature_gains(trees, names)
        if partial is not None:
            from ydf_amd.model.forest import concat_forests

            flat = concat_forests(partial.forest, flat)
            pg = (partial.metadata or {}).get("feature_gains", {})
            for k, v in pg.items():
                gains[k] = gains.get(k, 0.0) + v
        model = make_model(flat, init_preds, gains)
        model.training_logs = logs
        model.metadata["early_stopping_triggered"] = bool(
            len(trees) // max(C, 1) < hp["num_trees"]
            and hp["early_stopping"] != "NONE")
        self._finalize_model(model)
        if not hp.get("keep_non_leaf_label_distribution", True):
            # drop non-leaf training distributions (reference
            # keep_non_leaf_label_distribution=false: smaller model,
            # disables distribution-dependent analyses like TreeSHAP)
            model.forest.cover = model.forest.cover.copy()
            model.forest.cover[model.forest.feat >= 0] = 0.0
        if hp.get("compute_permutation_variable_importance"):
            from ydf_amd.utils.analysis import permutation_importances

            key = ("MEAN_DECREASE_IN_ACCURACY"
                   if self._task == Task.CLASSIFICATION
                   else "MEAN_INCREASE_IN_RMSE")
            vi = permutation_importances(model, ds, ds.label_values,
                                         device=device)
            model.metadata["permutation_importances"] = {
                key: [(float(s), n) for s, n in vi]}
        if snapshot_cb is not None:
            # final state also becomes the snapshot (enables continuing
            # with a larger num_trees later)
            snapshot_cb(trees, hp["num_trees"], init_preds)
        return model

    def _prepare_valid(self, valid, train_ds: VerticalDataset, device):
        """Bins a user-provided validation dataset with the TRAIN dataspec."""
        from ydf_amd.dataset.dataset import create_vertical_dataset
        from ydf_amd.model.forest import padded_boundaries

        vds = create_vertical_dataset(valid, dataspec=train_ds.dataspec)
        bnd = padded_boundaries(vds.dataspec.feature_columns)
        bins = self._bin_matrix(vds.X, self._cat_feature_flags(vds), bnd,
                                device)
        labels = torch.from_numpy(
            np.ascontiguousarray(vds.label_values)).to(device)
        return bins, labels, vds


class RandomForestLearner(GenericLearner):
    """Random forest learner (reference learner/random_forest/
    random_forest.cc:917 bagging loop).

    Deviations (documented): bootstrap resampling is Poisson(1)-approximated
    (same expectation as sampling-with-replacement); classification
    aggregates per-tree leaf probabilities (winner_take_all=False
    semantics), which is the reference's recommended setting for
    calibrated probabilities."""

    def __init__(self, label: Optional[str] = None,
                 task: Task = Task.CLASSIFICATION,
                 features: Optional[Sequence[Union[str, Column]]] = None,
                 num_trees: int = 300, max_depth: int = 16,
                 min_examples: int = 5,
                 bootstrap_training_dataset: bool = True,
                 bootstrap_size_ratio: float = 1.0,
                 num_candidate_attributes: int = 0,
                 num_candidate_attributes_ratio: float = -1.0,
                 sampling_with_replacement: bool = True,
                 winner_take_all: bool = True,
                 compute_oob_performances: bool = True,
                 compute_oob_variable_importances: bool = False,
                 num_oob_variable_importances_permutations: int = 1,
                 honest: bool = False,
                 honest_ratio_leaf_examples: float = 0.5,
                 honest_fixed_separation: bool = False,
                 uplift_treatment: Optional[str] = None,
                 uplift_split_score: str = "KULLBACK_LEIBLER",
                 uplift_min_examples_in_treatment: int = 5,
                 adapt_bootstrap_size_ratio_for_maximum_training_duration:
                 bool = False,
                 maximum_training_duration_seconds: float = -1.0,
                 split_axis: str = "AXIS_ALIGNED",
                 sparse_oblique_num_projections_exponent: float = 2.0,
                 sparse_oblique_max_num_projections: int = 6000,
                 sparse_oblique_projection_density_factor: float = 2.0,
                 sparse_oblique_normalization: str = "NONE",
                 sparse_oblique_weights: str = "BINARY",
                 hyperparameter_template: Optional[str] = None,
                 random_seed: int = 123456, **kwargs):
        super().__init__(label=label, task=task, features=features,
                         random_seed=random_seed, **kwargs)
        self._hp_template = hyperparameter_template
        self.hyperparameters = dict(
            num_trees=num_trees, max_depth=max_depth,
            min_examples=min_examples,
            bootstrap_training_dataset=bootstrap_training_dataset,
            bootstrap_size_ratio=bootstrap_size_ratio,
            num_candidate_attributes=num_candidate_attributes,
            num_candidate_attributes_ratio=num_candidate_attributes_ratio,
            sampling_with_replacement=sampling_with_replacement,
            winner_take_all=winner_take_all,
            num_oob_variable_importances_permutations=(
                num_oob_variable_importances_permutations),
            compute_oob_performances=compute_oob_performances,
            compute_oob_variable_importances=(
                compute_oob_variable_importances),
            honest=honest,
            honest_ratio_leaf_examples=honest_ratio_leaf_examples,
            honest_fixed_separation=honest_fixed_separation,
            uplift_treatment=uplift_treatment,
            uplift_split_score=uplift_split_score,
            uplift_min_examples_in_treatment=(
                uplift_min_examples_in_treatment),
            maximum_training_duration_seconds=(
                maximum_training_duration_seconds),
            adapt_bootstrap_size_ratio_for_maximum_training_duration=(
                adapt_bootstrap_size_ratio_for_maximum_training_duration),
            split_axis=split_axis,
            sparse_oblique_num_projections_exponent=(
                sparse_oblique_num_projections_exponent),
            sparse_oblique_max_num_projections=(
                sparse_oblique_max_num_projections),
            sparse_oblique_projection_density_factor=(
                sparse_oblique_projection_density_factor),
            sparse_oblique_normalization=sparse_oblique_normalization,
            sparse_oblique_weights=sparse_oblique_weights,
        )

    def _train_uplift(self, data, device) -> RandomForestModel:
        """Uplift forest (reference uplift tasks; trees split on
        treatment/control divergence, leaves store
        E[outcome|treatment] - E[outcome|control])."""
        from ydf_amd.dataset.dataset import _to_column_dict
        from ydf_amd.learner.uplift import train_uplift_forest
        from ydf_amd.model.forest import padded_boundaries

        hp = self.hyperparameters
        tcol = hp.get("uplift_treatment")
        if not tcol:
            raise ValueError("uplift tasks need uplift_treatment=")
        cols = _to_column_dict(data)
        if tcol not in cols:
            raise ValueError(f"treatment column {tcol!r} missing")
        tvals = np.asarray(cols.pop(tcol))
        if self.features is None:
            self.features = [k for k in cols if k != self.label]
        # binary treatment: positive = second vocab item
        # (frequency-ordered like a categorical label)
        if tvals.dtype.kind in "UOS":
            uniq, counts = np.unique(tvals.astype(str), return_counts=True)
            order = np.argsort(-counts, kind="stable")
            tvocab = [str(uniq[i]) for i in order]
            treat = (tvals.astype(str) == tvocab[1]).astype(np.float32)
        else:
            uniq = np.unique(tvals)
            if len(uniq) != 2:
                raise ValueError("treatment must be binary")
            tvocab = [str(uniq[0]), str(uniq[1])]
            treat = (tvals == uniq[1]).astype(np.float32)
        inner_task = Task.CLASSIFICATION \
            if self._task == Task.CATEGORICAL_UPLIFT else Task.REGRESSION
        saved_task = self._task
        self._task = inner_task
        try:
            ds, bins, labels, bnd, cat_flags, weights, mono = \
                self._prepare(cols, device)
        finally:
            self._task = saved_task
        if labels is None:
            raise ValueError(f"label column {self.label!r} missing")
        F = bins.shape[0]
        trees = train_uplift_forest(
            bins, labels, torch.from_numpy(treat).to(device),
            num_trees=hp["num_trees"], max_depth=hp["max_depth"],
            min_examples=hp["min_examples"],
            min_examples_in_treatment=hp.get(
                "uplift_min_examples_in_treatment", 5),
            split_score=hp.get("uplift_split_score", "KULLBACK_LEIBLER"),
            num_candidate_features=self._num_candidate(F),
            bootstrap=hp["bootstrap_training_dataset"],
            seed=self.random_seed, weights=weights, log=info)
        flat = self._finalize_forest(build_flat_forest(
            trees, bnd, leaf_scale=1.0,
            cat_feats=self._cat_feature_flags(ds)))
        classes = self._label_classes(ds) \
            if self._task == Task.CATEGORICAL_UPLIFT else None
        model = RandomForestModel(
            forest=flat, dataspec=ds.dataspec, task=self._task,
            label_classes=classes, init_predictions=[0.0],
            num_trees_per_iter=1, activation="identity",
            metadata={"uplift_treatment": tcol,
                      "treatment_vocab": tvocab,
                      "feature_gains": self._feature_gains(
                          trees,
                          [c.name for c in ds.dataspec.feature_columns])})
        self._last_trees = trees
        self._last_ds = ds
        return model

    def _num_candidate(self, F: int) -> int:
        hp = self.hyperparameters
        if hp["num_candidate_attributes_ratio"] > 0:
            return max(1, int(round(hp["num_candidate_attributes_ratio"]
                                    * F)))
        k = hp["num_candidate_attributes"]
        if k > 0:
            return min(k, F)
        if k == -1:
            return F
        # k == 0: reference default — sqrt(F) classification, F/3 regression
        if self._task == Task.CLASSIFICATION:
            return max(1, int(math.sqrt(F) + 0.5))
        return max(1, F // 3)

    def train(self, data, valid=None, verbose=None) -> RandomForestModel:
        if getattr(self, "feature_selector", None) is not None:
            return self._train_with_feature_selection(data, valid)
        if self.tuner is not None:
            return self._train_with_tuner(data, valid=valid)
        hp = self.hyperparameters
        device = self._resolve_device()
        if self._task in (Task.CATEGORICAL_UPLIFT, Task.NUMERICAL_UPLIFT):
            return self._train_uplift(data, device)
        ds, bins, labels, bnd, cat_flags, weights, mono = self._prepare(
            data, device)
        if labels is None:
            raise ValueError(f"label column {self.label!r} missing")
        classes = self._label_classes(ds) \
            if self._task == Task.CLASSIFICATION else None
        n_classes = len(classes) if classes else 2
        F = bins.shape[0]
        obl = self._oblique_cfg(F, cat_flags)
        raw_t = None
        if obl:
            raw_t = torch.from_numpy(np.ascontiguousarray(ds.X)).to(device)
        # with oblique projections the candidate pool is F + P virtual
        # features; the reference evaluates every projection, so feature
        # sampling is disabled unless explicitly requested
        ncand = self._num_candidate(F)
        if obl and hp["num_candidate_attributes"] == 0 \
                and hp["num_candidate_attributes_ratio"] <= 0:
            ncand = 0
        cfg = trainer_lib.TrainerConfig(
            loss=trainer_lib.LOSS_RF, num_trees=hp["num_trees"],
            max_depth=hp["max_depth"], shrinkage=1.0, lambda_l2=0.0,
            min_examples=hp["min_examples"], min_hessian=0.0,
            n_classes=n_classes, seed=self.random_seed,
            bootstrap=hp["bootstrap_training_dataset"],
            bootstrap_ratio=hp.get("bootstrap_size_ratio", 1.0),
            with_replacement=hp.get("sampling_with_replacement", True),
            oob_vi_permutations=hp.get(
                "num_oob_variable_importances_permutations", 1),
            num_candidate_features=ncand,
            max_duration_seconds=hp.get(
                "maximum_training_duration_seconds", -1.0),
            adapt_sample_for_duration=hp.get(
                "adapt_bootstrap_size_ratio_for_maximum_training_duration",
                False),
            honest=hp.get("honest", False),
            na_mode=(getattr(self, "missing_value_policy",
                             "GLOBAL_IMPUTATION") == "LOCAL_IMPUTATION"),
            honest_ratio=hp.get("honest_ratio_leaf_examples", 0.5),
            honest_fixed_separation=hp.get("honest_fixed_separation",
                                           False),
            **obl,
        )
        t = trainer_lib.ForestTrainer(bins, labels, cfg,
                                      cat_flags=cat_flags, weights=weights,
                                      mono=mono, raw=raw_t)
        compute_oob = (hp["compute_oob_performances"]
                       and hp["bootstrap_training_dataset"])
        result = trainer_lib.train_rf(t, log=info, compute_oob=compute_oob)
        oob_eval = None
        if compute_oob:
            trees, oob_sum, oob_cnt = result
            from ydf_amd.metric.metric import evaluate_predictions

            cnt = oob_cnt.cpu().numpy()
            covered = cnt > 0
            if covered.any():
                preds = (oob_sum.cpu().numpy()[:, covered]
                         / cnt[covered]).T
                if preds.shape[1] == 1:
                    preds = preds[:, 0]
                y = labels.cpu().numpy()[covered]
                oob_eval = evaluate_predictions(
                    preds, y, self._task, n_classes)
        else:
            trees = result
        # winner_take_all (reference RF default): each tree votes its
        # majority class; for binary trees this is a leaf-value threshold,
        # so the vote transform happens at model-build time
        self._last_trees = trees
        self._last_ds = ds
        wta = (hp["winner_take_all"]
               and self._task == Task.CLASSIFICATION and n_classes == 2)
        if wta:
            for tr in trees:
                tr.leaf_value = (tr.leaf_value > 0.5).astype(np.float32)
        flat = self._finalize_forest(build_flat_forest(
            trees, bnd, leaf_scale=1.0,
            cat_feats=self._cat_feature_flags(ds)))
        C = n_classes if (classes and n_classes > 2) else 1
        model = RandomForestModel(
            forest=flat, dataspec=ds.dataspec, task=self._task,
            label_classes=classes, init_predictions=[0.0] * max(C, 1),
            num_trees_per_iter=C, activation="identity",
            metadata={"feature_gains": self._feature_gains(
                trees, [c.name for c in ds.dataspec.feature_columns]),
                "missing_value_policy": getattr(
                    self, "missing_value_policy", "GLOBAL_IMPUTATION"),
                "winner_take_all": wta})
        model._self_evaluation = oob_eval
        self._finalize_model(model)
        if hp.get("compute_oob_variable_importances") \
                and hp["bootstrap_training_dataset"]:
            vi = _oob_permutation_vi(model, ds, cfg, self._task, device)
            if vi:
                model.metadata["oob_permutation_importances"] = vi
        return model


def _oob_permutation_vi(model, ds, cfg, task, device):
    """Per-tree OOB permutation variable importances (reference
    random_forest.cc:1411-1477 ComputeVariableImportancesFromAccumulated):
    each tree is evaluated on ITS out-of-bag rows (bootstrap weights
    regenerated from the per-tree seed) with each feature permuted in
    turn; the metric drop is averaged over trees. Binary classification
    -> MEAN_DECREASE_IN_ACCURACY (+ MEAN_DECREASE_IN_AUC omitted);
    regression -> MEAN_INCREASE_IN_RMSE."""
    from ydf_amd.learner import trainer as trainer_lib
    from ydf_amd.model.generic_model import _DeviceForest

    if task == Task.CLASSIFICATION and model.label_classes \
            and len(model.label_classes) > 2:
        return None  # multi-class per-tree votes: not supported yet
    X = np.ascontiguousarray(ds.X)
    y = ds.label_values.astype(np.float32)
    F, N = X.shape
    T = model.forest.n_trees
    names = model.input_feature_names()
    cpu = torch.device("cpu")
    df = _DeviceForest(model.forest, cpu)
    rng = np.random.RandomState(cfg.seed ^ 0x00bafeed)
    acc_drop = np.zeros(F, dtype=np.float64)
    n_used = 0

    def tree_metric(Xsub_t, yb, t):
        out = torch.empty(Xsub_t.shape[1], dtype=torch.float32)
        from ydf_amd import ops as _ops

        _ops.predict_forest(Xsub_t, df.feat, df.thr, df.left, df.roots,
                            out, tree_start=t, tree_step=1, n_trees=1,
                            cat_idx=df.cat_idx, masks=df.masks,
                            obl_ranges=df.obl_ranges, obl_attr=df.obl_attr,
                            obl_w=df.obl_w)
        p = out.numpy()
        if task == Task.REGRESSION:
            return -float(np.sqrt(np.mean((p - yb) ** 2)))  # higher=better
        return float(((p > 0.5) == (yb > 0.5)).mean())

    n_perm = max(1, int(getattr(cfg, "oob_vi_permutations", 1)))
    for t in range(T):
        w = trainer_lib.rf_bootstrap_weights(cfg.seed, t, N, device,
                                             cfg.bootstrap_ratio,
                                             cfg.with_replacement)
        oob = (w == 0).cpu().numpy()
        if oob.sum() < 10:
            continue
        n_used += 1
        Xs = np.ascontiguousarray(X[:, oob])
        yb = y[oob]
        Xt = torch.from_numpy(Xs)
        base = tree_metric(Xt, yb, t)
        for f in range(F):
            saved = Xs[f].copy()
            drop = 0.0
            for _ in range(n_perm):
                Xs[f] = saved[rng.permutation(len(saved))]
                drop += base - tree_metric(Xt, yb, t)
            acc_drop[f] += drop / n_perm
            Xs[f] = saved
    if n_used == 0:
        return None
    acc_drop /= n_used
    key = "MEAN_INCREASE_IN_RMSE" if task == Task.REGRESSION \
        else "MEAN_DECREASE_IN_ACCURACY"
    order = np.argsort(-acc_drop, kind="stable")
    return {key: [[float(acc_drop[i]), names[i]] for i in order]}


class CartLearner(RandomForestLearner):
    """CART: a single tree, no bagging, all features as candidates, with
    validation-set pruning (reference learner/cart/cart.h:44)."""

    def __init__(self, label: Optional[str] = None,
                 task: Task = Task.CLASSIFICATION,
                 max_depth: int = 16, min_examples: int = 5,
                 validation_ratio: float = 0.1, **kwargs):
        kwargs.setdefault("num_trees", 1)
        kwargs.setdefault("bootstrap_training_dataset", False)
        kwargs.setdefault("num_candidate_attributes", -1)
        kwargs.setdefault("compute_oob_performances", False)
        super().__init__(label=label, task=task, max_depth=max_depth,
                         min_examples=min_examples, **kwargs)
        self.hyperparameters["validation_ratio"] = validation_ratio

    def train(self, data, valid=None, verbose=None):
        vr = self.hyperparameters.get("validation_ratio", 0.1)
        from ydf_amd.dataset.dataset import _to_column_dict

        if vr <= 0 or self._task not in (Task.CLASSIFICATION,
                                         Task.REGRESSION):
            return super().train(data, valid=valid, verbose=verbose)
        cols = _to_column_dict(data)
        n = len(next(iter(cols.values())))
        rng = np.random.RandomState(self.random_seed)
        perm = rng.permutation(n)
        n_valid = int(n * vr)
        if n_valid < 10:
            return super().train(data, valid=valid, verbose=verbose)
        vidx, tidx = perm[:n_valid], perm[n_valid:]
        train_cols = {k: np.asarray(v)[tidx] for k, v in cols.items()}
        model = super().train(train_cols, verbose=verbose)
        # prune with the held-out rows (reference cart.cc pruning)
        from ydf_amd.learner.pruning import prune_tree
        from ydf_amd.model.forest import build_flat_forest, \
            padded_boundaries

        tree = self._last_trees[0]
        ds = self._last_ds
        bnd = padded_boundaries(ds.dataspec.feature_columns)
        cat_feats = self._cat_feature_flags(ds)
        valid_cols = {k: np.asarray(v)[vidx] for k, v in cols.items()}
        from ydf_amd.dataset.dataset import create_vertical_dataset

        vds = create_vertical_dataset(valid_cols, dataspec=ds.dataspec)
        pruned = prune_tree(tree, vds.X, vds.label_values, bnd, cat_feats,
                            self._task)
        info(f"CART pruning removed {pruned} nodes")
        model.forest = self._finalize_forest(build_flat_forest(
            [tree], bnd, leaf_scale=1.0, cat_feats=cat_feats))
        model._dev_forest = {}
        model._thr_on_cuts = None  # forest changed: re-check eligibility
        return model


class IsolationForestLearner(GenericLearner):
    """Isolation forest (reference learner/isolation_forest/
    isolation_forest.h:40): subsampled trees with uniform random
    axis-aligned splits; max depth = ceil(log2(subsample)) per
    isolation_forest.cc:670. Trees are built host-side (tiny subsamples);
    scoring runs through the batch inference kernels."""

    def __init__(self, label: Optional[str] = None,
                 task: Task = Task.ANOMALY_DETECTION,
                 features: Optional[Sequence[Union[str, Column]]] = None,
                 num_trees: int = 300, subsample_count: int = 256,
                 subsample_ratio: Optional[float] = None,
                 max_depth: int = -2, random_seed: int = 123456, **kwargs):
        super().__init__(label=label, task=task, features=features,
                         random_seed=random_seed, **kwargs)
        self.hyperparameters = dict(
            num_trees=num_trees, subsample_count=subsample_count,
            subsample_ratio=subsample_ratio, max_depth=max_depth)

    def train(self, data, valid=None, verbose=None) -> IsolationForestModel:
        from ydf_amd.dataset.dataset import create_vertical_dataset

        hp = self.hyperparameters
        if isinstance(data, VerticalDataset):
            ds = data
        else:
            ds = create_vertical_dataset(data, label=self.label,
                                         task=self._task,
                                         features=self.features)
        X = ds.X  # [F, N]
        F, N = X.shape
        rng = np.random.RandomState(self.random_seed)
        if hp["subsample_ratio"] is not None:
            sub = max(2, int(N * hp["subsample_ratio"]))
        else:
            sub = min(hp["subsample_count"], N)
        max_depth = hp["max_depth"]
        if max_depth < 0:  # -2: reference default ceil(log2(subsample))
            max_depth = max(1, int(math.ceil(math.log2(max(sub, 2)))))

        c = IsolationForestModel.expected_path_length
        feats, thrs, lefts, roots, covers = [], [], [], [], []

        def new_node() -> int:
            feats.append(-1)
            thrs.append(0.0)
            lefts.append(0)
            covers.append(0.0)
            return len(feats) - 1

        def build(root_rows: np.ndarray) -> None:
            # children are allocated as adjacent pairs (flat layout needs
            # right = left + 1)
            root = new_node()
            stack = [(root, root_rows, 0)]
            while stack:
                my, rows, depth = stack.pop()
                n = len(rows)
                covers[my] = float(n)
                split = None
                if depth < max_depth and n > 1:
                    # random feature with a non-constant range, uniform cut
                    for _ in range(8):
                        f = rng.randint(F)
                        vals = X[f, rows]
                        lo, hi = float(vals.min()), float(vals.max())
                        if hi > lo:
                            split = (f, float(rng.uniform(lo, hi)), vals)
                            break
                if split is None:
                    thrs[my] = float(depth + c(n))
                    continue
                f, cut, vals = split
                go_right = vals > cut
                feats[my] = f
                thrs[my] = cut
                li = new_node()
                ri = new_node()
                lefts[my] = li
                stack.append((li, rows[~go_right], depth + 1))
                stack.append((ri, rows[go_right], depth + 1))

        for _ in range(hp["num_trees"]):
            rows = rng.choice(N, size=sub, replace=False)
            roots.append(len(feats))
            build(rows)

        flat = FlatForest(feat=np.asarray(feats, np.int32),
                          thr=np.asarray(thrs, np.float32),
                          left=np.asarray(lefts, np.int32),
                          roots=np.asarray(roots, np.int32),
                          cover=np.asarray(covers, np.float32))
        return IsolationForestModel(
            forest=flat, dataspec=ds.dataspec, task=self._task,
            label_classes=None, init_predictions=[0.0],
            num_trees_per_iter=1, activation="isolation",
            num_examples_per_tree=sub)


DecisionTreeLearner = CartLearner
