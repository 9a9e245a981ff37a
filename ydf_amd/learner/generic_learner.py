"""Learner base class (capability analogue of the reference AbstractLearner,
learner/abstract_learner.h:42, and PYDF GenericLearner,
port/python/ydf/learner/generic_learner.py:255)."""
from __future__ import annotations

from typing import Dict, Optional, Sequence, Union

import numpy as np
import torch

from ydf_amd.dataset.dataset import VerticalDataset, create_vertical_dataset
from ydf_amd.dataset.dataspec import Column, Semantic, Task
from ydf_amd.model.forest import padded_boundaries
from ydf_amd import ops


# Shared decision-tree hyperparameters accepted by EVERY tree learner
# (reference GetGenericHyperParameterSpecification surface; PYDF
# generates each learner signature from it). A name listed here is
# accepted as a keyword by any learner; settings this framework cannot
# honor raise NotImplementedError at train time (see
# _validate_extra_hp). None = "not set" sentinel.
SHARED_TREE_PARAMS = {
    "explicit_args": None,
    "class_weights": None,              # implemented (example weights)
    "label_classes": None,
    "include_all_columns": False,
    "max_num_scanned_rows_to_infer_semantic": None,       # perf cap
    "max_num_scanned_rows_to_compute_statistics": None,   # perf cap
    "num_discretized_numerical_bins": None,  # implemented (max_bins)
    "discretize_numerical_columns": None,    # 256-bin path is core design
    "categorical_set_split_greedy_sampling": None,
    "categorical_set_split_max_num_items": None,
    "categorical_set_split_min_item_frequency": None,
    "categorical_set_split_greedy_maximum_mask_size": None,
    "categorical_random_max_num_trials": None,
    "categorical_random_num_trial_exponent": None,
    "sparse_oblique_weights_integer_minimum": None,
    "sparse_oblique_weights_integer_maximum": None,
    "sparse_oblique_weights_power_of_two_min_exponent": None,
    "sparse_oblique_weights_power_of_two_max_exponent": None,
    "numerical_vector_sequence_num_examples": None,
    "numerical_vector_sequence_num_random_anchors": None,
    "numerical_vector_sequence_enable_closer_than_conditions": None,
    "numerical_vector_sequence_enable_projected_more_than_conditions":
        None,
    "sorting_strategy": None,
    "in_split_min_examples_check": None,
    "keep_non_leaf_label_distribution": None,
    "growing_strategy": None,
    "max_num_nodes": None,
    "honest": None,
    "honest_ratio_leaf_examples": None,
    "honest_fixed_separation": None,
    "num_candidate_attributes": None,
    "num_candidate_attributes_ratio": None,
    "mhld_oblique_max_num_attributes": None,
    "mhld_oblique_sample_attributes": None,
    "split_axis": None,
    "sparse_oblique_max_num_features": None,
    "sparse_oblique_max_num_projections": None,
    "sparse_oblique_normalization": None,
    "sparse_oblique_num_projections_exponent": None,
    "sparse_oblique_projection_density_factor": None,
    "sparse_oblique_weights": None,
    "min_examples": None,
    "working_dir": None,
    "ranking_group": None,
    "uplift_treatment": None,
    "uplift_split_score": None,
    "uplift_min_examples_in_treatment": None,
    "maximum_training_duration_seconds": None,
}


class GenericLearner:
    """Base learner: dataset ingestion, binning, device selection."""

    def __init__(self, label: Optional[str], task: Task = Task.CLASSIFICATION,
                 features: Optional[Sequence[Union[str, Column]]] = None,
                 weights: Optional[str] = None, tuner=None,
                 max_vocab_count: int = 2000, min_vocab_frequency: int = 1,
                 allow_na_conditions: bool = False,
                 pure_serving_model: bool = False,
                 missing_value_policy: str = "GLOBAL_IMPUTATION",
                 categorical_algorithm: str = "CART",
                 maximum_model_size_in_memory_in_bytes: float = -1.0,
                 random_seed: int = 123456, device=None,
                 feature_selector=None,
                 num_threads: Optional[int] = None,
                 data_spec=None, extra_training_config=None, **extra_hp):
        self._extra_hp = {}
        for k, v in extra_hp.items():
            if k not in SHARED_TREE_PARAMS:
                raise TypeError(
                    f"{type(self).__name__} got an unexpected keyword "
                    f"argument {k!r}")
            if v is not None:
                self._extra_hp[k] = v
        self.feature_selector = feature_selector
        # pre-built dataspec (reference data_spec constructor arg):
        # skips dataspec inference; data is encoded with THIS spec
        self.data_spec_override = data_spec
        if extra_training_config is not None:
            raise NotImplementedError(
                "extra_training_config (raw TrainingConfig proto "
                "extensions) is not supported; use the keyword "
                "hyperparameters instead")
        self.allow_na_conditions = allow_na_conditions
        self.pure_serving_model = pure_serving_model
        if missing_value_policy not in ("GLOBAL_IMPUTATION",
                                        "LOCAL_IMPUTATION",
                                        "RANDOM_LOCAL_IMPUTATION"):
            raise ValueError(
                f"unknown missing_value_policy {missing_value_policy!r}")
        self.missing_value_policy = missing_value_policy
        if categorical_algorithm != "CART":
            raise NotImplementedError(
                "categorical_algorithm: only CART (sorted set-splits) "
                "is implemented")
        self.max_model_bytes = maximum_model_size_in_memory_in_bytes
        self.label = label
        self._task = task
        self.features = features
        self.weights_col = weights
        self.tuner = tuner
        self.max_vocab_count = max_vocab_count
        self.min_vocab_frequency = min_vocab_frequency
        self.random_seed = random_seed
        self.device = device
        self.num_threads = num_threads
        self.hyperparameters: Dict = {}

    def task(self) -> Task:
        return self._task

    @property
    def learner_name(self) -> str:
        """Registry name of this learner, e.g. "RANDOM_FOREST" (PYDF
        learner.learner_name)."""
        from ydf_amd.utils import registry

        registry._bootstrap()
        for name, cls in registry.learner_registry._items.items():
            if cls is type(self):
                return name
        # fall back to CamelCase -> SNAKE of the class name
        import re

        return re.sub(r"(?<!^)(?=[A-Z])", "_",
                      type(self).__name__.replace("Learner", "")).upper()

    def extract_input_feature_names(self, ds) -> list:
        """Input-feature column names of `ds` for this learner: every
        column except the label/weights/group special columns, or the
        explicit `features` list filtered to what the data provides
        (PYDF learner.extract_input_feature_names)."""
        from ydf_amd.dataset.dataset import _to_column_dict

        cols = _to_column_dict(ds)
        non_input = {self.label, self.weights_col,
                     getattr(self, "ranking_group", None),
                     getattr(self, "uplift_treatment", None),
                     getattr(self, "label_event_observed", None),
                     getattr(self, "label_entry_age", None)}
        if self.features is not None:
            want = [f if isinstance(f, str) else f.name
                    for f in self.features]
            return [f for f in want if f in cols]
        return [c for c in cols if c not in non_input]

    # ------------------------------------------------------------------
    def _resolve_device(self) -> torch.device:
        if self.device is not None:
            return torch.device(self.device)
        return torch.device("cuda") if torch.cuda.is_available() else \
            torch.device("cpu")

    def _cat_feature_flags(self, ds: VerticalDataset) -> np.ndarray:
        """bool [F]: which features use categorical set-splits (CATEGORICAL
        semantic; categories beyond 255 share the last bin)."""
        return np.asarray(
            [c.semantic == Semantic.CATEGORICAL
             for c in ds.dataspec.feature_columns], dtype=bool)

    def _finalize_model(self, model) -> None:
        """pure_serving_model strips training-only payloads (reference
        MakePureServing, abstract_model.h:433); the in-memory size cap
        (maximum_model_size_in_memory_in_bytes) is checked post-train."""
        if getattr(self, "pure_serving_model", False):
            model.training_logs = None
            model.tuner_logs = None
            model._self_evaluation = None
            if model.metadata:
                model.metadata.pop("feature_gains", None)
            model.forest.cover = np.zeros_like(model.forest.cover)
        cap = getattr(self, "max_model_bytes", -1.0)
        if cap and cap > 0:
            f = model.forest
            size = sum(a.nbytes for a in (f.feat, f.thr, f.left, f.roots,
                                          f.cat_idx, f.masks, f.cover,
                                          f.obl_attr, f.obl_w,
                                          f.na_right))
            if size > cap:
                from ydf_amd.utils.log import info as _info

                _info(f"model size {size}B exceeds "
                      f"maximum_model_size_in_memory_in_bytes={cap:g}")

    def _oblique_cfg(self, F: int, cat_flags) -> Dict:
        """TrainerConfig kwargs for sparse-oblique splits (reference
        SparseObliqueSplit defaults, decision_tree.proto:173-296):
        P = clamp(ceil(n_numerical ^ exponent), 1, max)."""
        hp = self.hyperparameters
        if hp.get("split_axis", "AXIS_ALIGNED") != "SPARSE_OBLIQUE":
            return {}
        import math as _math

        n_num = F - int(cat_flags.sum().item()) \
            if cat_flags is not None else F
        if n_num <= 0:
            return {}
        P = int(min(hp.get("sparse_oblique_max_num_projections", 6000),
                    max(1, _math.ceil(n_num ** hp.get(
                        "sparse_oblique_num_projections_exponent", 2.0)))))
        return dict(
            oblique_projections=P,
            oblique_density=hp.get(
                "sparse_oblique_projection_density_factor", 2.0),
            oblique_weights=hp.get("sparse_oblique_weights", "BINARY"),
            oblique_norm=hp.get("sparse_oblique_normalization", "NONE"),
            oblique_max_features=hp.get(
                "sparse_oblique_max_num_features", -1),
        )

    def _train_with_feature_selection(self, data, valid):
        """PYDF learner(feature_selector=...) integration: run the
        selector, train the final model on the selected features and
        attach the selection logs (PYDF set_feature_selection_logs)."""
        import copy as _copy

        fs = self.feature_selector
        base = _copy.copy(self)
        base.feature_selector = None
        logs = fs.run(base, data, valid if valid is not None else data)
        final = _copy.copy(self)
        final.feature_selector = None
        final.features = list(logs.selected_features)
        model = final.train(data)
        model.set_feature_selection_logs(logs)
        return model

    def _finalize_forest(self, flat):
        """Post-build forest fixups: expands group-space masks on
        large-vocab categorical features into full-dictionary set
        conditions."""
        bigcat = getattr(self, "_bigcat_group_of_code", None)
        if bigcat:
            from ydf_amd.model.forest import expand_bigcat_masks

            flat = expand_bigcat_masks(flat, bigcat)
        return flat

    def _compute_bigcat_maps(self, ds, cat_feats: np.ndarray):
        """Large-vocab categorical support (vocab > 256): orders the
        full dictionary by the global mean-label statistic (CART-style
        ordering, reference splitter_scanner.h:859 applies it per node)
        and packs contiguous runs into <=256 mass-balanced groups. The
        kernels train on group indices; chosen group masks expand back
        to full-dictionary set conditions at model build."""
        na_mode = getattr(self, "missing_value_policy",
                          "GLOBAL_IMPUTATION") == "LOCAL_IMPUTATION"
        n_groups = 255 if na_mode else 256
        specs = ds.dataspec.feature_columns
        y = ds.label_values
        maps = {}
        for fi in np.nonzero(cat_feats)[0]:
            spec = specs[fi]
            V = spec.vocab_size
            if V <= n_groups or y is None:
                continue
            codes = ds.X[fi].astype(np.int64)
            valid = codes >= 0
            cnts = np.bincount(codes[valid], minlength=V).astype(
                np.float64)
            sums = np.bincount(codes[valid], weights=y[valid],
                               minlength=V)
            stat = sums / np.maximum(cnts, 1.0)
            order = np.argsort(stat, kind="stable")
            csum = np.cumsum(np.maximum(cnts[order], 1.0))
            g_of_rank = np.minimum(
                ((csum - 1.0) * n_groups / csum[-1]).astype(np.int64),
                n_groups - 1)
            group_of_code = np.empty(V, dtype=np.int32)
            group_of_code[order] = g_of_rank
            maps[int(fi)] = group_of_code
        return maps or None

    def _bin_matrix(self, ds_X: np.ndarray, cat_feats: np.ndarray,
                    bnd: np.ndarray, device: torch.device) -> torch.Tensor:
        """Bins numericals by quantile cuts; categorical codes pass through
        as their own bin index (clamped to 255; large vocabularies remap
        through the CART-ordered group map). Under LOCAL_IMPUTATION NaN
        rows land in reserved bin 255."""
        policy = getattr(self, "missing_value_policy",
                         "GLOBAL_IMPUTATION")
        na_mode = policy == "LOCAL_IMPUTATION"
        random_na = policy == "RANDOM_LOCAL_IMPUTATION"
        X = torch.from_numpy(np.ascontiguousarray(ds_X)).to(device)
        bnd_t = torch.from_numpy(bnd).to(device)
        bins = torch.empty(X.shape, dtype=torch.uint8, device=device)
        ops.bin_data(X, bnd_t, bins, na_to_255=na_mode or random_na)
        if random_na:
            # RANDOM_LOCAL_IMPUTATION (reference decision_tree.proto:
            # 99-103, Random Survival Forests): missing values imputed
            # by randomly sampled observed values. Deviation
            # (documented): sampled once per forest from the feature's
            # GLOBAL observed bin distribution (the reference redraws
            # per node) — preserves the feature distribution, which is
            # the policy's point, while keeping the binned store
            # immutable across trees. Serving imputes the global mean
            # (deterministic predictions).
            g = torch.Generator(device="cpu")
            g.manual_seed(int(self.random_seed) * 7919 + 13)
            for f in range(bins.shape[0]):
                na = bins[f] == 255
                n_na = int(na.sum().item())
                if n_na == 0:
                    continue
                obs = bins[f][~na]
                if obs.numel() == 0:
                    bins[f][na] = 0
                    continue
                pick = torch.randint(0, obs.numel(), (n_na,),
                                     generator=g).to(device)
                bins[f][na] = obs[pick]
        ci = np.nonzero(cat_feats)[0]
        bigcat = getattr(self, "_bigcat_group_of_code", None) or {}
        if ci.size:
            idx_np = np.asarray([i for i in ci if i not in bigcat])
            if idx_np.size:
                idx = torch.from_numpy(idx_np).to(device)
                bins[idx] = X[idx].clamp_(0, 255).to(torch.uint8)
            for fi, g_of_c in bigcat.items():
                gt = torch.from_numpy(
                    g_of_c.astype(np.int64)).to(device)
                codes = X[fi].long()
                na = codes < 0
                grp = gt[codes.clamp(0, len(g_of_c) - 1)]
                if na_mode:
                    grp = torch.where(na, torch.full(
                        (), 255, dtype=torch.int64, device=device), grp)
                bins[fi] = grp.to(torch.uint8)
        del X
        return bins

    def _notify_usage_start(self, ds) -> float:
        import time as _time

        from ydf_amd.utils import usage

        usage.on_training_start(type(self).__name__, ds.n_examples)
        return _time.monotonic()

    def _notify_usage_end(self, ds, model, t0: float) -> None:
        import time as _time

        from ydf_amd.utils import usage

        usage.on_training_end(
            type(self).__name__, ds.n_examples,
            model.num_trees() if hasattr(model, "num_trees") else None,
            _time.monotonic() - t0)

    def _validate_extra_hp(self) -> None:
        eh = getattr(self, "_extra_hp", {})
        if not eh:
            return
        # merge names the specialized trainers read from hp (explicit
        # constructor params always win over the shared-table form)
        for k, v in eh.items():
            if hasattr(self, "hyperparameters"):
                self.hyperparameters.setdefault(k, v)
        if eh.get("label_classes"):
            raise NotImplementedError(
                "label_classes (explicit class order) is not "
                "implemented; classes are frequency-ordered as in the "
                "default reference dataspec inference")
        if eh.get("include_all_columns"):
            raise NotImplementedError(
                "include_all_columns=True is not implemented; list the "
                "columns in `features` instead")
        if eh.get("mhld_oblique_sample_attributes"):
            raise NotImplementedError(
                "mhld_oblique_sample_attributes is not implemented")
        if eh.get("in_split_min_examples_check") is False:
            raise NotImplementedError(
                "in_split_min_examples_check=False is not implemented")
        if eh.get("categorical_random_max_num_trials") or \
                eh.get("categorical_random_num_trial_exponent"):
            raise NotImplementedError(
                "categorical_algorithm=RANDOM trials are not "
                "implemented (CART set-splits are used)")
        ss = eh.get("sorting_strategy")
        if ss is not None and ss not in ("IN_NODE", "PRESORT",
                                         "FORCE_PRESORT", "AUTO",
                                         "LAYER"):
            raise ValueError(f"unknown sorting_strategy {ss!r}")

    def _prepare(self, data, device: torch.device):
        """Dataset -> (VerticalDataset, binned u8 [F,N] on device,
        labels f32 [N] on device, padded boundary matrix np [F,n_cuts],
        cat_flags u8 tensor or None)."""
        self._validate_extra_hp()
        weights_np = None
        if isinstance(data, VerticalDataset):
            ds = data
        else:
            from ydf_amd.dataset.dataset import _to_column_dict

            features = self.features
            cols = _to_column_dict(data)
            if self.weights_col is not None and features is None:
                features = [c for c in cols
                            if c not in (self.label, self.weights_col)]
            local_na = getattr(self, "missing_value_policy",
                               "GLOBAL_IMPUTATION") in (
                "LOCAL_IMPUTATION", "RANDOM_LOCAL_IMPUTATION")
            nb = getattr(self, "_extra_hp", {}).get(
                "num_discretized_numerical_bins")
            max_bins = 255 if local_na else 256
            if nb:
                max_bins = max(2, min(int(nb), max_bins))
            if getattr(self, "data_spec_override", None) is not None:
                ds = create_vertical_dataset(
                    cols, dataspec=self.data_spec_override)
            else:
                ds = create_vertical_dataset(
                    cols, label=self.label, task=self._task,
                    features=features,
                    max_vocab_count=self.max_vocab_count,
                    min_vocab_frequency=self.min_vocab_frequency,
                    allow_na_conditions=self.allow_na_conditions,
                    keep_na=local_na,
                    max_bins=max_bins)
            if self.weights_col is not None:
                if self.weights_col not in cols:
                    raise ValueError(
                        f"weights column {self.weights_col!r} missing")
                weights_np = np.asarray(cols[self.weights_col],
                                        dtype=np.float32)
        bnd = padded_boundaries(ds.dataspec.feature_columns)
        cat_feats = self._cat_feature_flags(ds)
        self._bigcat_group_of_code = self._compute_bigcat_maps(
            ds, cat_feats)
        bins = self._bin_matrix(ds.X, cat_feats, bnd, device)
        labels = None
        if ds.label_values is not None:
            labels = torch.from_numpy(
                np.ascontiguousarray(ds.label_values)).to(device)
        cat_flags = None
        if cat_feats.any():
            cat_flags = torch.from_numpy(
                cat_feats.astype(np.uint8)).to(device)
        mono = None
        if self.features is not None:
            dirs = {f.name: int(getattr(f, "monotonic", 0) or 0)
                    for f in self.features if isinstance(f, Column)}
            if any(dirs.values()):
                arr = np.zeros(len(ds.dataspec.feature_columns),
                               dtype=np.int8)
                for i, c in enumerate(ds.dataspec.feature_columns):
                    arr[i] = dirs.get(c.name, 0)
                mono = torch.from_numpy(arr).to(device)
        cw = getattr(self, "_extra_hp", {}).get("class_weights")
        if cw:
            # class weights as per-example weights (reference
            # class_weights: Dict[label value -> weight])
            if self._task != Task.CLASSIFICATION or \
                    ds.label_values is None:
                raise ValueError(
                    "class_weights requires a classification task")
            vocab = list(ds.dataspec.label_column.vocab or [])
            wmap = np.ones(max(len(vocab), 1), dtype=np.float32)
            for name, w in cw.items():
                if str(name) not in vocab:
                    raise ValueError(
                        f"class_weights key {name!r} is not a label "
                        f"class (classes: {vocab})")
                wmap[vocab.index(str(name))] = float(w)
            cls_w = wmap[np.clip(
                ds.label_values.astype(np.int64), 0, len(wmap) - 1)]
            weights_np = cls_w if weights_np is None \
                else weights_np * cls_w
        weights = None
        if weights_np is not None:
            # packed-u64 histogram path needs per-example h <= 16: scale
            # down uniformly if needed (leaf values are scale-invariant at
            # the default l2=0; documented in README)
            mx = float(weights_np.max()) if len(weights_np) else 1.0
            if mx > 8.0:
                weights_np = weights_np * (8.0 / mx)
            weights = torch.from_numpy(
                np.ascontiguousarray(weights_np)).to(device)
        return ds, bins, labels, bnd, cat_flags, weights, mono

    def _label_classes(self, ds: VerticalDataset):
        if self._task != Task.CLASSIFICATION:
            return None
        lspec = ds.dataspec.label_column
        if lspec.semantic != Semantic.CATEGORICAL:
            raise ValueError("classification label must be categorical")
        return list(lspec.vocab[1:])

    @staticmethod
    def _feature_gains(trees, names):
        """Total split gain per feature (persisted for SUM_SCORE variable
        importances; reference AbstractModel precomputed importances)."""
        g = np.zeros(len(names), dtype=np.float64)
        for t in trees:
            if t.gain is None:
                continue
            valid = t.feat >= 0
            np.add.at(g, t.feat[valid], t.gain[valid].astype(np.float64))
        return {names[i]: float(g[i]) for i in range(len(names))
                if g[i] > 0}

    def train(self, data, valid=None, verbose=None):
        raise NotImplementedError

    def cross_validation(self, data, folds: int = 10, seed: int = 1234):
        """K-fold cross validation (reference utils/fold_generator +
        AbstractLearner evaluation, abstract_learner.h:270). Returns the
        pooled out-of-fold Evaluation."""
        from ydf_amd.dataset.dataset import _to_column_dict
        from ydf_amd.metric.metric import evaluate_predictions

        cols = _to_column_dict(data)
        n = len(next(iter(cols.values())))
        rng = np.random.RandomState(seed)
        perm = rng.permutation(n)
        all_preds = None
        all_labels = np.empty(n, dtype=np.float32)
        for k in range(folds):
            test_idx = perm[k::folds]
            train_idx = np.setdiff1d(perm, test_idx)
            tr = {c: v[train_idx] for c, v in cols.items()}
            te = {c: v[test_idx] for c, v in cols.items()}
            model = self.train(tr)
            p = model.predict(te)
            if all_preds is None:
                all_preds = np.empty((n,) + p.shape[1:], dtype=np.float32)
            all_preds[test_idx] = p
            lspec = model.dataspec.label_column
            from ydf_amd.dataset.dataspec import Semantic

            if lspec.semantic == Semantic.CATEGORICAL:
                lookup = {v: i for i, v in enumerate(lspec.vocab)}
                all_labels[test_idx] = [
                    lookup.get(s, 0) - 1
                    for s in te[self.label].astype(str)]
            else:
                all_labels[test_idx] = te[self.label]
        n_classes = 2
        if model.label_classes:
            n_classes = len(model.label_classes)
        return evaluate_predictions(all_preds, all_labels, self._task,
                                    n_classes)

    def _train_with_tuner(self, data, valid=None):
        """Random-search trials; keeps the best model by validation loss
        (reference hyperparameters_optimizer.cc random trials)."""
        from ydf_amd.learner.tuner import (OptimizerLogs, TrialLog,
                                           run_parallel_trials,
                                           trial_score)

        if getattr(self.tuner, "parallel_trials", 1) > 1:
            # one trial per process slot (one per GPU when available):
            # the MI355X mapping of the reference's distributed HPO
            # (hyperparameters_optimizer.h:46 + generic_worker/)
            return run_parallel_trials(self, self.tuner, data,
                                       valid=valid)
        rng = np.random.RandomState(self.tuner.seed)
        tuner = self.tuner
        best = None
        logs = []
        base_hp = dict(self.hyperparameters)
        self.tuner = None  # avoid recursion
        try:
            for trial in range(tuner.num_trials):
                hp = tuner.sample(rng)
                self.hyperparameters = dict(base_hp)
                self.hyperparameters.update(
                    {k: v for k, v in hp.items()
                     if k in self.hyperparameters})
                if self.hyperparameters.get("validation_ratio", 0) == 0:
                    self.hyperparameters["validation_ratio"] = 0.1
                model = self.train(data, valid=valid)
                score = trial_score(
                    model, valid if valid is not None else data,
                    getattr(tuner, "optimize_metric", None))
                logs.append(TrialLog(hyperparameters=hp, score=score))
                if best is None or score > best[0]:
                    best = (score, model)
        finally:
            self.tuner = tuner
            self.hyperparameters = base_hp
        model = best[1]
        model.tuner_logs = OptimizerLogs(trials=logs)
        return model

    def validate_hyperparameters(self) -> None:
        pass


def hyperparameter_specification(learner_cls) -> dict:
    """Machine-readable hyper-parameter spec generated from the learner
    signature (reference GetGenericHyperParameterSpecification,
    abstract_learner.h:126 — the spec the reference uses to generate
    PYDF signatures and docs; here the signature IS the source and the
    spec is derived from it, eliminating signature/spec drift).

    Returns {name: {"type": "int|float|str|bool|...", "default": v}}.
    """
    import inspect

    out = {}
    for cls in reversed(learner_cls.__mro__):
        if cls is object:
            continue
        try:
            sig = inspect.signature(cls.__init__)
        except (TypeError, ValueError):
            continue
        for name, p in sig.parameters.items():
            if name in ("self", "args", "kwargs", "label", "task",
                        "features", "tuner", "device"):
                continue
            if p.kind in (inspect.Parameter.VAR_POSITIONAL,
                          inspect.Parameter.VAR_KEYWORD):
                continue
            default = None if p.default is inspect.Parameter.empty \
                else p.default
            out[name] = {
                "type": type(default).__name__
                if default is not None else "optional",
                "default": default,
            }
    return out
