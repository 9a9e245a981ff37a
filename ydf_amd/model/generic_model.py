"""GenericModel: prediction, evaluation, persistence.

Capability analogue of the reference AbstractModel
(model/abstract_model.h:63) + the PYDF GenericModel surface
(port/python/ydf/model/generic_model.py: predict:438, evaluate:552,
save:358, describe:277, benchmark:320). Serving runs through the flat-forest
HIP kernel on GPU (ydf_amd/serving) or its C++ CPU twin.
"""
from __future__ import annotations

import dataclasses
import json
import os
import time
from typing import Dict, List, Optional

import numpy as np
import torch

from ydf_amd import ops
from ydf_amd.dataset.dataset import (VerticalDataset, _to_column_dict,
                                     encode_column)
from ydf_amd.dataset.dataspec import DataSpecification, Semantic, Task
from ydf_amd.metric.metric import Evaluation, evaluate_predictions
from ydf_amd.model.forest import FlatForest


@dataclasses.dataclass
class ModelIOOptions:
    file_prefix: Optional[str] = None


_WEIGHTS_DISABLED = object()  # evaluate(weighted=False) sentinel


class ShapValues(tuple):
    """predict_shap result: the reference tuple shape
    (values: Dict[feature -> phi[N]], initial_value) that ALSO answers
    the legacy mapping form (shap["feature"], shap["__BIAS__"],
    .items()). Unpack `values, initial = model.predict_shap(data)` or
    index by feature name."""

    def __new__(cls, values, initial):
        return super().__new__(cls, (values, initial))

    def __getitem__(self, k):
        if isinstance(k, str):
            if k == "__BIAS__":
                return tuple.__getitem__(self, 1)
            return tuple.__getitem__(self, 0)[k]
        return tuple.__getitem__(self, k)

    def __contains__(self, k):
        return k == "__BIAS__" or k in tuple.__getitem__(self, 0)

    def items(self):
        yield from tuple.__getitem__(self, 0).items()
        yield "__BIAS__", tuple.__getitem__(self, 1)

    def keys(self):
        return list(tuple.__getitem__(self, 0).keys()) + ["__BIAS__"]


def default_device() -> torch.device:
    return torch.device("cuda") if torch.cuda.is_available() else \
        torch.device("cpu")


class _DeviceForest:
    """Forest arrays resident on one device (upload once, reuse)."""

    def __init__(self, forest: FlatForest, device: torch.device):
        self.feat = torch.from_numpy(forest.feat).to(device)
        self.thr = torch.from_numpy(forest.thr).to(device)
        self.left = torch.from_numpy(forest.left).to(device)
        self.roots = torch.from_numpy(forest.roots).to(device)
        if forest.has_cats:
            self.cat_idx = torch.from_numpy(forest.cat_idx).to(device)
            self.masks = torch.from_numpy(
                forest.masks.view(np.int64)).to(device)
        else:
            self.cat_idx = None
            self.masks = None
        if len(forest.obl_ranges):
            self.obl_ranges = torch.from_numpy(
                np.ascontiguousarray(forest.obl_ranges)).to(device)
            self.obl_attr = torch.from_numpy(forest.obl_attr).to(device)
            self.obl_w = torch.from_numpy(forest.obl_w).to(device)
        else:
            self.obl_ranges = self.obl_attr = self.obl_w = None
        self.na_right = torch.from_numpy(forest.na_right).to(device) \
            if forest.has_na_routing else None
        self.packed = None
        if device.type == "cuda":
            ci = self.cat_idx if self.cat_idx is not None else torch.full(
                (self.feat.numel(),), -1, dtype=torch.int32, device=device)
            self.packed = ops.pack_forest_nodes(self.feat, self.thr,
                                                self.left, ci)


class _CallableList(list):
    """list that also answers `x()` — reference-API compatibility
    (PYDF exposes label_classes/training_logs as METHODS; this
    framework historically as attributes; both forms work here)."""

    def __call__(self):
        return self


class _CallableDict(dict):
    def __call__(self):
        return self


class _CallableStr(str):
    def __call__(self):
        return self


class _CallableInt(int):
    def __call__(self):
        return self


class TrainingLogEntry(dict):
    """One training-log record. Dict form ({"iteration", "valid_loss",
    ...}) plus the reference attribute form (entry.iteration,
    entry.evaluation — PYDF generic_model.TrainingLogEntry)."""

    def __getattr__(self, k):
        if k == "evaluation":
            from ydf_amd.metric.metric import Evaluation

            ev = Evaluation(loss=self.get("valid_loss"))
            ev.custom_metrics = {kk: v for kk, v in self.items()
                                 if kk not in ("iteration", "valid_loss")}
            return ev
        try:
            return self[k]
        except KeyError:
            raise AttributeError(k) from None


class GenericModel:
    """Base decision-forest model."""

    _model_type = "GENERIC"

    def __init__(self, forest: FlatForest, dataspec: DataSpecification,
                 task: Task, label_classes: Optional[List[str]] = None,
                 init_predictions: Optional[List[float]] = None,
                 num_trees_per_iter: int = 1, activation: str = "identity",
                 metadata: Optional[Dict] = None):
        self.forest = forest
        self.dataspec = dataspec
        self._task = task
        self.label_classes = label_classes
        self.init_predictions = init_predictions or [0.0]
        self.num_trees_per_iter = num_trees_per_iter
        self.activation = activation
        self.metadata = metadata or {}
        self._dev_forest: Dict[str, _DeviceForest] = {}
        self._engine = None  # None = automatic (flat kernel)
        self.training_logs = None
        self.tuner_logs = None
        self._self_evaluation = None

    # reference-API compat: these read as attributes AND call as methods
    @property
    def label_classes(self):
        return self._label_classes

    @label_classes.setter
    def label_classes(self, v):
        self._label_classes = _CallableList(v) if isinstance(
            v, (list, tuple)) else v

    @property
    def metadata(self):
        return self._metadata

    @metadata.setter
    def metadata(self, v):
        self._metadata = _CallableDict(v) if isinstance(v, dict) else v

    @property
    def activation(self):
        return self._activation

    @activation.setter
    def activation(self, v):
        self._activation = _CallableStr(v) if isinstance(v, str) else v

    @property
    def training_logs(self):
        return self._training_logs

    @training_logs.setter
    def training_logs(self, v):
        if isinstance(v, (list, tuple)):
            v = _CallableList(
                TrainingLogEntry(e) if isinstance(e, dict) else e
                for e in v)
        self._training_logs = v

    def __getstate__(self):
        # models pickle (reference PYDF __getstate__/__setstate__ via
        # serialized blob); device/engine caches are rebuilt lazily
        d = dict(self.__dict__)
        d["_dev_forest"] = {}
        return d

    def __setstate__(self, d):
        self.__dict__.update(d)
        self._dev_forest = {}

    def set_data_spec(self, data_spec) -> None:
        """Replaces the dataspec (PYDF model.set_data_spec); engine
        caches are invalidated because thresholds/vocabularies may no
        longer match the packed tables."""
        self.dataspec = data_spec
        self._dev_forest.clear()
        if hasattr(self, "_thr_on_cuts"):
            del self._thr_on_cuts

    def set_node_format(self, node_format: str) -> None:
        """On-disk node container format (PYDF model.set_node_format).
        Only BLOB_SEQUENCE (the reference default) is written."""
        if str(node_format).upper() not in ("BLOB_SEQUENCE",):
            raise ValueError(
                f"unsupported node format {node_format!r}; the exporter "
                "writes BLOB_SEQUENCE (reference blob_sequence.h)")
        self.metadata["node_format"] = str(node_format).upper()

    # ------------------------------------------------------------------
    def task(self) -> Task:
        return self._task

    def name(self) -> str:
        return self._model_type

    def label(self) -> Optional[str]:
        return self.dataspec.label

    def label_col_idx(self) -> int:
        for i, c in enumerate(self.dataspec.columns):
            if c.name == self.dataspec.label:
                return i
        return -1

    def input_feature_names(self) -> List[str]:
        return [c.name for c in self.dataspec.feature_columns]

    def input_features(self):
        """[(name, semantic)] like PYDF model.input_features()."""
        return [(c.name, c.semantic) for c in self.dataspec.feature_columns]

    def data_spec(self):
        return self.dataspec

    def to_tensorflow_saved_model(self, *a, **k):
        raise ImportError(
            "TensorFlow is not available in this environment; use "
            "to_cpp()/to_java()/to_docker() for deployment, or "
            "export_ydf_model() for the reference on-disk format")

    def to_jax_function(self, *a, **k):
        raise ImportError(
            "JAX is not available in this environment; use to_cpp()/"
            "to_java()/to_docker() for deployment")

    def to_tensorflow_function(self, *a, **k):
        raise ImportError(
            "TensorFlow is not available in this environment; use "
            "to_cpp()/to_java()/to_js() for embedding")

    def update_with_jax_params(self, *a, **k):
        raise ImportError("JAX is not available in this environment")

    # -- PYDF method-surface parity (thin delegates) -------------------
    def predict_class(self, data, device=None) -> np.ndarray:
        """Most likely class name per example (PYDF
        model.predict_class; classification only)."""
        if self._task != Task.CLASSIFICATION or not self.label_classes:
            raise ValueError("predict_class requires a classification "
                             "model")
        p = self.predict(data, device=device)
        names = np.asarray(self.label_classes)
        if p.ndim == 1:
            return names[(p >= 0.5).astype(np.int64)]
        return names[p.argmax(axis=1)]

    def serialize(self) -> bytes:
        """In-memory serialization (PYDF model.serialize; pair with
        ydf.deserialize_model)."""
        from ydf_amd.model.model_lib import serialize_model

        return serialize_model(self)

    def to_cpp(self, key: str = "my_model") -> str:
        from ydf_amd.serving.embed import to_cpp

        return to_cpp(self, key)

    def to_standalone_cc(self, name: str = "ydf_model",
                         algorithm: str = "ROUTING") -> str:
        from ydf_amd.serving.embed import to_cpp

        return to_cpp(self, name, algorithm=algorithm)

    def to_standalone_java(self, class_name: str = "YdfModel") -> str:
        from ydf_amd.serving.embed import to_java

        return to_java(self, class_name)

    def to_docker(self, path: str, **kwargs) -> None:
        from ydf_amd.serving.deploy import to_docker

        to_docker(self, path, **kwargs)

    def input_features_col_idxs(self):
        label = self.dataspec.label
        return [i for i, c in enumerate(self.dataspec.columns)
                if c.name != label]

    def hyperparameter_optimizer_logs(self):
        """Tuning trial logs when the model came from a tuner (PYDF
        model.hyperparameter_optimizer_logs)."""
        return self.tuner_logs

    def set_metadata(self, metadata) -> None:
        self.metadata = dict(metadata) if metadata else {}

    def set_feature_selection_logs(self, logs) -> None:
        self._feature_selection_logs = logs

    def feature_selection_logs(self):
        return getattr(self, "_feature_selection_logs", None)

    def num_trees(self) -> int:
        return self.forest.n_trees

    def num_nodes(self) -> int:
        return self.forest.n_nodes

    def self_evaluation(self):
        """Out-of-bag evaluation (RF) or last validation evaluation (GBT);
        mirrors ydf model.self_evaluation (reference OOB evaluations,
        random_forest.cc:557 / GBT validation logs)."""
        if self._self_evaluation is not None:
            return self._self_evaluation
        if self.training_logs:
            from ydf_amd.metric.metric import Evaluation

            return Evaluation(loss=self.training_logs[-1].get("valid_loss"))
        return None

    # ------------------------------------------------------------------
    def _encode_features(self, data) -> np.ndarray:
        """data -> feature-major [F,N] float32 matrix per the dataspec."""
        if isinstance(data, VerticalDataset):
            return data.X
        cols = _to_column_dict(data)
        specs = self.dataspec.feature_columns
        n = len(next(iter(cols.values()))) if cols else 0
        X = np.empty((len(specs), n), dtype=np.float32)
        keep_na = self.forest.has_na_routing or (
            (self.metadata or {}).get("missing_value_policy")
            == "LOCAL_IMPUTATION")
        has_vecseq = False
        for i, spec in enumerate(specs):
            if spec.vecseq_source is not None:
                has_vecseq = True
                continue
            if spec.semantic == Semantic.NUMERICAL_VECTOR_SEQUENCE:
                # ragged source column (imported reference models keep
                # it in the spec): its values feed the virtual
                # projection columns, not a dense X row
                X[i] = 0.0
                continue
            src = spec.set_source or spec.name
            if src not in cols:
                raise ValueError(f"missing input feature {src!r}")
            X[i] = encode_column(cols[src], spec, keep_na=keep_na)
        if has_vecseq:
            from ydf_amd.dataset.vecseq import fill_vecseq_columns

            fill_vecseq_columns(X, specs, cols)
        return X

    def list_compatible_engines(self):
        """Names of serving engines usable by this model (reference
        ListCompatibleFastEngines; mirrors PYDF
        model.list_compatible_fast_engines)."""
        out = ["flat"]
        pure_numerical = (len(self.forest.masks) == 0
                          and len(self.forest.obl_ranges) == 0
                          and not self.forest.has_na_routing)
        if pure_numerical:
            out.append("8bit")
            if self._thresholds_on_cuts():
                # compact-node engine: exact only when every split
                # threshold sits on a training cut (always true for
                # models trained here; imported models may not)
                out.append("binned8")
            # QuickScorer needs <= 64 leaves per tree
            f = self.forest
            ok = True
            for t in range(f.n_trees):
                lo, hi = f.tree_slice(t)
                if int((f.feat[lo:hi] < 0).sum()) > 64:
                    ok = False
                    break
            if ok:
                out.append("qs")
        return out

    def _thresholds_on_cuts(self) -> bool:
        cached = getattr(self, "_thr_on_cuts", None)
        if cached is not None:
            return cached
        from ydf_amd.model.forest import padded_boundaries

        try:
            bnd = padded_boundaries(self.dataspec.feature_columns)
        except ValueError:
            self._thr_on_cuts = False
            return False
        f = self.forest
        internal = f.feat >= 0
        feats = f.feat[internal]
        thrs = f.thr[internal]
        ok = True
        for fi in np.unique(feats):
            cuts = bnd[int(fi)]
            t = thrs[feats == fi]
            idx = np.searchsorted(cuts, t)
            valid = idx < len(cuts)
            if not valid.all() or not np.array_equal(
                    cuts[idx[valid]], t[valid]):
                ok = False
                break
        self._thr_on_cuts = bool(ok)
        return self._thr_on_cuts

    def force_engine(self, name) -> None:
        """Pins the serving engine used by predict() on GPU
        (mirrors PYDF model.force_engine; None = automatic = flat)."""
        if name is not None and name not in self.list_compatible_engines():
            raise ValueError(
                f"engine {name!r} not compatible; options: "
                f"{self.list_compatible_engines()}")
        self._engine = name
        self._dev_forest.clear()

    def _forest_on(self, device: torch.device) -> _DeviceForest:
        key = str(device)
        if key not in self._dev_forest:
            self._dev_forest[key] = _DeviceForest(self.forest, device)
        return self._dev_forest[key]

    def _n_outputs(self) -> int:
        if (self._task == Task.CLASSIFICATION and self.label_classes
                and len(self.label_classes) > 2):
            return len(self.label_classes)
        return 1

    def _leaf_scale(self) -> float:
        return 1.0

    def predict_margin(self, X: torch.Tensor) -> torch.Tensor:
        """Raw per-output forest sums/means. X [F,N] f32 on any device."""
        eng = getattr(self, "_engine", None)
        if eng is None and X.is_cuda \
                and len(self.forest.masks) == 0 \
                and len(self.forest.obl_ranges) == 0 \
                and not self.forest.has_na_routing \
                and self._thresholds_on_cuts():
            # auto-select: the compact-node binned engine is ~2x the
            # flat engine on large batches (half the L2 node traffic,
            # profiles/serving_engines_r02.md) and bit-equivalent when
            # thresholds sit on training cuts; small batches route to
            # its tree-parallel grid (binned4_tp) instead of serially
            # walking the whole forest per thread
            eng = "binned8"
        if eng == "binned8" and X.is_cuda:
            return self._predict_margin_binned8(X)
        if eng in ("qs", "8bit") and X.is_cuda \
                and self._n_outputs() == 1:
            return self._predict_margin_engine(X, eng)
        df = self._forest_on(X.device)
        C = self._n_outputs()
        N = X.shape[1]
        out = torch.empty((C, N), dtype=torch.float32, device=X.device)
        T = self.forest.n_trees
        per = T // C if C > 1 else T
        for c in range(C):
            ops.predict_forest(X, df.feat, df.thr, df.left, df.roots,
                               out[c], tree_start=c, tree_step=C,
                               n_trees=per if C > 1 else T,
                               init=float(self.init_predictions[c]
                                          if c < len(self.init_predictions)
                                          else self.init_predictions[0]),
                               scale=self._leaf_scale(),
                               cat_idx=df.cat_idx, masks=df.masks,
                               packed=df.packed, obl_ranges=df.obl_ranges,
                               obl_attr=df.obl_attr, obl_w=df.obl_w,
                               na_right=df.na_right)
        return out

    def _predict_margin_engine(self, X: torch.Tensor,
                               eng: str) -> torch.Tensor:
        from ydf_amd.model.forest import (build_quickscorer,
                                          pack_binned_nodes,
                                          padded_boundaries)

        dev = X.device
        key = f"{eng}:{dev}"
        cache = self._dev_forest
        N = X.shape[1]
        out = torch.empty((1, N), dtype=torch.float32, device=dev)
        init = float(self.init_predictions[0])
        scale = self._leaf_scale()
        if eng == "qs":
            if key not in cache:
                conds, offs, lv = build_quickscorer(self.forest)
                cache[key] = (torch.from_numpy(conds).to(dev),
                              torch.from_numpy(offs).to(dev),
                              torch.from_numpy(lv).to(dev))
            c, o, lv = cache[key]
            ops.predict_forest_qs(X, c, o, lv, out[0], init=init,
                                  scale=scale)
            return out
        if key not in cache:
            bnd = padded_boundaries(self.dataspec.feature_columns)
            cache[key] = (
                torch.from_numpy(
                    pack_binned_nodes(self.forest, bnd)).to(dev),
                torch.from_numpy(self.forest.roots).to(dev),
                torch.from_numpy(bnd).to(dev))
        packed, roots, bnd_t = cache[key]
        bins = torch.empty(X.shape, dtype=torch.uint8, device=dev)
        ops.bin_data(X, bnd_t, bins)
        ops.predict_forest_binned(bins, packed, roots, out[0], init=init,
                                  scale=scale)
        return out

    def _predict_margin_binned8(self, X: torch.Tensor) -> torch.Tensor:
        """Compact-node binned engine (multi-output capable: class
        trees stride through the shared node table)."""
        from ydf_amd.model.forest import (pack_binned8_nodes,
                                          padded_boundaries)

        from ydf_amd.model.forest import pack_binned4_nodes

        dev = X.device
        key = f"binned8:{dev}"
        cache = self._dev_forest
        if key not in cache:
            bnd = padded_boundaries(self.dataspec.feature_columns)
            try:
                # 4-byte nodes when the forest fits the packing limits
                # (F < 63, < 2^18 nodes/leaves) — measured slightly
                # faster than the 8-byte form
                n4, lv4 = pack_binned4_nodes(
                    self.forest, bnd, leaf_scale=self._leaf_scale())
                packed = (torch.from_numpy(n4).to(dev),
                          torch.from_numpy(lv4).to(dev))
            except ValueError:
                packed = (torch.from_numpy(pack_binned8_nodes(
                    self.forest, bnd,
                    leaf_scale=self._leaf_scale())).to(dev),)
            cache[key] = (packed,
                          torch.from_numpy(self.forest.roots).to(dev),
                          torch.from_numpy(bnd).to(dev))
        packed, roots, bnd_t = cache[key]
        C = self._n_outputs()
        N = X.shape[1]
        out = torch.empty((C, N), dtype=torch.float32, device=dev)
        bins = torch.empty(X.shape, dtype=torch.uint8, device=dev)
        ops.bin_data(X, bnd_t, bins)
        T = self.forest.n_trees
        for c in range(C):
            init_c = float(self.init_predictions[c]
                           if c < len(self.init_predictions)
                           else self.init_predictions[0])
            if len(packed) == 2:
                ops.predict_forest_binned4(
                    bins, packed[0], packed[1], roots, out[c],
                    init=init_c, tree_start=c,
                    tree_step=C if C > 1 else 1,
                    n_trees=T // C if C > 1 else T)
            else:
                ops.predict_forest_binned8(
                    bins, packed[0], roots, out[c], init=init_c,
                    tree_start=c, tree_step=C if C > 1 else 1,
                    n_trees=T // C if C > 1 else T)
        return out

    def predict(self, data, device=None, *, use_slow_engine=False,
                num_threads=None) -> np.ndarray:
        """Predictions as numpy: binary classification -> P(class_2) [N];
        multi-class -> [N, C]; regression/anomaly -> [N]. (Mirrors
        ydf GenericModel.predict semantics; use_slow_engine forces the
        generic CPU path instead of the packed GPU engines, and
        num_threads bounds the CPU op parallelism.)"""
        if num_threads is not None:
            torch.set_num_threads(int(num_threads))
        if use_slow_engine:
            device = "cpu"
        dev = torch.device(device) if device is not None else default_device()
        from ydf_amd.utils import usage

        if self.forest.has_set_conditions:
            # categorical-set conditions: ragged token inputs — walks
            # run on host (reference CategoricalSetContains engines)
            m = self._margin_set_model(data)
            usage.on_inference(m.shape[1])
            return self._apply_activation(m).cpu().numpy()
        X_np = self._encode_features(data)
        usage.on_inference(X_np.shape[1] if X_np.ndim == 2 else len(X_np))
        X = torch.from_numpy(np.ascontiguousarray(X_np)).to(dev)
        m = self.predict_margin(X)
        out = self._apply_activation(m)
        return out.cpu().numpy()

    def _margin_set_model(self, data) -> torch.Tensor:
        """Host tree walk for models with categorical-SET conditions
        (reference ContainsVector over CATEGORICAL_SET columns): cells
        are token sets; a set node goes right iff the cell intersects
        its item list. Vectorized per node over the rows reaching it."""
        cols = _to_column_dict(data)
        specs = self.dataspec.feature_columns
        f = self.forest
        n = len(next(iter(cols.values()))) if cols else 0
        dense = {}
        setcol = {}
        for i, spec in enumerate(specs):
            if spec.semantic == Semantic.CATEGORICAL_SET:
                lut = {v: j for j, v in enumerate(spec.vocab or [])}

                def to_set(cell, lut=lut):
                    if isinstance(cell, (list, tuple, set, frozenset,
                                         np.ndarray)):
                        toks = [str(t) for t in cell]
                    else:
                        toks = str(cell).split(" ")
                    return frozenset(lut.get(t, 0) for t in toks if t)

                setcol[i] = [to_set(c) for c in cols[spec.name]]
            else:
                dense[i] = encode_column(
                    cols[spec.name], spec,
                    keep_na=self.forest.has_na_routing)
        C = self._n_outputs()
        out = np.zeros((C, n), dtype=np.float64)
        offs, items = f.set_offs, f.set_items
        scale = self._leaf_scale()

        def rec(node, idx, acc):
            fi = int(f.feat[node])
            if fi < 0:
                acc[idx] += float(f.thr[node])
                return
            si = int(f.set_idx[node])
            ci = int(f.cat_idx[node])
            if si >= 0:
                cond_items = items[offs[si]:offs[si + 1]]
                if fi in setcol:
                    cond = frozenset(int(v) for v in cond_items)
                    col = setcol[fi]
                    right = np.fromiter(
                        (not cond.isdisjoint(col[i]) for i in idx),
                        bool, count=len(idx))
                else:
                    # plain CATEGORICAL feature with a full-dictionary
                    # set condition (large-vocab training): code in set
                    codes = dense[fi][idx].astype(np.int64)
                    right = np.isin(codes, np.asarray(cond_items,
                                                      dtype=np.int64))
            elif ci >= 0:
                cb = dense[fi][idx].astype(np.int64).clip(0, 255)
                right = (f.masks[ci][cb >> 6]
                         >> (cb & 63).astype(np.uint64)) & np.uint64(1)
                right = right.astype(bool)
            else:
                xv = dense[fi][idx]
                right = xv > f.thr[node]
                if f.has_na_routing:
                    nanm = np.isnan(xv)
                    if nanm.any():
                        right = np.where(nanm, bool(f.na_right[node]),
                                         right)
            left = int(f.left[node])
            rec(left, idx[~right], acc)
            rec(left + 1, idx[right], acc)

        T = f.n_trees
        per = T // C if C > 1 else T
        all_idx = np.arange(n)
        for c in range(C):
            acc = out[c]
            acc += float(self.init_predictions[c]
                         if c < len(self.init_predictions)
                         else self.init_predictions[0])
            for t in range(c, T, C) if C > 1 else range(T):
                rec(int(f.roots[t]), all_idx, acc)
            init = float(self.init_predictions[c]
                         if c < len(self.init_predictions)
                         else self.init_predictions[0])
            out[c] = init + (acc - init) * scale
        return torch.from_numpy(out.astype(np.float32))

    def _apply_activation(self, m: torch.Tensor) -> torch.Tensor:
        if self.activation == "sigmoid":
            out = torch.empty_like(m[0])
            ops.sigmoid(m[0], out)
            return out
        if self.activation == "softmax":
            return torch.softmax(m, dim=0).T.contiguous()
        if self.activation == "exp":  # Poisson log link
            return torch.exp(m[0].clamp(max=30))
        if m.shape[0] == 1:
            return m[0]
        return m.T.contiguous()

    # ------------------------------------------------------------------
    def evaluate(self, data, device=None, weights=None, *,
                 weighted=None, task=None, label=None, group=None,
                 bootstrapping=False, ndcg_truncation=5,
                 mrr_truncation=5, map_truncation=5,
                 use_slow_engine=False,
                 num_threads=None) -> Evaluation:
        """`weights` names a column in `data` (or passes an array) for
        example-weighted metrics; defaults to the training weights
        column if the model recorded one. Reference keyword surface:
        weighted=False disables the recorded weights; task/label/group
        override the evaluation task, label column and ranking group;
        bootstrapping=True|n attaches percentile-bootstrap CIs
        (ev.bootstrap_cis); *_truncation set the ranking metric
        cutoffs."""
        if weighted is False:
            weights = _WEIGHTS_DISABLED
        task_override = task
        cols = _to_column_dict(data) if not isinstance(data, VerticalDataset) \
            else None
        preds = self.predict(data, device=device,
                             use_slow_engine=use_slow_engine,
                             num_threads=num_threads)
        if isinstance(data, VerticalDataset):
            labels = data.label_values
        else:
            lname = label if label is not None else self.dataspec.label
            if lname is None or lname not in cols:
                raise ValueError("dataset has no label column")
            use_spec = label is None or label == self.dataspec.label
            if use_spec and self.dataspec.label_column.semantic \
                    == Semantic.CATEGORICAL:
                lspec = self.dataspec.label_column
                lookup = {item: i for i, item in enumerate(lspec.vocab)}
                labels = np.fromiter(
                    (lookup.get(s, 0) - 1 for s in cols[lname].astype(str)),
                    dtype=np.float32, count=len(cols[lname]))
            else:
                arr = np.asarray(cols[lname])
                if arr.dtype.kind in "USO":
                    # overridden label with string classes: map through
                    # the model's class order
                    lk = {c: i for i, c in enumerate(
                        self.label_classes or [])}
                    labels = np.fromiter(
                        (lk.get(s, 0) for s in arr.astype(str)),
                        dtype=np.float32, count=len(arr))
                else:
                    labels = arr.astype(np.float32)
        w = None
        if weights is _WEIGHTS_DISABLED:
            weights = None
        elif weights is None:
            weights = (self.metadata or {}).get("weights_column")
        if weights is not None and cols is not None:
            w = np.asarray(cols[weights], np.float64) \
                if isinstance(weights, str) and weights in cols \
                else (np.asarray(weights, np.float64)
                      if not isinstance(weights, str) else None)
        n_classes = len(self.label_classes) if self.label_classes else 2
        etask = task_override if task_override is not None else self._task
        if etask == Task.SURVIVAL_ANALYSIS:
            from ydf_amd.learner.survival import CoxData
            from ydf_amd.metric.survival import concordance_index

            ecol = (self.metadata or {}).get("label_event_observed")
            if cols is None or ecol not in cols:
                raise ValueError(
                    f"survival evaluation needs the event column "
                    f"{ecol!r} in the dataset")
            events = np.asarray(cols[ecol]).astype(bool)
            acol = (self.metadata or {}).get("label_entry_age")
            entry = np.asarray(cols[acol], np.float64) \
                if acol and acol in cols else None
            ev = Evaluation(num_examples=len(labels))
            ev.cindex = concordance_index(labels, events, preds)
            cd = CoxData(labels, events, torch.device("cpu"), entry)
            ev.loss = cd.loss(torch.from_numpy(
                np.asarray(preds, np.float32)))
            return ev
        if etask in (Task.CATEGORICAL_UPLIFT, Task.NUMERICAL_UPLIFT):
            from ydf_amd.metric.uplift import auuc_qini

            tcol = (self.metadata or {}).get("uplift_treatment")
            if cols is None or tcol not in cols:
                raise ValueError(
                    f"uplift evaluation needs the treatment column "
                    f"{tcol!r} in the dataset")
            treat = np.asarray(cols[tcol])
            if treat.dtype.kind in "UOS":
                tvocab = (self.metadata or {}).get("treatment_vocab")
                pos = tvocab[1] if tvocab and len(tvocab) > 1 else None
                treat = (treat.astype(str) == pos).astype(np.float32) \
                    if pos is not None else (treat != treat[0]).astype(
                        np.float32)
            ev = Evaluation(num_examples=len(labels))
            ev.auuc, ev.qini = auuc_qini(labels, treat, preds)
            return ev
        ev = evaluate_predictions(preds, labels, etask, n_classes,
                                  weights=w)
        if etask == Task.CLASSIFICATION and self.label_classes:
            ev.classes = tuple(self.label_classes)
        if etask == Task.RANKING:
            gcol = group if group is not None else \
                (self.metadata or {}).get("ranking_group")
            if gcol and cols is not None and gcol in cols:
                from ydf_amd.metric.metric import mean_average_precision
                from ydf_amd.metric.metric import mrr as mrr_fn
                from ydf_amd.metric.metric import ndcg as ndcg_fn

                g = np.asarray(cols[gcol])
                nt = ndcg_truncation if ndcg_truncation != 5 else \
                    (self.metadata or {}).get("ndcg_truncation", 5)
                ev.ndcg = ndcg_fn(labels, preds, g, truncation=nt)
                ev.mrr = mrr_fn(labels, preds, g,
                                truncation=mrr_truncation)
                ev.map = mean_average_precision(
                    labels, preds, g, truncation=map_truncation)
                ev.loss = -ev.ndcg
        if bootstrapping:
            from ydf_amd.metric.metric import (
                bootstrap_confidence_intervals)

            n_boot = 2000 if bootstrapping is True \
                else max(int(bootstrapping), 10)
            ev.bootstrap_cis = bootstrap_confidence_intervals(
                labels, preds, etask, n_samples=n_boot)
        return ev

    # ------------------------------------------------------------------
    def predict_shap(self, data, *, num_threads=None):
        """Path-dependent TreeSHAP values (reference utils/shap.h:83;
        mirrors ydf model.predict_shap). Returns {feature: phi [N]} plus
        the expected value under "__BIAS__"; margins (pre-activation)
        for classification. CPU implementation."""
        from ydf_amd import ops as _ops
        from ydf_amd._ydf_ops import (cpu_forest_expected_value,
                                      cpu_tree_shap)

        if self._n_outputs() > 1:
            raise NotImplementedError(
                "predict_shap supports single-output models for now")
        if not (self.forest.cover > 0).any():
            raise ValueError("model has no node covers (old format?)")
        X = np.ascontiguousarray(self._encode_features(data))
        F, N = X.shape
        phi = np.zeros((N, F + 1), dtype=np.float32)
        f = self.forest
        feat = np.ascontiguousarray(f.feat)
        thr = np.ascontiguousarray(f.thr)
        left = np.ascontiguousarray(f.left)
        cover = np.ascontiguousarray(f.cover)
        roots = np.ascontiguousarray(f.roots)
        cat_idx = np.ascontiguousarray(f.cat_idx)
        masks = np.ascontiguousarray(f.masks)
        scale = self._leaf_scale()
        obl_ranges = np.ascontiguousarray(f.obl_ranges)
        obl_attr = np.ascontiguousarray(f.obl_attr)
        obl_w = np.ascontiguousarray(f.obl_w)
        na_arr = np.ascontiguousarray(f.na_right)
        cpu_tree_shap(X.ctypes.data, N, F, feat.ctypes.data, thr.ctypes.data,
                      left.ctypes.data,
                      cat_idx.ctypes.data if f.has_cats else 0,
                      masks.ctypes.data if f.has_cats else 0,
                      obl_ranges.ctypes.data, obl_attr.ctypes.data,
                      obl_w.ctypes.data,
                      na_arr.ctypes.data if f.has_na_routing else 0,
                      cover.ctypes.data, roots.ctypes.data, 0, 1,
                      f.n_trees, scale, 0.0, phi.ctypes.data)
        ev = cpu_forest_expected_value(
            feat.ctypes.data, thr.ctypes.data, left.ctypes.data,
            cover.ctypes.data, roots.ctypes.data, 0, 1, f.n_trees, scale)
        phi[:, F] = float(self.init_predictions[0]) + ev
        if num_threads is not None:
            torch.set_num_threads(int(num_threads))
        out = {name: phi[:, i]
               for i, name in enumerate(self.input_feature_names())}
        return ShapValues(out, phi[:, F])

    def analyze_prediction(self, single_example) -> Dict:
        """Per-example prediction analysis (mirrors PYDF
        model.analyze_prediction, generic_model.py:674 family): TreeSHAP
        feature attributions for one (or a few) example(s). Returns
        {feature: contribution} plus "__BIAS__"."""
        shap = self.predict_shap(single_example)
        return {k: (float(v[0]) if len(v) == 1 else v)
                for k, v in shap.items()}

    def variable_importances(self) -> Dict:
        """Structure-based variable importances (reference
        AbstractModel::GetVariableImportance; SUM_SCORE requires a model
        trained by this framework — it is persisted in the header)."""
        from ydf_amd.utils import analysis as analysis_lib

        return analysis_lib.structure_importances(self)

    def analyze(self, data, sampling: float = 1.0,
                num_bins: int = 20,
                partial_dependence_plot: bool = True,
                conditional_expectation_plot: bool = True,
                permutation_variable_importance: bool = True,
                shap_values: bool = True,
                permutation_variable_importance_rounds: int = 1,
                num_threads: Optional[int] = None,
                maximum_duration: Optional[float] = 20,
                features: Optional[List[str]] = None, device=None):
        """Full model analysis: variable importances (+SHAP summary) +
        PDPs + CEPs (mirrors ydf model.analyze; reference
        utils/model_analysis.h:36-89). `sampling` subsamples the rows;
        maximum_duration is accepted for API parity (the GPU analysis
        path finishes well inside it for the supported sizes)."""
        from ydf_amd.utils import analysis as analysis_lib

        if num_threads is not None:
            torch.set_num_threads(int(num_threads))
        if sampling < 1.0:
            cols0 = _to_column_dict(data) \
                if not isinstance(data, VerticalDataset) else None
            if cols0 is not None:
                n0 = len(next(iter(cols0.values())))
                rng = np.random.RandomState(1234)
                keep = rng.rand(n0) < sampling
                data = {k: np.asarray(v)[keep] for k, v in cols0.items()}
        labels = None
        if self.dataspec.label is not None:
            try:
                cols = _to_column_dict(data) \
                    if not isinstance(data, VerticalDataset) else None
                if isinstance(data, VerticalDataset):
                    labels = data.label_values
                elif self.dataspec.label in cols:
                    lspec = self.dataspec.label_column
                    if lspec.semantic == Semantic.CATEGORICAL:
                        lookup = {v: i for i, v in enumerate(lspec.vocab)}
                        labels = np.fromiter(
                            (lookup.get(s, 0) - 1
                             for s in cols[self.dataspec.label].astype(str)),
                            dtype=np.float32,
                            count=len(cols[self.dataspec.label]))
                    else:
                        labels = np.asarray(cols[self.dataspec.label],
                                            dtype=np.float32)
            except Exception:
                labels = None
        return analysis_lib.analyze(
            self, data, labels=labels,
            permutation_variable_importance=(
                permutation_variable_importance
                and permutation_variable_importance_rounds > 0),
            partial_dependence=partial_dependence_plot,
            conditional_expectation=conditional_expectation_plot,
            shap_values=shap_values,
            permutation_rounds=permutation_variable_importance_rounds,
            features=features, num_grid_points=num_bins, device=device)

    def serving_session(self, batch_size: int, device=None):
        """Graph-captured repeated inference at a fixed batch size —
        the MI355X answer to the reference's latency-critical serving
        engines: the eager bin+walk launch/sync overhead (~1 ms/batch
        measured at small batches, profiles/serving_engines_r02.md) is
        replaced by ONE hipGraph replay. Returns a ServingSession with
        .predict(data) for repeated same-shape batches."""
        return ServingSession(self, batch_size, device=device)

    # ------------------------------------------------------------------
    def benchmark(self, data, benchmark_duration: float = 3.0,
                  warmup_duration: float = 0.5, batch_size: int = 0,
                  device=None):
        """Inference throughput benchmark (mirrors ydf model.benchmark;
        reference cli/benchmark_inference.cc)."""
        dev = torch.device(device) if device is not None else default_device()
        from ydf_amd.utils import usage

        X_np = self._encode_features(data)
        usage.on_inference(X_np.shape[1] if X_np.ndim == 2 else len(X_np))
        X = torch.from_numpy(np.ascontiguousarray(X_np)).to(dev)
        n = X.shape[1]

        def run_once():
            m = self.predict_margin(X)
            self._apply_activation(m)

        t_end = time.perf_counter() + warmup_duration
        while time.perf_counter() < t_end:
            run_once()
        if dev.type == "cuda":
            torch.cuda.synchronize()
        runs = 0
        t0 = time.perf_counter()
        t_end = t0 + benchmark_duration
        while time.perf_counter() < t_end:
            run_once()
            runs += 1
        if dev.type == "cuda":
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        return BenchmarkResult(
            examples_per_second=n * runs / dt,
            num_runs=runs,
            duration_seconds=dt,
            batch_size=n,
        )

    # ------------------------------------------------------------------
    def describe(self, output_format: str = "auto",
                 full_details: bool = False):
        """Model card (reference model/describe.{h,cc}; ydf
        model.describe()). output_format: "auto" (text on a terminal),
        "text", "html", or "notebook" (object with _repr_html_);
        full_details appends hyperparameter metadata and the first
        tree."""
        if output_format == "auto":
            output_format = "text"
        lines = [
            f'type: "{self._model_type}"',
            f"task: {self._task.name}",
            f"label: {self.dataspec.label!r}",
            f"input features ({len(self.input_feature_names())}): "
            + ", ".join(self.input_feature_names()[:40]),
            f"trees: {self.num_trees()}",
            f"nodes: {self.num_nodes()}",
        ]
        if self.label_classes:
            lines.append(f"classes: {self.label_classes}")
        if self.num_trees():
            sizes = np.diff(np.append(self.forest.roots,
                                      self.forest.n_nodes))
            lines.append(f"nodes per tree: min={int(sizes.min())} "
                         f"mean={float(sizes.mean()):.1f} "
                         f"max={int(sizes.max())}")
        vi = self.variable_importances()
        top = (vi.get("SUM_SCORE") or vi.get("NUM_NODES") or [])[:8]
        if top:
            lines.append("top features (" +
                         ("SUM_SCORE" if "SUM_SCORE" in vi else "NUM_NODES")
                         + "): " + ", ".join(f"{n} ({s:.3g})"
                                             for s, n in top))
        if self.training_logs:
            last = self.training_logs[-1]
            lines.append(f"training: {len(self.training_logs)} iterations, "
                         f"final validation loss "
                         f"{last.get('valid_loss'):.6g}")
        ev = self.self_evaluation()
        if ev is not None and getattr(ev, "accuracy", None) is not None:
            lines.append(f"self evaluation (OOB/validation): "
                         f"accuracy={ev.accuracy:.4f}")
        if self.tuner_logs is not None and self.tuner_logs.trials:
            bt = self.tuner_logs.best_trial
            lines.append(
                f"tuning: {len(self.tuner_logs.trials)} trials, best "
                f"score {bt.score:.6g}, best hyperparameters "
                + ", ".join(f"{k}={v}"
                            for k, v in bt.hyperparameters.items()))
        if full_details:
            if self.metadata:
                lines.append("metadata: " + ", ".join(
                    f"{k}={v}" for k, v in sorted(self.metadata.items())
                    if not isinstance(v, (dict, list))))
            if self.num_trees() and hasattr(self, "print_tree"):
                lines.append("tree 0:")
                lines.append(self.print_tree(0))
        if output_format in ("html", "notebook"):
            rows = "".join(f"<tr><td>{ln.split(':', 1)[0]}</td>"
                           f"<td>{ln.split(':', 1)[1] if ':' in ln else ''}"
                           "</td></tr>" for ln in lines)
            html = f"<h2>{self._model_type}</h2><table>{rows}</table>"
            if self.training_logs:
                pts = [(d['iteration'], d.get('valid_loss'))
                       for d in self.training_logs
                       if d.get('valid_loss') is not None]
                if pts:
                    xs = [p[0] for p in pts]
                    ys = [p[1] for p in pts]
                    w, h = 480, 160
                    ymin, ymax = min(ys), max(ys)
                    yr = (ymax - ymin) or 1.0
                    poly = " ".join(
                        f"{10 + (x - xs[0]) / max(xs[-1] - xs[0], 1) * (w - 20):.1f},"
                        f"{h - 10 - (y - ymin) / yr * (h - 20):.1f}"
                        for x, y in pts)
                    html += (f'<h3>validation loss</h3>'
                             f'<svg width="{w}" height="{h}">'
                             f'<polyline fill="none" stroke="steelblue" '
                             f'points="{poly}"/></svg>')
            if output_format == "notebook":
                class _Display(str):
                    def _repr_html_(self):
                        return str(self)

                return _Display(html)
            return html
        return "\n".join(lines)

    def _repr_html_(self) -> str:
        return self.describe(output_format="html")

    def __str__(self) -> str:
        return self.describe()

    # ------------------------------------------------------------------
    def _header(self) -> dict:
        return {
            "framework": "ydf_amd",
            "version": 1,
            "model_type": self._model_type,
            "task": self._task.name,
            "label_classes": self.label_classes,
            "init_predictions": [float(v) for v in self.init_predictions],
            "num_trees_per_iter": self.num_trees_per_iter,
            "activation": self.activation,
            "metadata": self.metadata,
            "training_logs": self.training_logs,
        }

    def _load_extra(self, header: dict) -> None:
        pass

    def save(self, path: str, advanced_options: ModelIOOptions = None,
             format: str = "ydf") -> None:
        """Saves the model directory.

        Default format="ydf": the REFERENCE on-disk layout
        (model/model_library.cc:92-107: header.pb / data_spec.pb /
        <family>_header.pb / nodes-00000-of-00001 / done) readable by
        the reference C++/Go/JS/TF-DF loaders, plus `extra.json`
        carrying state the reference format has no field for (training
        logs, framework metadata, exact activation) which the reference
        reader ignores. Models the layout cannot express (multi-class
        RF, uplift, survival, custom losses) fall back to the npz
        container with a log line. format="npz" forces the fast npz
        container (used for training snapshots)."""
        if format == "ydf":
            import shutil
            import tempfile

            try:
                from ydf_amd.model.export_ydf import export_ydf_model

                os.makedirs(path, exist_ok=True)
                tmp = tempfile.mkdtemp(
                    dir=os.path.dirname(os.path.abspath(path)) or ".")
                try:
                    export_ydf_model(self, tmp)
                    with open(os.path.join(tmp, "extra.json"), "w") as f:
                        json.dump(self._header(), f, indent=1)
                    for name in os.listdir(tmp):
                        os.replace(os.path.join(tmp, name),
                                   os.path.join(path, name))
                finally:
                    shutil.rmtree(tmp, ignore_errors=True)
                return
            except NotImplementedError as e:
                from ydf_amd.utils.log import info

                info(f"reference-format save not available for this "
                     f"model ({e}); writing npz container")
        os.makedirs(path, exist_ok=True)
        with open(os.path.join(path, "header.json"), "w") as f:
            json.dump(self._header(), f, indent=1)
        with open(os.path.join(path, "dataspec.json"), "w") as f:
            json.dump(self.dataspec.to_json(), f, indent=1)
        np.savez(os.path.join(path, "forest.npz"), feat=self.forest.feat,
                 thr=self.forest.thr, left=self.forest.left,
                 roots=self.forest.roots, cat_idx=self.forest.cat_idx,
                 masks=self.forest.masks, cover=self.forest.cover,
                 obl_ranges=self.forest.obl_ranges,
                 obl_attr=self.forest.obl_attr, obl_w=self.forest.obl_w,
                 na_right=self.forest.na_right,
                 set_idx=self.forest.set_idx,
                 set_offs=self.forest.set_offs,
                 set_items=self.forest.set_items)
        with open(os.path.join(path, "done"), "w") as f:
            f.write("")


@dataclasses.dataclass
class BenchmarkResult:
    examples_per_second: float
    num_runs: int
    duration_seconds: float
    batch_size: int

    def __str__(self) -> str:
        return (f"{self.examples_per_second:,.0f} examples/s "
                f"({self.num_runs} runs over {self.duration_seconds:.2f}s)")


class ServingSession:
    """Fixed-batch-size inference with the whole pipeline (binning +
    forest walk + activation) captured in one hipGraph. Feed
    same-shaped batches through predict(); input upload and output
    download stay outside the graph."""

    def __init__(self, model, batch_size: int, device=None):
        dev = torch.device(device) if device is not None \
            else default_device()
        if dev.type != "cuda":
            raise ValueError(
                "serving_session requires a GPU device (use predict() "
                "on CPU)")
        self._model = model
        self._n = int(batch_size)
        F = len(model.dataspec.feature_columns)
        self._X = torch.zeros((F, self._n), dtype=torch.float32,
                              device=dev)
        self._pin = torch.empty((F, self._n), dtype=torch.float32,
                                pin_memory=True)
        # warm every lazy cache (device forest, packed engines,
        # threshold checks) before capture
        for _ in range(2):
            model._apply_activation(model.predict_margin(self._X))
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._out = model._apply_activation(
                model.predict_margin(self._X))
        self._graph = g
        # pre-bound pinned staging on both sides: predict() does zero
        # tensor allocation
        self._pin_np = self._pin.numpy()
        self._out_pin = torch.empty(self._out.shape,
                                    dtype=self._out.dtype,
                                    pin_memory=True)
        self._out_np = self._out_pin.numpy()

    def predict(self, data) -> np.ndarray:
        # fast path: a preformed feature-major [F, batch] f32 matrix
        # skips per-call column encoding (the dominant Python cost at
        # small batches)
        if isinstance(data, np.ndarray) and data.ndim == 2 \
                and data.shape == self._X.shape:
            X_np = data
        else:
            X_np = self._model._encode_features(data)
        if X_np.shape[1] != self._n:
            raise ValueError(
                f"serving_session captured for batch size {self._n}, "
                f"got {X_np.shape[1]} examples")
        np.copyto(self._pin_np, X_np)
        self._X.copy_(self._pin, non_blocking=True)
        self._graph.replay()
        self._out_pin.copy_(self._out, non_blocking=True)
        torch.cuda.synchronize(self._X.device)
        return self._out_np.copy()

    __call__ = predict
