"""scikit-learn model import (capability analogue of PYDF's
ydf.from_sklearn, port/python/ydf/model/export_sklearn.py): converts a
trained scikit-learn tree-ensemble into a ydf_amd model that serves
through the flat-forest HIP/CPU kernels.

Supported: DecisionTreeClassifier/Regressor, ExtraTree*,
RandomForestClassifier/Regressor, ExtraTreesClassifier/Regressor,
GradientBoostingClassifier (binary) / GradientBoostingRegressor,
IsolationForest.
"""
from __future__ import annotations

import math
from typing import List, Optional

import numpy as np

from ydf_amd.dataset.dataspec import (ColumnSpec, DataSpecification,
                                      Semantic, Task)
from ydf_amd.model.forest import FlatForest
from ydf_amd.model.specialized import (GradientBoostedTreesModel,
                                       IsolationForestModel,
                                       RandomForestModel)


def _convert_tree(tree, leaf_fn, feats, thrs, lefts, covers):
    """Appends one sklearn `tree_` to the flat arrays. sklearn routes
    x <= threshold LEFT, which matches our layout (left = "no" child of
    `x > thr`, right = left + 1). Children are allocated as an adjacent
    pair so right == left + 1 holds."""
    cl = tree.children_left
    cr = tree.children_right
    feature = tree.feature
    threshold = tree.threshold
    n_samples = tree.n_node_samples

    def new_slot():
        feats.append(-1)
        thrs.append(0.0)
        lefts.append(0)
        covers.append(0.0)
        return len(feats) - 1

    root = new_slot()
    stack = [(0, root, 0)]  # (sklearn node, our slot, depth)
    while stack:
        n, slot, depth = stack.pop()
        covers[slot] = float(n_samples[n])
        if cl[n] == -1:  # leaf
            thrs[slot] = float(leaf_fn(n, depth))
            continue
        feats[slot] = int(feature[n])
        thrs[slot] = float(threshold[n])
        li = new_slot()
        new_slot()
        lefts[slot] = li
        stack.append((cl[n], li, depth + 1))
        stack.append((cr[n], li + 1, depth + 1))
    return root


def _build_forest(sk_trees, leaf_fns) -> FlatForest:
    feats: List[int] = []
    thrs: List[float] = []
    lefts: List[int] = []
    covers: List[float] = []
    roots = []
    for tree, leaf_fn in zip(sk_trees, leaf_fns):
        roots.append(_convert_tree(tree, leaf_fn, feats, thrs, lefts,
                                   covers))
    return FlatForest(
        feat=np.asarray(feats, np.int32),
        thr=np.asarray(thrs, np.float32),
        left=np.asarray(lefts, np.int32),
        roots=np.asarray(roots, np.int32),
        cover=np.asarray(covers, np.float32))


def _dataspec(n_features: int, label: str, classes=None,
              feature_names=None) -> DataSpecification:
    cols = []
    for i in range(n_features):
        name = feature_names[i] if feature_names is not None \
            else f"features[{i}]"
        cols.append(ColumnSpec(name=str(name), semantic=Semantic.NUMERICAL))
    from ydf_amd.dataset.dataspec import OOV_ITEM

    if classes is not None:
        cols.append(ColumnSpec(name=label, semantic=Semantic.CATEGORICAL,
                               vocab=[OOV_ITEM] + [str(c) for c in classes]))
    else:
        cols.append(ColumnSpec(name=label, semantic=Semantic.NUMERICAL))
    return DataSpecification(columns=cols, label=label)


def from_sklearn(sklearn_model, label_name: str = "label",
                 feature_names: Optional[List[str]] = None):
    """Converts a trained scikit-learn model into a ydf_amd model
    (mirrors ydf.from_sklearn)."""
    name = type(sklearn_model).__name__
    if name in ("DecisionTreeClassifier", "ExtraTreeClassifier"):
        return _from_tree_classifier([sklearn_model], sklearn_model,
                                     label_name, feature_names)
    if name in ("DecisionTreeRegressor", "ExtraTreeRegressor"):
        return _from_tree_regressor([sklearn_model], sklearn_model,
                                    label_name, feature_names)
    if name in ("RandomForestClassifier", "ExtraTreesClassifier"):
        return _from_tree_classifier(sklearn_model.estimators_,
                                     sklearn_model, label_name,
                                     feature_names)
    if name in ("RandomForestRegressor", "ExtraTreesRegressor"):
        return _from_tree_regressor(sklearn_model.estimators_,
                                    sklearn_model, label_name,
                                    feature_names)
    if name == "GradientBoostingClassifier":
        return _from_gbt_classifier(sklearn_model, label_name,
                                    feature_names)
    if name == "GradientBoostingRegressor":
        return _from_gbt_regressor(sklearn_model, label_name, feature_names)
    if name == "IsolationForest":
        return _from_isolation_forest(sklearn_model, label_name,
                                      feature_names)
    raise NotImplementedError(
        f"cannot convert sklearn model of type {name}")


def _from_tree_classifier(estimators, sk, label, feature_names):
    classes = list(sk.classes_)
    C = len(classes)
    F = sk.n_features_in_
    if C == 2:
        def leaf_fn_for(est):
            v = est.tree_.value

            def fn(n, depth):
                row = v[n][0]
                s = row.sum()
                return row[1] / s if s else 0.0
            return fn

        forest = _build_forest([e.tree_ for e in estimators],
                               [leaf_fn_for(e) for e in estimators])
        return RandomForestModel(
            forest=forest,
            dataspec=_dataspec(F, label, classes, feature_names),
            task=Task.CLASSIFICATION, label_classes=[str(c) for c in
                                                     classes],
            init_predictions=[0.0], num_trees_per_iter=1,
            activation="identity",
            metadata={"imported_from": "sklearn"})
    # multi-class: one tree per (sklearn tree, class), interleaved by class
    trees, fns = [], []
    for e in estimators:
        v = e.tree_.value
        for c in range(C):
            def fn(n, depth, v=v, c=c):
                row = v[n][0]
                s = row.sum()
                return row[c] / s if s else 0.0
            trees.append(e.tree_)
            fns.append(fn)
    forest = _build_forest(trees, fns)
    return RandomForestModel(
        forest=forest, dataspec=_dataspec(F, label, classes, feature_names),
        task=Task.CLASSIFICATION, label_classes=[str(c) for c in classes],
        init_predictions=[0.0] * C, num_trees_per_iter=C,
        activation="identity", metadata={"imported_from": "sklearn"})


def _from_tree_regressor(estimators, sk, label, feature_names):
    F = sk.n_features_in_

    def leaf_fn_for(est):
        v = est.tree_.value

        def fn(n, depth):
            return v[n][0][0]
        return fn

    forest = _build_forest([e.tree_ for e in estimators],
                           [leaf_fn_for(e) for e in estimators])
    return RandomForestModel(
        forest=forest, dataspec=_dataspec(F, label, None, feature_names),
        task=Task.REGRESSION, init_predictions=[0.0],
        num_trees_per_iter=1, activation="identity",
        metadata={"imported_from": "sklearn"})


def _gbt_init_value(sk) -> float:
    init = sk.init_
    if init == "zero" or init is None:
        return 0.0
    if hasattr(init, "constant_"):           # DummyRegressor (mean)
        return float(np.asarray(init.constant_).ravel()[0])
    if hasattr(init, "class_prior_"):        # DummyClassifier prior
        p = float(np.clip(init.class_prior_[1], 1e-9, 1 - 1e-9))
        return math.log(p / (1 - p))
    return 0.0


def _from_gbt_classifier(sk, label, feature_names):
    classes = list(sk.classes_)
    if len(classes) != 2:
        raise NotImplementedError(
            "GradientBoostingClassifier import supports binary labels")
    F = sk.n_features_in_
    lr = float(sk.learning_rate)
    ests = [e[0] for e in sk.estimators_]

    def leaf_fn_for(est):
        v = est.tree_.value

        def fn(n, depth):
            return v[n][0][0] * lr
        return fn

    forest = _build_forest([e.tree_ for e in ests],
                           [leaf_fn_for(e) for e in ests])
    return GradientBoostedTreesModel(
        forest=forest, dataspec=_dataspec(F, label, classes, feature_names),
        task=Task.CLASSIFICATION, label_classes=[str(c) for c in classes],
        init_predictions=[_gbt_init_value(sk)], num_trees_per_iter=1,
        activation="sigmoid", metadata={"imported_from": "sklearn"})


def _from_gbt_regressor(sk, label, feature_names):
    F = sk.n_features_in_
    lr = float(sk.learning_rate)
    ests = [e[0] for e in sk.estimators_]

    def leaf_fn_for(est):
        v = est.tree_.value

        def fn(n, depth):
            return v[n][0][0] * lr
        return fn

    forest = _build_forest([e.tree_ for e in ests],
                           [leaf_fn_for(e) for e in ests])
    return GradientBoostedTreesModel(
        forest=forest, dataspec=_dataspec(F, label, None, feature_names),
        task=Task.REGRESSION, init_predictions=[_gbt_init_value(sk)],
        num_trees_per_iter=1, activation="identity",
        metadata={"imported_from": "sklearn"})


def _average_path_length(n: float) -> float:
    if n <= 1:
        return 0.0
    if n == 2:
        return 1.0
    h = math.log(n - 1) + 0.5772156649
    return 2.0 * h - 2.0 * (n - 1) / n


def _from_isolation_forest(sk, label, feature_names):
    F = sk.n_features_in_

    def leaf_fn_for(est):
        t = est.tree_
        ns = t.n_node_samples

        def fn(n, depth):
            return depth + _average_path_length(float(ns[n]))
        return fn

    forest = _build_forest([e.tree_ for e in sk.estimators_],
                           [leaf_fn_for(e) for e in sk.estimators_])
    return IsolationForestModel(
        forest=forest, dataspec=_dataspec(F, label, None, feature_names),
        task=Task.ANOMALY_DETECTION, init_predictions=[0.0],
        num_trees_per_iter=1, activation="identity",
        num_examples_per_tree=int(sk.max_samples_),
        metadata={"imported_from": "sklearn"})
