"""Model families: GBT / RF / Isolation Forest / CART.

Capability analogues of the reference model classes
(model/gradient_boosted_trees/gradient_boosted_trees.h:57,
model/random_forest/random_forest.h:52,
model/isolation_forest/isolation_forest.h:51).
"""
from __future__ import annotations

import math

import numpy as np
import torch

from ydf_amd.dataset.dataspec import Task
from ydf_amd.model.generic_model import GenericModel


class DecisionForestModel(GenericModel):
    """Shared base for tree-ensemble models (mirrors
    ydf.DecisionForestModel)."""

    def get_tree(self, idx: int):
        from ydf_amd.model import tree as tree_lib

        return tree_lib.extract_tree(self.forest, idx)

    def get_all_trees(self):
        return [self.get_tree(i) for i in range(self.forest.n_trees)]

    def print_tree(self, idx: int = 0, max_depth: int = 6) -> str:
        from ydf_amd.model import tree as tree_lib

        return tree_lib.format_tree(
            self.get_tree(idx), self.dataspec, max_depth=max_depth)

    def plot_tree(self, idx: int = 0, max_depth: int = 6):
        """SVG rendering of one tree (mirrors PYDF model.plot_tree,
        port/python/ydf/model/decision_forest_model/
        decision_forest_model.py:102)."""
        from ydf_amd.model import tree as tree_lib

        return tree_lib.plot_tree(self.get_tree(idx), self.dataspec,
                                  max_depth=max_depth,
                                  label_classes=self.label_classes)

    def _rebuild_from_trees(self, trees) -> None:
        from ydf_amd.model import tree as tree_lib

        self.forest = tree_lib.build_forest_from_trees(
            trees, n_features=len(self.dataspec.feature_columns))
        # drop every derived/device representation of the old forest
        self._dev_forest.clear()
        if hasattr(self, "_thr_on_cuts"):
            del self._thr_on_cuts

    def set_tree(self, idx: int, tree) -> None:
        """Replaces tree `idx` (mirrors PYDF model.set_tree,
        decision_forest_model.py:148). The edited model serves the new
        structure everywhere (all engines are rebuilt lazily)."""
        trees = self.get_all_trees()
        if not 0 <= idx < len(trees):
            raise ValueError(f"tree index {idx} out of range")
        trees[idx] = tree
        self._rebuild_from_trees(trees)

    def add_tree(self, tree) -> None:
        """Appends a tree (mirrors PYDF model.add_tree,
        decision_forest_model.py:158). For multi-output GBT models the
        caller is responsible for keeping class striding consistent
        (append num_trees_per_iter trees per round), as in the
        reference."""
        trees = self.get_all_trees()
        trees.append(tree)
        self._rebuild_from_trees(trees)

    def remove_tree(self, idx: int) -> None:
        """Removes tree `idx` (mirrors PYDF model.remove_tree,
        decision_forest_model.py:167)."""
        trees = self.get_all_trees()
        if not 0 <= idx < len(trees):
            raise ValueError(f"tree index {idx} out of range")
        del trees[idx]
        self._rebuild_from_trees(trees)

    def iter_trees(self):
        """Iterator over the trees (PYDF model.iter_trees)."""
        for i in range(self.forest.n_trees):
            yield self.get_tree(i)

    def predict_leaves(self, data) -> np.ndarray:
        """Index of the active leaf per tree, [n, num_trees] (PYDF
        model.predict_leaves). Alias of leaf_indices."""
        return self.leaf_indices(data)

    def leaf_indices(self, data) -> np.ndarray:
        """[n, num_trees] i32: the leaf NODE index each example reaches
        in each tree (vectorized recursive partition walk; handles
        numerical, categorical-mask, oblique and NA-routing
        conditions)."""
        f = self.forest
        if f.has_set_conditions:
            raise NotImplementedError(
                "leaf_indices/distance on categorical-SET models is not "
                "supported")
        X = self._encode_features(data)
        n = X.shape[1]
        out = np.empty((n, f.n_trees), dtype=np.int32)
        has_na = f.has_na_routing

        def rec(node, idx, col):
            fi = int(f.feat[node])
            if fi < 0:
                col[idx] = node
                return
            ci = int(f.cat_idx[node])
            if ci >= 0:
                cb = X[fi, idx].astype(np.int64).clip(0, 255)
                right = ((f.masks[ci][cb >> 6]
                          >> (cb & 63).astype(np.uint64))
                         & np.uint64(1)).astype(bool)
            elif ci <= -2:
                oi = -(ci + 2)
                s0, nn = int(f.obl_ranges[oi, 0]), int(f.obl_ranges[oi, 1])
                acc = (f.obl_w[s0:s0 + nn, None]
                       * X[f.obl_attr[s0:s0 + nn]][:, idx]).sum(axis=0)
                right = acc > f.thr[node]
            else:
                xv = X[fi, idx]
                right = xv > f.thr[node]
                if has_na:
                    nanm = np.isnan(xv)
                    if nanm.any():
                        right = np.where(nanm, bool(f.na_right[node]),
                                         right)
            left = int(f.left[node])
            rec(left, idx[~right], col)
            rec(left + 1, idx[right], col)

        all_idx = np.arange(n)
        for t in range(f.n_trees):
            rec(int(f.roots[t]), all_idx, out[:, t])
        return out

    def distance(self, data1, data2=None) -> np.ndarray:
        """Pairwise tree-ensemble distance between examples (mirrors
        PYDF model.distance, port/python/ydf/model/
        decision_forest_model/decision_forest_model.py:196): 1 minus
        the fraction of trees in which the two examples reach the same
        leaf. distance[i, j] is between example i of data1 and example
        j of data2 (data2 defaults to data1). In [0, 1]; not a metric
        (no triangle inequality)."""
        l1 = self.leaf_indices(data1)
        l2 = l1 if data2 is None else self.leaf_indices(data2)
        n1, T = l1.shape
        n2 = l2.shape[0]
        d = np.empty((n1, n2), dtype=np.float32)
        step = max(1, 20_000_000 // max(T * max(n2, 1), 1))
        for i0 in range(0, n1, step):
            eq = l1[i0:i0 + step, None, :] == l2[None, :, :]
            d[i0:i0 + step] = 1.0 - eq.mean(axis=-1, dtype=np.float32)
        return d


class GradientBoostedTreesModel(DecisionForestModel):
    _model_type = "GRADIENT_BOOSTED_TREES"

    def initial_predictions(self) -> np.ndarray:
        return np.asarray(self.init_predictions, dtype=np.float32)

    def num_trees_per_iteration(self) -> int:
        """Trees per boosting iteration (PYDF
        gradient_boosted_trees_model.py:256): 1, or the class count
        for multi-class models."""
        return int(self.num_trees_per_iter)

    def early_stopping_triggered(self):
        """Whether training stopped early on the validation loss
        (None when unknown, e.g. imported models)."""
        v = (self.metadata or {}).get("early_stopping_triggered")
        return None if v is None else bool(v)

    def set_initial_predictions(self, initial_predictions) -> None:
        """Sets the model bias (PYDF set_initial_predictions)."""
        self.init_predictions = [float(v) for v in initial_predictions]
        self._dev_forest.clear()

    def output_logits(self) -> bool:
        """True when classification predictions are raw margins
        instead of probabilities (PYDF output_logits)."""
        return self.activation == "identity" \
            and self._task == Task.CLASSIFICATION

    def set_output_logits(self, output_logits: bool) -> None:
        if self._task != Task.CLASSIFICATION:
            raise ValueError(
                "output_logits only applies to classification models")
        if output_logits:
            self.activation = "identity"
        else:
            self.activation = "softmax" if (
                self.label_classes and len(self.label_classes) > 2) \
                else "sigmoid"

    def validation_evaluation(self):
        """Evaluation on the training-time validation set, from the
        last training-log entry (PYDF validation_evaluation)."""
        if not self.training_logs:
            return None
        from ydf_amd.metric.metric import Evaluation

        last = self.training_logs[-1]
        ev = Evaluation(loss=last.get("valid_loss"))
        ev.custom_metrics = {k: v for k, v in last.items()
                             if k not in ("iteration", "valid_loss")}
        return ev

    def validation_loss(self):
        if self.training_logs:
            return self.training_logs[-1].get("valid_loss")
        return None


class RandomForestModel(DecisionForestModel):
    _model_type = "RANDOM_FOREST"

    def out_of_bag_evaluations(self):
        """OOB evaluation logs (PYDF random_forest_model.py:31; alias
        of training_logs for Random Forests)."""
        return self.training_logs or []

    def winner_takes_all(self) -> bool:
        """Whether classification aggregates one vote per tree instead
        of summed leaf probabilities (PYDF winner_takes_all)."""
        return bool((self.metadata or {}).get("winner_take_all", False))

    def _leaf_scale(self) -> float:
        C = self._n_outputs()
        per = self.forest.n_trees // C if C > 1 else self.forest.n_trees
        return 1.0 / max(per, 1)


# A CART model is a random forest with a single tree (ydf convention).
CARTModel = RandomForestModel


class IsolationForestModel(DecisionForestModel):
    """Anomaly score = 2^(-E[path length] / c(n)) (reference
    model/isolation_forest/isolation_forest.h:51; leaf values store
    depth + c(n_leaf))."""

    _model_type = "ISOLATION_FOREST"

    @property
    def num_examples_per_tree(self):
        """Examples used to grow each tree (PYDF
        isolation_forest_model.py:26; attribute AND method form)."""
        return self._num_examples_per_tree

    @num_examples_per_tree.setter
    def num_examples_per_tree(self, v):
        from ydf_amd.model.generic_model import _CallableInt

        self._num_examples_per_tree = _CallableInt(int(v))

    def __init__(self, *args, num_examples_per_tree: int = 256, **kwargs):
        super().__init__(*args, **kwargs)
        self.num_examples_per_tree = num_examples_per_tree

    @staticmethod
    def expected_path_length(n: float) -> float:
        if n <= 1:
            return 0.0
        h = math.log(n - 1) + 0.5772156649
        return 2.0 * h - 2.0 * (n - 1) / n

    def _leaf_scale(self) -> float:
        return 1.0 / max(self.forest.n_trees, 1)

    def _apply_activation(self, m: torch.Tensor) -> torch.Tensor:
        denom = self.expected_path_length(float(self.num_examples_per_tree))
        return torch.exp2(-m[0] / denom)

    def _header(self) -> dict:
        h = super()._header()
        h["num_examples_per_tree"] = self.num_examples_per_tree
        return h

    def _load_extra(self, header: dict) -> None:
        self.num_examples_per_tree = header.get("num_examples_per_tree", 256)


MODEL_CLASSES = {
    c._model_type: c
    for c in (GenericModel, GradientBoostedTreesModel, RandomForestModel,
              IsolationForestModel)
}
