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


class GenericLearner:
    """Base learner: dataset ingestion, binning, device selection."""

    def __init__(self, label: Optional[str], task: Task = Task.CLASSIFICATION,
                 features: Optional[Sequence[Union[str, Column]]] = None,
                 max_vocab_count: int = 2000, min_vocab_frequency: int = 1,
                 random_seed: int = 123456, device=None,
                 num_threads: Optional[int] = None):
        self.label = label
        self._task = task
        self.features = features
        self.max_vocab_count = max_vocab_count
        self.min_vocab_frequency = min_vocab_frequency
        self.random_seed = random_seed
        self.device = device
        self.num_threads = num_threads
        self.hyperparameters: Dict = {}

    def task(self) -> Task:
        return self._task

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

    def _bin_matrix(self, ds_X: np.ndarray, cat_feats: np.ndarray,
                    bnd: np.ndarray, device: torch.device) -> torch.Tensor:
        """Bins numericals by quantile cuts; categorical codes pass through
        as their own bin index (clamped to 255)."""
        X = torch.from_numpy(np.ascontiguousarray(ds_X)).to(device)
        bnd_t = torch.from_numpy(bnd).to(device)
        bins = torch.empty(X.shape, dtype=torch.uint8, device=device)
        ops.bin_data(X, bnd_t, bins)
        ci = np.nonzero(cat_feats)[0]
        if ci.size:
            idx = torch.from_numpy(ci).to(device)
            bins[idx] = X[idx].clamp_(0, 255).to(torch.uint8)
        del X
        return bins

    def _prepare(self, data, device: torch.device):
        """Dataset -> (VerticalDataset, binned u8 [F,N] on device,
        labels f32 [N] on device, padded boundary matrix np [F,n_cuts],
        cat_flags u8 tensor or None)."""
        if isinstance(data, VerticalDataset):
            ds = data
        else:
            ds = create_vertical_dataset(
                data, label=self.label, task=self._task,
                features=self.features, max_vocab_count=self.max_vocab_count,
                min_vocab_frequency=self.min_vocab_frequency)
        bnd = padded_boundaries(ds.dataspec.feature_columns)
        cat_feats = self._cat_feature_flags(ds)
        bins = self._bin_matrix(ds.X, cat_feats, bnd, device)
        labels = None
        if ds.label_values is not None:
            labels = torch.from_numpy(
                np.ascontiguousarray(ds.label_values)).to(device)
        cat_flags = None
        if cat_feats.any():
            cat_flags = torch.from_numpy(
                cat_feats.astype(np.uint8)).to(device)
        return ds, bins, labels, bnd, cat_flags

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

    def validate_hyperparameters(self) -> None:
        pass
