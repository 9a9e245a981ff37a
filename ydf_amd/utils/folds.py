"""Train/test fold generation (reference utils/fold_generator.h:47
GenerateFolds): deterministic k-fold index splits, optionally grouped so
whole groups stay in one fold."""
from __future__ import annotations

from typing import List, Optional, Tuple

import numpy as np


def generate_folds(num_examples: int, num_folds: int = 10,
                   seed: int = 1234,
                   groups: Optional[np.ndarray] = None
                   ) -> List[np.ndarray]:
    """Returns `num_folds` disjoint index arrays covering [0, n)."""
    rng = np.random.RandomState(seed)
    if groups is None:
        perm = rng.permutation(num_examples)
        return [np.sort(perm[i::num_folds]) for i in range(num_folds)]
    uniq = np.unique(groups)
    gperm = rng.permutation(len(uniq))
    fold_of_group = {uniq[g]: i % num_folds
                     for i, g in enumerate(gperm)}
    fold_of_row = np.fromiter((fold_of_group[g] for g in groups),
                              dtype=np.int64, count=num_examples)
    return [np.nonzero(fold_of_row == f)[0] for f in range(num_folds)]


def fold_splits(num_examples: int, num_folds: int = 10, seed: int = 1234,
                groups: Optional[np.ndarray] = None
                ) -> List[Tuple[np.ndarray, np.ndarray]]:
    """[(train_idx, test_idx)] per fold."""
    folds = generate_folds(num_examples, num_folds, seed, groups)
    out = []
    for f in range(num_folds):
        test = folds[f]
        train = np.concatenate([folds[g] for g in range(num_folds)
                                if g != f])
        out.append((np.sort(train), test))
    return out
