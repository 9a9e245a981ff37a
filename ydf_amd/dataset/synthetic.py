"""Config-driven synthetic dataset generator (capability analogue of the
reference dataset/synthetic_dataset.{h,cc,proto}: numerical / categorical
/ boolean / multidimensional features with classification, regression or
ranking labels driven by a random ground-truth function; used by tests
and training benchmarks)."""
from __future__ import annotations

import dataclasses
from typing import Dict, Optional

import numpy as np


@dataclasses.dataclass
class SyntheticDatasetOptions:
    """Mirrors the knobs of the reference synthetic_dataset.proto."""

    num_examples: int = 10000
    num_numerical: int = 8
    num_categorical: int = 2
    categorical_vocab_size: int = 10
    num_boolean: int = 1
    num_multidimensional_numerical: int = 0
    multidimensional_dim: int = 4
    missing_ratio: float = 0.0
    # label
    task: str = "classification"   # classification | regression | ranking
    num_classes: int = 2
    num_examples_per_ranking_group: int = 10
    label_name: str = "LABEL"
    seed: int = 1234


def generate_synthetic_dataset(options: Optional[SyntheticDatasetOptions]
                               = None, **kwargs) -> Dict[str, np.ndarray]:
    """Returns a column dict; the label depends on a random linear +
    interaction function of the features so models can learn it."""
    opt = options or SyntheticDatasetOptions(**kwargs)
    rng = np.random.RandomState(opt.seed)
    n = opt.num_examples
    cols: Dict[str, np.ndarray] = {}
    signal = np.zeros(n, dtype=np.float64)

    num_feats = []
    for i in range(opt.num_numerical):
        x = rng.randn(n).astype(np.float32)
        cols[f"num_{i}"] = x
        num_feats.append(x)
        signal += rng.uniform(-1, 1) * x
    # pairwise interaction from the first two numericals
    if len(num_feats) >= 2:
        signal += 0.5 * num_feats[0] * num_feats[1]
    for i in range(opt.num_categorical):
        codes = rng.randint(0, opt.categorical_vocab_size, n)
        effect = rng.uniform(-1, 1, opt.categorical_vocab_size)
        signal += effect[codes]
        cols[f"cat_{i}"] = np.array(
            [f"v_{c}" for c in codes], dtype=object)
    for i in range(opt.num_boolean):
        b = rng.randint(0, 2, n)
        signal += rng.uniform(-1, 1) * b
        cols[f"bool_{i}"] = b.astype(bool)
    for i in range(opt.num_multidimensional_numerical):
        for d in range(opt.multidimensional_dim):
            x = rng.randn(n).astype(np.float32)
            cols[f"multi_{i}.{d}"] = x
            signal += rng.uniform(-0.3, 0.3) * x

    if opt.missing_ratio > 0:
        for name, v in cols.items():
            miss = rng.random_sample(n) < opt.missing_ratio
            if v.dtype.kind == "f":
                v = v.copy()
                v[miss] = np.nan
                cols[name] = v
            elif v.dtype == object:
                v = v.copy()
                v[miss] = ""
                cols[name] = v

    noise = rng.randn(n) * 0.3
    if opt.task == "regression":
        cols[opt.label_name] = (signal + noise).astype(np.float32)
    elif opt.task == "ranking":
        g = opt.num_examples_per_ranking_group
        cols["GROUP"] = np.repeat(np.arange((n + g - 1) // g), g)[:n]
        rel = signal + noise
        # 5-level relevance by within-dataset quantile
        qs = np.quantile(rel, [0.4, 0.65, 0.85, 0.95])
        cols[opt.label_name] = np.searchsorted(qs, rel).astype(np.float32)
    else:
        if opt.num_classes == 2:
            p = 1.0 / (1.0 + np.exp(-(signal + noise)))
            y = (rng.random_sample(n) < p).astype(int)
            cols[opt.label_name] = np.where(y == 1, "pos", "neg")
        else:
            qs = np.quantile(signal + noise,
                             np.linspace(0, 1, opt.num_classes + 1)[1:-1])
            y = np.searchsorted(qs, signal + noise)
            cols[opt.label_name] = np.array(
                [f"c_{v}" for v in y], dtype=object)
    return cols
