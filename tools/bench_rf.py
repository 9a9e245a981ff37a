#!/usr/bin/env python3
"""Random-forest benchmark (BASELINE config #4): 1000 trees on 1M x 500
mixed numerical+categorical synthetic data (categoricals currently enter as
integer codes; native set-splits tracked in ROADMAP).

Mirrors the reference harness shape (cli/monitoring/benchmark_training.cc
synthetic configs); measures trees/sec and full-forest wall-clock.
"""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ydf_amd as ydf  # noqa: E402
from ydf_amd.dataset.dataset import VerticalDataset  # noqa: E402
from ydf_amd.dataset.dataspec import (ColumnSpec, DataSpecification,  # noqa
                                      Semantic)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=1_000_000)
    ap.add_argument("--num-features", type=int, default=400)
    ap.add_argument("--cat-features", type=int, default=100)
    ap.add_argument("--trees", type=int, default=1000)
    ap.add_argument("--max-depth", type=int, default=16)
    ap.add_argument("--device", default=None)
    args = ap.parse_args()

    rng = np.random.RandomState(3)
    F = args.num_features + args.cat_features
    n = args.rows
    print(f"# generating {n}x{F} mixed data...", file=sys.stderr, flush=True)
    X = np.empty((F, n), dtype=np.float32)
    X[:args.num_features] = rng.randn(args.num_features, n)
    # categoricals: integer codes 0..31 with skewed frequencies
    for i in range(args.cat_features):
        X[args.num_features + i] = rng.zipf(1.5, n).clip(1, 32) - 1
    w = rng.randn(12)
    margin = (X[:12] * w[:, None]).sum(axis=0) + (X[args.num_features] < 2)
    y = (margin + rng.randn(n) > 0)

    cols = []
    from ydf_amd.dataset.dataspec import numerical_boundaries

    for i in range(F):
        cols.append(ColumnSpec(
            name=f"f{i}", semantic=Semantic.NUMERICAL,
            mean=float(X[i].mean()),
            boundaries=numerical_boundaries(X[i], max_sample=200_000)))
    cols.append(ColumnSpec(name="label", semantic=Semantic.CATEGORICAL,
                           vocab=["<OOD>", "n", "p"]))
    ds = VerticalDataset(
        X=X, dataspec=DataSpecification(columns=cols, label="label"),
        label_values=y.astype(np.float32))

    t0 = time.perf_counter()
    learner = ydf.RandomForestLearner(
        label="label", num_trees=args.trees, max_depth=args.max_depth,
        device=args.device)
    model = learner.train(ds)
    dt = time.perf_counter() - t0
    print(json.dumps({
        "metric": "rf_1Mx500_train_trees_per_s",
        "value": args.trees / dt,
        "unit": "trees/s",
        "rows": n,
        "features": F,
        "trees": args.trees,
        "max_depth": args.max_depth,
        "wall_clock_s": dt,
        "model_nodes": model.num_nodes(),
        "data": "synthetic",
    }))


if __name__ == "__main__":
    main()
