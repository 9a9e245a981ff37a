"""Shared CLI helpers."""
import argparse
import sys

import ydf_amd as ydf


def read_dataset(path: str):
    """Typed dataset path: 'csv:/path' or a bare csv path (reference
    dataset/formats typed-path convention)."""
    import pandas as pd

    if path.startswith("csv:"):
        path = path[4:]
    return pd.read_csv(path)


LEARNERS = {
    "GRADIENT_BOOSTED_TREES": ydf.GradientBoostedTreesLearner,
    "RANDOM_FOREST": ydf.RandomForestLearner,
    "CART": ydf.CartLearner,
    "ISOLATION_FOREST": ydf.IsolationForestLearner,
}

TASKS = {
    "CLASSIFICATION": ydf.Task.CLASSIFICATION,
    "REGRESSION": ydf.Task.REGRESSION,
    "ANOMALY_DETECTION": ydf.Task.ANOMALY_DETECTION,
}
