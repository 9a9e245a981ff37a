"""Shared CLI helpers."""
import argparse
import sys

import ydf_amd as ydf


def read_dataset(path: str):
    """Typed dataset path: 'csv:/path', 'tfrecord:/path', 'avro:/path'
    or a bare csv path (reference dataset/formats typed-path
    convention). Returns a column dict."""
    from ydf_amd.dataset.dataset import _to_column_dict

    if ":" not in path or path.split(":", 1)[0] not in (
            "csv", "tfrecord", "avro"):
        path = "csv:" + path
    return _to_column_dict(path)


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


def write_dataset(path: str, cols) -> None:
    """Writes a column dict to a typed path (csv:/tfrecord:/avro:)."""
    import numpy as np

    fmt = "csv"
    if ":" in path and path.split(":", 1)[0] in ("csv", "tfrecord",
                                                 "avro"):
        fmt, path = path.split(":", 1)
    cols = {k: np.asarray(v) for k, v in cols.items()}
    if fmt == "csv":
        import pandas as pd

        pd.DataFrame(cols).to_csv(path, index=False)
    elif fmt == "tfrecord":
        from ydf_amd.dataset.tfrecord import write_tfrecord_columns

        write_tfrecord_columns(path, cols)
    else:
        from ydf_amd.dataset.avro import write_avro

        fields = []
        for k, v in cols.items():
            if v.dtype.kind in "fiub" and v.dtype.kind != "b":
                fields.append({"name": k, "type": "double"})
            elif v.dtype.kind == "b":
                fields.append({"name": k, "type": "boolean"})
            else:
                fields.append({"name": k, "type": "string"})
        schema = {"type": "record", "name": "row", "fields": fields}
        n = len(next(iter(cols.values())))
        recs = []
        for i in range(n):
            row = {}
            for k, v in cols.items():
                x = v[i]
                if v.dtype.kind == "b":
                    row[k] = bool(x)
                elif v.dtype.kind in "fiu":
                    row[k] = float(x)
                else:
                    row[k] = str(x)
            recs.append(row)
        write_avro(path, schema, recs)
