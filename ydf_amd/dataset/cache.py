"""On-disk binned dataset cache for out-of-core (>HBM) training.

Reference analogue: the distributed dataset cache — an on-disk binned
column store built once and streamed during training
(learner/distributed_decision_tree/dataset_cache/dataset_cache.h:15-58)
— and the out-of-core sharded-sampling GBT path
(gradient_boosted_trees.cc:655 ShardedSamplingTrain).

MI355X design: the cache holds the 256-bin representation (u8), laid
out as ROW CHUNKS of a feature-major matrix (`chunk-XXXXX.bin` =
[F, rows_in_chunk] u8, C-contiguous) plus labels (f32) per chunk and a
dataspec. Training streams one chunk at a time through the histogram
kernels (H2D per chunk on GPU), so device/host memory holds one chunk
+ per-level histograms + per-row state — the dataset itself can exceed
HBM. Chunks are memory-mapped on CPU.
"""
from __future__ import annotations

import json
import os
from typing import Iterator, Tuple

import numpy as np

from ydf_amd.dataset.dataspec import DataSpecification


class DatasetCache:
    """Reader over a cache directory written by create_dataset_cache."""

    def __init__(self, path: str):
        self.path = path
        with open(os.path.join(path, "cache_meta.json")) as f:
            meta = json.load(f)
        self.n_rows = int(meta["n_rows"])
        self.n_features = int(meta["n_features"])
        self.chunk_rows = int(meta["chunk_rows"])
        self.n_chunks = int(meta["n_chunks"])
        self.cat_flags = np.asarray(meta["cat_flags"], dtype=bool)
        with open(os.path.join(path, "dataspec.json")) as f:
            self.dataspec = DataSpecification.from_json(json.load(f))

    def chunk(self, c: int) -> Tuple[np.ndarray, np.ndarray]:
        """(bins u8 [F, rows], labels f32 [rows]) — memory-mapped."""
        rows = min(self.chunk_rows,
                   self.n_rows - c * self.chunk_rows)
        bins = np.memmap(os.path.join(self.path, f"chunk-{c:05d}.bin"),
                         dtype=np.uint8, mode="r",
                         shape=(self.n_features, rows))
        labels = np.memmap(
            os.path.join(self.path, f"labels-{c:05d}.f32"),
            dtype=np.float32, mode="r", shape=(rows,))
        return bins, labels

    def chunks(self) -> Iterator[Tuple[int, np.ndarray, np.ndarray]]:
        for c in range(self.n_chunks):
            bins, labels = self.chunk(c)
            yield c, bins, labels


def create_dataset_cache(data, cache_dir: str, label: str,
                         task=None,
                         chunk_rows: int = 1_000_000,
                         max_vocab_count: int = 2000,
                         learner=None) -> DatasetCache:
    """Builds the on-disk binned cache from in-memory columns or a
    typed/sharded dataset path ("csv:...", "...@N", globs).

    The dataspec (quantile boundaries, vocabularies) is inferred from
    the data; binning then writes row ranges to chunk files. Cache
    CREATION currently materializes the raw columns in host RAM (the
    u8 cache is 4x smaller); TRAINING streams chunks, so the GPU only
    ever holds one chunk — the >HBM property this cache exists for."""
    from ydf_amd.dataset.dataset import (_to_column_dict,
                                         create_vertical_dataset)
    from ydf_amd.dataset.dataspec import Task

    os.makedirs(cache_dir, exist_ok=True)
    cols = _to_column_dict(data)
    ds = create_vertical_dataset(
        cols, label=label,
        task=task if task is not None else Task.CLASSIFICATION,
        max_vocab_count=max_vocab_count)
    from ydf_amd.model.forest import padded_boundaries

    bnd = padded_boundaries(ds.dataspec.feature_columns)
    cat_flags = np.asarray(
        [c.semantic.name == "CATEGORICAL"
         for c in ds.dataspec.feature_columns], dtype=bool)
    N = ds.n_examples
    F = ds.n_features
    n_chunks = (N + chunk_rows - 1) // chunk_rows

    import torch

    from ydf_amd import ops

    for c in range(n_chunks):
        lo = c * chunk_rows
        hi = min(lo + chunk_rows, N)
        X = torch.from_numpy(
            np.ascontiguousarray(ds.X[:, lo:hi]))
        bins = torch.empty(X.shape, dtype=torch.uint8)
        ops.bin_data(X, torch.from_numpy(bnd), bins)
        b = bins.numpy()
        ci = np.nonzero(cat_flags)[0]
        if ci.size:
            b[ci] = np.clip(ds.X[ci, lo:hi], 0, 255).astype(np.uint8)
        b.tofile(os.path.join(cache_dir, f"chunk-{c:05d}.bin"))
        np.ascontiguousarray(
            ds.label_values[lo:hi], dtype=np.float32).tofile(
            os.path.join(cache_dir, f"labels-{c:05d}.f32"))
    with open(os.path.join(cache_dir, "dataspec.json"), "w") as f:
        json.dump(ds.dataspec.to_json(), f)
    with open(os.path.join(cache_dir, "cache_meta.json"), "w") as f:
        json.dump({"version": 1,
                   "n_rows": int(N), "n_features": int(F),
                   "chunk_rows": int(chunk_rows),
                   "n_chunks": int(n_chunks),
                   "cat_flags": [bool(v) for v in cat_flags]}, f)
    with open(os.path.join(cache_dir, "done"), "w") as f:
        f.write("")
    return DatasetCache(cache_dir)
