"""Data specification: column semantics, vocabularies, binning boundaries.

Capability analogue of the reference's DataSpecification proto +
dataspec inference (yggdrasil_decision_forests/dataset/data_spec.proto:49,
data_spec_inference.h:55), redesigned around a GPU-ready binned column
store: every numerical feature carries quantile bin boundaries so the
training path can materialize the uint8 binned matrix directly on device.
"""
from __future__ import annotations

import dataclasses
import enum
from typing import Dict, List, Optional

import numpy as np


class Semantic(enum.Enum):
    """Column semantic (reference data_spec.proto ColumnType)."""

    NUMERICAL = 1
    CATEGORICAL = 2
    BOOLEAN = 3
    HASH = 4
    CATEGORICAL_SET = 5
    DISCRETIZED_NUMERICAL = 6
    NUMERICAL_VECTOR_SEQUENCE = 7


class Task(enum.Enum):
    """Learning task (reference abstract_model.proto Task)."""

    CLASSIFICATION = 1
    REGRESSION = 2
    RANKING = 3
    CATEGORICAL_UPLIFT = 4
    NUMERICAL_UPLIFT = 5
    ANOMALY_DETECTION = 6
    SURVIVAL_ANALYSIS = 7


@dataclasses.dataclass
class Column:
    """User-facing feature declaration (mirrors ydf.Column)."""

    name: str
    semantic: Optional[Semantic] = None
    # monotonic constraint direction: +1 increasing, -1 decreasing, 0 none
    # (mirrors ydf.Feature(monotonic=...))
    monotonic: int = 0


# Out-of-vocabulary item: index 0 of every categorical vocabulary
# (reference convention, data_spec.proto VocabValue).
OOV_ITEM = "<OOD>"


@dataclasses.dataclass
class ColumnSpec:
    """Inferred per-column specification."""

    name: str
    semantic: Semantic
    # CATEGORICAL: vocabulary, index -> item (index 0 is OOV).
    vocab: Optional[List[str]] = None
    # NUMERICAL: statistics + quantile bin boundaries (ascending, <=255).
    mean: float = 0.0
    min_value: float = 0.0
    max_value: float = 0.0
    num_nas: int = 0
    boundaries: Optional[np.ndarray] = None
    # CATEGORICAL_SET expansion: this column is the virtual boolean
    # "set_token in data[set_source]" (reference categorical-set
    # conditions, approximated by per-token contains conditions)
    set_source: Optional[str] = None
    set_token: Optional[str] = None
    # NUMERICAL_VECTOR_SEQUENCE (reference data_spec.proto:73-81 + the
    # only GPU code in the reference, learner/decision_tree/gpu.cu.cc):
    # the parent column stores vecseq_dim; virtual projection columns
    # store (vecseq_source, vecseq_kind, vecseq_anchor) and hold
    #   kind "dot":  max_k <vec_k, anchor>    (ProjectedMoreThan)
    #   kind "dist": -min_k |vec_k - anchor|^2 (CloserThan, negated so
    #                 every condition keeps the ">= threshold" shape)
    vecseq_dim: int = 0
    vecseq_source: Optional[str] = None
    vecseq_kind: Optional[str] = None
    vecseq_anchor: Optional[np.ndarray] = None

    @property
    def vocab_size(self) -> int:
        return len(self.vocab) if self.vocab is not None else 0

    def to_json(self) -> dict:
        d = {
            "name": self.name,
            "semantic": self.semantic.name,
            "mean": float(self.mean),
            "min_value": float(self.min_value),
            "max_value": float(self.max_value),
            "num_nas": int(self.num_nas),
        }
        if self.vocab is not None:
            d["vocab"] = list(self.vocab)
        if self.boundaries is not None:
            d["boundaries"] = [float(v) for v in self.boundaries]
        if self.set_source is not None:
            d["set_source"] = self.set_source
            d["set_token"] = self.set_token
        if self.vecseq_dim:
            d["vecseq_dim"] = int(self.vecseq_dim)
        if self.vecseq_source is not None:
            d["vecseq_source"] = self.vecseq_source
            d["vecseq_kind"] = self.vecseq_kind
            d["vecseq_anchor"] = [float(v) for v in self.vecseq_anchor]
        return d

    @classmethod
    def from_json(cls, d: dict) -> "ColumnSpec":
        return cls(
            name=d["name"],
            semantic=Semantic[d["semantic"]],
            vocab=d.get("vocab"),
            mean=d.get("mean", 0.0),
            min_value=d.get("min_value", 0.0),
            max_value=d.get("max_value", 0.0),
            num_nas=d.get("num_nas", 0),
            boundaries=np.asarray(d["boundaries"], dtype=np.float32)
            if "boundaries" in d
            else None,
            set_source=d.get("set_source"),
            set_token=d.get("set_token"),
            vecseq_dim=d.get("vecseq_dim", 0),
            vecseq_source=d.get("vecseq_source"),
            vecseq_kind=d.get("vecseq_kind"),
            vecseq_anchor=np.asarray(d["vecseq_anchor"], dtype=np.float32)
            if "vecseq_anchor" in d else None,
        )


@dataclasses.dataclass
class DataSpecification:
    """Full dataset schema: ordered columns + label column index."""

    columns: List[ColumnSpec]
    label: Optional[str] = None

    def column(self, name: str) -> ColumnSpec:
        for c in self.columns:
            if c.name == name:
                return c
        raise KeyError(f"no column named {name!r}")

    @property
    def feature_columns(self) -> List[ColumnSpec]:
        return [c for c in self.columns if c.name != self.label]

    @property
    def label_column(self) -> ColumnSpec:
        assert self.label is not None
        return self.column(self.label)

    def to_json(self) -> dict:
        return {
            "columns": [c.to_json() for c in self.columns],
            "label": self.label,
        }

    @classmethod
    def from_json(cls, d: dict) -> "DataSpecification":
        return cls(
            columns=[ColumnSpec.from_json(c) for c in d["columns"]],
            label=d.get("label"),
        )


def categorical_vocab(values: np.ndarray, max_vocab_count: int = 2000,
                      min_vocab_frequency: int = 1) -> List[str]:
    """Vocabulary sorted by descending frequency then item value.

    Matches the reference's ordering (most frequent first, index 0 reserved
    for out-of-vocabulary).
    """
    items, counts = np.unique(values.astype(str), return_counts=True)
    order = np.lexsort((items, -counts))
    vocab = [OOV_ITEM]
    for idx in order[: max_vocab_count - 1]:
        if counts[idx] >= min_vocab_frequency:
            vocab.append(str(items[idx]))
    return vocab


def numerical_boundaries(values: np.ndarray, max_bins: int = 256,
                         max_sample: int = 1_000_000) -> np.ndarray:
    """Quantile bin boundaries (ascending, deduplicated, <= max_bins-1 cuts).

    The GPU-ready analogue of the reference's DISCRETIZED_NUMERICAL /
    dataset-cache binning (dataset_cache.h:15-58)."""
    v = values[np.isfinite(values)]
    if v.size == 0:
        return np.zeros((0,), dtype=np.float32)
    if v.size > max_sample:
        rng = np.random.RandomState(1234)
        v = v[rng.randint(0, v.size, max_sample)]
    uniq = np.unique(v)
    if uniq.size <= 1:
        return np.zeros((0,), dtype=np.float32)
    if uniq.size <= max_bins:
        # few distinct values: exact midpoints (reference
        # DISCRETIZED_NUMERICAL behavior for low-cardinality columns)
        cuts = (uniq[1:].astype(np.float64)
                + uniq[:-1].astype(np.float64)) / 2.0
    else:
        qs = np.linspace(0.0, 1.0, max_bins + 1)[1:-1]
        cuts = np.quantile(v, qs).astype(np.float64)
        cuts = np.unique(cuts)
        cuts = cuts[cuts < float(uniq[-1])]  # a cut at max separates nothing
    return cuts.astype(np.float32)
