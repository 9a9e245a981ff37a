"""Experiment tracker (capability analogue of ydf.util.LogBook,
port/python/ydf/util/log_book.py): a tiny SQLite-backed log of
(key, result) experiment records with dataframe export.

Written MI355X-framework-first: no behavioral coupling to the reference
implementation beyond the public API (directory store, SQLITE mode,
default-key augmentation, superset key filtering).
"""
from __future__ import annotations

import datetime
import enum
import json
import os
import sqlite3
from typing import Any, Dict, Optional

ExperimentKey = Dict[str, Any]
ExperimentResult = Dict[str, Any]

_RESERVED_KEYS = ("id", "timestamp")


class Mode(enum.Enum):
    SQLITE = "SQLITE"


class LogBook:
    """Keeps track of ML experiments in a directory.

    Usage:
        lb = LogBook(path)
        key = {"param1": 1, "param2": "abc"}
        if not lb.exist(key):
            lb.add(key, {"accuracy": 0.9})
        df = lb.to_dataframe({"param1": 1})
    """

    def __init__(self, directory: str,
                 print_num_experiments: bool = True,
                 mode: Mode = Mode.SQLITE,
                 default_keys: Optional[ExperimentKey] = None):
        if mode != Mode.SQLITE:
            raise NotImplementedError("only Mode.SQLITE is supported")
        self._default_keys = default_keys or {}
        self._directory = os.path.expanduser(directory)
        os.makedirs(self._directory, exist_ok=True)
        self._conn = sqlite3.connect(
            os.path.join(self._directory, "log_book.sqlite"), timeout=60)
        self._conn.execute(
            "CREATE TABLE IF NOT EXISTS experiments ("
            "id INTEGER PRIMARY KEY AUTOINCREMENT, "
            "timestamp TEXT, key TEXT, result TEXT)")
        self._conn.commit()
        if print_num_experiments:
            print(f"Found {self.num_experiments()} experiments")

    def _augment_key(self, key: ExperimentKey) -> ExperimentKey:
        out = dict(self._default_keys)
        out.update(key)
        return out

    def num_experiments(self) -> int:
        return self._conn.execute(
            "SELECT COUNT(*) FROM experiments").fetchone()[0]

    def _iter_keys(self):
        for (k,) in self._conn.execute("SELECT key FROM experiments"):
            yield self._augment_key(json.loads(k))

    def exist(self, key: ExperimentKey) -> bool:
        key = self._augment_key(key)
        return any(k == key for k in self._iter_keys())

    def count_key(self, key: ExperimentKey) -> int:
        key = self._augment_key(key)
        return sum(1 for k in self._iter_keys() if k == key)

    def add(self, key: ExperimentKey, result: ExperimentResult) -> None:
        """Records a new experiment; fails if the exact key exists."""
        if not isinstance(key, dict):
            raise ValueError("`key` is not a dictionary")
        if not isinstance(result, dict):
            raise ValueError("`result` is not a dictionary")
        for r in _RESERVED_KEYS:
            if r in key:
                raise ValueError(f"`key` contains a reserved key `{r}`")
            if r in result:
                raise ValueError(
                    f"`result` contains a reserved key `{r}`")
        if self.exist(key):
            raise ValueError(f"experiment {key} already recorded")
        self._conn.execute(
            "INSERT INTO experiments (timestamp, key, result) "
            "VALUES (?, ?, ?)",
            (datetime.datetime.now().isoformat(), json.dumps(key),
             json.dumps(result)))
        self._conn.commit()

    def to_dataframe(self, key_filter: Optional[ExperimentKey] = None):
        """All experiments (optionally those whose key is a SUPERSET of
        key_filter) as a pandas DataFrame with id/timestamp columns."""
        import pandas as pd

        records = []
        rows = self._conn.execute(
            "SELECT id, timestamp, key, result FROM experiments")
        for rid, ts, k, r in rows:
            key = self._augment_key(json.loads(k))
            if key_filter is not None and any(
                    kk not in key or key[kk] != vv
                    for kk, vv in key_filter.items()):
                continue
            rec = {"id": rid, "timestamp": ts}
            rec.update(key)
            rec.update(json.loads(r))
            records.append(rec)
        return pd.DataFrame(records)
