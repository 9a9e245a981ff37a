"""ydf_amd — an MI355X-native decision-forests framework.

A from-scratch reimplementation of the capability surface of Yggdrasil
Decision Forests (google/yggdrasil-decision-forests) designed for AMD
Instinct MI355X (gfx950/CDNA4): binned GPU-resident column store,
hand-written HIP kernels for the training and serving hot paths, and
RCCL-over-xGMI data-parallel training.

Public API mirrors the `ydf` Python package
(reference port/python/ydf/__init__.py:38-128).
"""

__version__ = "0.1.0"

# Learners
from ydf_amd.learner.generic_learner import GenericLearner
from ydf_amd.learner.specialized_learners import (
    CartLearner,
    DecisionTreeLearner,
    GradientBoostedTreesLearner,
    IsolationForestLearner,
    RandomForestLearner,
)

# Models
from ydf_amd.model.generic_model import GenericModel, ModelIOOptions
from ydf_amd.model.specialized import (
    CARTModel,
    DecisionForestModel,
    GradientBoostedTreesModel,
    IsolationForestModel,
    RandomForestModel,
)
from ydf_amd.model.model_lib import (
    deserialize_model,
    load_model,
    serialize_model,
)
from ydf_amd.model.import_ydf import load_ydf_model
from ydf_amd.model.export_ydf import export_ydf_model

# alias mirroring ydf.from_tensorflow_decision_forests-style importers
from_yggdrasil_model = load_ydf_model
from ydf_amd.model import tree

# Model export / serving extras
from ydf_amd.serving.deploy import to_docker
from ydf_amd.utils import usage
from ydf_amd.utils.folds import fold_splits, generate_folds
from ydf_amd.utils.registry import get_learner
from ydf_amd.serving.embed import to_cpp, to_java, to_js
from ydf_amd.learner.extras import (
    BackwardSelectionFeatureSelector,
    MultitaskerLearner,
    MultitaskItem,
)

# Custom losses
from ydf_amd.learner.custom_metric import (
    AbstractCustomMetric,
    BinaryClassificationMetric,
    MultiClassificationMetric,
    RegressionMetric,
)
from ydf_amd import experimental, help
from ydf_amd.learner.custom_loss import (
    Activation,
    BinaryClassificationLoss,
    MultiClassificationLoss,
    RegressionLoss,
)

# Tuner
from ydf_amd.learner.tuner import (
    OptimizerLogs,
    RandomSearchTuner,
    VizierTuner,
)

# Dataset
from ydf_amd.dataset.cache import DatasetCache, create_dataset_cache
from ydf_amd.dataset.dataset import VerticalDataset, create_vertical_dataset
from ydf_amd.dataset.dataspec import (
    Column,
    DataSpecification,
    Semantic,
    Task,
)

Feature = Column


class Monotonic:
    """Monotonic direction constants (mirrors ydf.Monotonic)."""

    INCREASING = 1
    DECREASING = -1

# Metric
from ydf_amd.metric.comparison import ModelComparison, compare_models
from ydf_amd.dataset.synthetic import (SyntheticDatasetOptions,
                                        generate_synthetic_dataset)
from ydf_amd.deep import (DeepModel, MultiLayerPerceptronLearner,
                          TabularTransformerLearner)
from ydf_amd.model.sklearn_io import from_sklearn
from ydf_amd.model.tree import (Leaf, NonLeaf, Tree,
                               build_forest_from_trees,
                               build_model_from_trees,
                               extract_tree, format_tree)
from ydf_amd.metric.metric import Evaluation, evaluate_predictions

# Utilities
from ydf_amd.utils.log import strict, verbose


# --- remaining PYDF surface names ------------------------------------------
class DistributedGradientBoostedTreesLearner(GradientBoostedTreesLearner):
    """Distributed GBT (reference learner/distributed_gradient_boosted_
    trees). MI355X mapping: distribution is one process per GPU over
    RCCL/xGMI — launch any GBT training under `torch.distributed.run`
    (one rank per GPU; rows are sharded, per-level histograms
    all-reduce). The reference's gRPC worker-pool parameters are
    accepted for API parity; `workers` (remote machine addresses) has
    no RCCL analogue here and raises with guidance."""

    def __init__(self, *args, workers=None, worker_logs=True,
                 force_numerical_discretization=False,
                 max_unique_values_for_discretized_numerical=16000,
                 **kwargs):
        if workers:
            raise NotImplementedError(
                "gRPC worker pools are not used on MI355X: launch this "
                "training under `python -m torch.distributed.run "
                "--nproc-per-node <gpus>` instead (one RCCL rank per "
                "GPU); see docs/DESIGN.md")
        # numerical features are ALWAYS 256-bin discretized on this
        # framework (the GPU histogram path), so
        # force_numerical_discretization is effectively always true
        # and the unique-value cap is the bin count
        self._worker_logs = worker_logs
        super().__init__(*args, **kwargs)

from ydf_amd.learner.extras import FeatureSelectorLogs  # noqa: E402


def start_worker(*args, **kwargs):
    """The reference starts gRPC workers; this framework distributes as
    one process per GPU via `python -m torch.distributed.run` (RCCL over
    xGMI) instead, so there is no worker daemon to start."""
    raise NotImplementedError(
        "distributed training runs as one process per GPU under "
        "torch.distributed.run (RCCL/xGMI); gRPC workers do not exist "
        "in this design. See README 'Distributed'.")


import dataclasses as _dataclasses  # noqa: E402
from typing import Optional as _Optional  # noqa: E402


@_dataclasses.dataclass
class ModelMetadata:
    """Mirrors ydf.ModelMetadata (owner/created date/uid/framework)."""

    owner: _Optional[str] = None
    created_date: _Optional[int] = None
    uid: _Optional[int] = None
    framework: _Optional[str] = "ydf_amd"


class NodeFormat:
    """Node storage formats (reference decision_tree.proto NodeFormat);
    models here always write BLOB_SEQUENCE, matching the reference
    default."""

    BLOB_SEQUENCE = "BLOB_SEQUENCE"


def from_tensorflow_decision_forests(*args, **kwargs):
    raise ImportError(
        "TensorFlow Decision Forests is not available in this "
        "environment; use load_ydf_model() for reference model "
        "directories or from_sklearn() for sklearn models.")


from ydf_amd.utils import folds as _folds  # noqa: E402
from ydf_amd.utils import usage as _usage  # noqa: E402


def _util_read_tf_record(path, *, compressed=None, process=None,
                         verbose=False, threads=None):
    """ydf.util.read_tf_record analogue: TFRecord shards -> dict of
    numpy columns (compression auto-detected; `process` maps raw
    decoded row dicts before column assembly)."""
    from ydf_amd.dataset.dataset import expand_sharded_paths
    from ydf_amd.dataset.tfrecord import read_tfrecord_columns

    p = str(path)
    if not p.startswith(("tfrecord:", "tfrecord+gzip:")):
        p = "tfrecord:" + p
    _, paths = expand_sharded_paths(p)
    cols = read_tfrecord_columns(paths)
    if process is not None:
        import numpy as _np

        n = len(next(iter(cols.values()))) if cols else 0
        rows = [{k: v[i] for k, v in cols.items()} for i in range(n)]
        rows = [r for r in (process(dict(r)) for r in rows)
                if r is not None]
        cols = {k: _np.asarray([r[k] for r in rows])
                for k in (rows[0] if rows else {})}
    return cols


def _util_write_tf_record(data, path, *, compressed=None):
    """ydf.util.write_tf_record analogue: dict of columns -> TFRecord."""
    from ydf_amd.dataset.tfrecord import write_tfrecord_columns

    write_tfrecord_columns(str(path), data,
                           compress=bool(compressed))


class util:  # noqa: N801  (PYDF exposes a lowercase `util` namespace)
    """Utility namespace (ydf.util analogue)."""

    generate_folds = staticmethod(_folds.generate_folds)
    fold_splits = staticmethod(_folds.fold_splits)
    usage = _usage
    read_tf_record = staticmethod(_util_read_tf_record)
    write_tf_record = staticmethod(_util_write_tf_record)


from ydf_amd.utils.log_book import LogBook as _LogBook  # noqa: E402

util.LogBook = _LogBook

version = "2.0.0+mi355x"
__version__ = version
