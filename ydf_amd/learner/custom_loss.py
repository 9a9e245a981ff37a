"""User-defined losses for GBT (capability analogue of the reference's
custom-loss callbacks, learner/gradient_boosted_trees/loss/loss_imp_custom_*
and PYDF ydf/learner/custom_loss.py).

The callbacks receive numpy arrays (labels, predictions are margins) and
return numpy gradients/hessians; the trainer uploads them to the device per
iteration, so custom losses run at host speed (documented)."""
from __future__ import annotations

import dataclasses
import enum
from typing import Callable, Optional

import numpy as np


class Activation(enum.Enum):
    IDENTITY = "identity"
    SIGMOID = "sigmoid"
    SOFTMAX = "softmax"


@dataclasses.dataclass
class RegressionLoss:
    """loss for Task.REGRESSION: gradient_and_hessian(labels, preds) ->
    (g, h) arrays; initial_predictions(labels, weights) -> float."""

    gradient_and_hessian: Callable
    initial_predictions: Optional[Callable] = None
    loss: Optional[Callable] = None
    activation: Activation = Activation.IDENTITY
    may_trigger_gc: bool = True


@dataclasses.dataclass
class BinaryClassificationLoss:
    """loss for binary Task.CLASSIFICATION; labels passed as 0/1 floats."""

    gradient_and_hessian: Callable
    initial_predictions: Optional[Callable] = None
    loss: Optional[Callable] = None
    activation: Activation = Activation.SIGMOID
    may_trigger_gc: bool = True


@dataclasses.dataclass
class MultiClassificationLoss:
    """loss for multi-class: gradient_and_hessian(labels, preds[C,N]) ->
    (g [C,N], h [C,N])."""

    gradient_and_hessian: Callable
    initial_predictions: Optional[Callable] = None
    loss: Optional[Callable] = None
    activation: Activation = Activation.SOFTMAX
    may_trigger_gc: bool = True


def default_initial_predictions(custom, labels: np.ndarray) -> float:
    if custom.initial_predictions is not None:
        return float(custom.initial_predictions(
            labels, np.ones_like(labels)))
    return 0.0
