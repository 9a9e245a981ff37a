"""Usage telemetry hooks (reference utils/usage.h:40-75): no-op by
default; a deployment can install callbacks observing training and
inference events."""
from __future__ import annotations

from typing import Callable, List, Optional

_on_training_start: List[Callable] = []
_on_training_end: List[Callable] = []
_on_inference: List[Callable] = []


def register_on_training_start(cb: Callable) -> None:
    _on_training_start.append(cb)


def register_on_training_end(cb: Callable) -> None:
    _on_training_end.append(cb)


def register_on_inference(cb: Callable) -> None:
    _on_inference.append(cb)


def clear() -> None:
    _on_training_start.clear()
    _on_training_end.clear()
    _on_inference.clear()


def on_training_start(learner_name: str, num_examples: int) -> None:
    for cb in _on_training_start:
        cb(learner_name=learner_name, num_examples=num_examples)


def on_training_end(learner_name: str, num_examples: int,
                    num_trees: Optional[int], wall_seconds: float) -> None:
    for cb in _on_training_end:
        cb(learner_name=learner_name, num_examples=num_examples,
           num_trees=num_trees, wall_seconds=wall_seconds)


def on_inference(num_examples: int) -> None:
    for cb in _on_inference:
        cb(num_examples=num_examples)
