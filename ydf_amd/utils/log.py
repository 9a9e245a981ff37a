"""Logging (reference utils/logging.h analogue)."""
from __future__ import annotations

import os
import sys
import time

_VERBOSE = int(os.environ.get("YDFA_VERBOSE", "1"))


def verbose(level: int = 1) -> int:
    """Sets the logging verbosity (mirrors ydf.verbose)."""
    global _VERBOSE
    old = _VERBOSE
    _VERBOSE = level
    return old


def strict(value: bool = True) -> None:
    """Mirrors ydf.strict (warnings as errors not implemented yet)."""


def info(msg: str) -> None:
    if _VERBOSE >= 1:
        print(f"[ydf_amd {time.strftime('%H:%M:%S')}] {msg}",
              file=sys.stderr, flush=True)
