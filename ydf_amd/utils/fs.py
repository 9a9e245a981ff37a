"""Pluggable filesystem abstraction.

Reference analogue: utils/filesystem.h — a registry of filesystem
backends keyed by path scheme (the reference ships POSIX + GCS + TF
filesystems behind one interface; this module gives the same
extensibility point). Paths with a "scheme://" prefix dispatch to the
registered backend; everything else is the local filesystem.

    class MyFS:
        def open(self, path, mode="rb"): ...
        def glob(self, pattern): ...
        def exists(self, path): ...
    register_filesystem("myfs", MyFS())
    open_file("myfs://bucket/data.csv")
"""
from __future__ import annotations

import builtins
import glob as _glob
import os
import re
from typing import Dict

_SCHEME_RE = re.compile(r"^([a-zA-Z][a-zA-Z0-9+.\-]*)://")


class LocalFileSystem:
    """Default backend: the POSIX filesystem."""

    def open(self, path: str, mode: str = "rb"):
        return builtins.open(path, mode)

    def glob(self, pattern: str):
        return sorted(_glob.glob(pattern))

    def exists(self, path: str) -> bool:
        return os.path.exists(path)


_REGISTRY: Dict[str, object] = {"file": LocalFileSystem()}


def register_filesystem(scheme: str, fs) -> None:
    """Registers a filesystem backend for `scheme`:// paths (reference
    filesystem registration)."""
    _REGISTRY[scheme] = fs


def resolve(path: str):
    """(filesystem, scheme-stripped path) for `path`."""
    m = _SCHEME_RE.match(path)
    if m:
        scheme = m.group(1)
        fs = _REGISTRY.get(scheme)
        if fs is None:
            raise ValueError(
                f"no filesystem registered for scheme {scheme!r} "
                f"(register_filesystem({scheme!r}, fs))")
        rest = path[m.end():]
        if scheme == "file":
            rest = "/" + rest if not rest.startswith("/") else rest
        return fs, rest
    return _REGISTRY["file"], path


def open_file(path: str, mode: str = "rb"):
    fs, p = resolve(path)
    return fs.open(p, mode)


def glob_files(pattern: str):
    fs, p = resolve(pattern)
    m = _SCHEME_RE.match(pattern)
    prefix = pattern[: m.end()] if m and m.group(1) != "file" else ""
    return [prefix + q for q in fs.glob(p)]


def exists(path: str) -> bool:
    fs, p = resolve(path)
    return fs.exists(p)
