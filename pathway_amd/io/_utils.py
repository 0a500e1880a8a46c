"""Shared io helpers (reference io/_utils.py)."""
from __future__ import annotations

import importlib
from typing import Any


class MissingServiceDependency(RuntimeError):
    pass


def require_client(module: str, connector: str):
    try:
        return importlib.import_module(module)
    except ImportError as e:
        raise MissingServiceDependency(
            f"pw.io.{connector} needs the {module!r} client library, which is "
            f"not installed in this offline environment"
        ) from e


class CsvParserSettings:
    def __init__(self, delimiter=",", quote='"', escape=None,
                 enable_double_quote_escapes=True, enable_quoting=True,
                 comment_character=None):
        self.delimiter = delimiter
        self.quote = quote
        self.escape = escape
        self.comment_character = comment_character


def check_deprecated_kwargs(kwargs, names, stacklevel=1):
    pass


class RawDataSchema:
    pass


def expand_paths(path: str) -> list[str]:
    """Directory walk / glob pattern / single file → concrete file list
    (the posix_like scanner's path semantics, data_storage sharding.rs)."""
    import glob as _glob
    import os as _os

    if _os.path.isdir(path):
        out = []
        for root, _, fnames in _os.walk(path):
            for f in sorted(fnames):
                out.append(_os.path.join(root, f))
        return out
    if any(ch in path for ch in "*?["):
        return sorted(_glob.glob(path))
    return [path]
