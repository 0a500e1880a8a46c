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


class EngineTimeMarker:
    """Type of the ``pw.io.ENGINE_TIME`` singleton (reference
    io/_utils.py:44): pass it where a connector accepts a column to
    select the engine (minibatch) time of each update instead."""

    def __repr__(self) -> str:
        return "pathway_amd.io.ENGINE_TIME"


ENGINE_TIME = EngineTimeMarker()

#: accepted wherever a duration is configured (reference io/_utils.py:107)
import datetime as _datetime  # noqa: E402

DurationLike = (int, float, _datetime.timedelta)


class CsvParserSettings:
    """CSV parser settings (reference io/_utils.py:217)."""

    def __init__(
        self,
        delimiter=",",
        quote='"',
        escape=None,
        enable_double_quote_escapes=True,
        enable_quoting=True,
        comment_character=None,
    ):
        self.delimiter = delimiter
        self.quote = quote
        self.escape = escape
        self.enable_double_quote_escapes = enable_double_quote_escapes
        self.enable_quoting = enable_quoting
        self.comment_character = comment_character


class TLSSettings:
    """TLS settings for connectors supporting encrypted transport
    (reference internals/_io_helpers.py:18).  This offline build records
    the configuration; sockets are upgraded with ``ssl`` when a mode
    other than "disable" is set and the server supports it."""

    def __init__(
        self,
        mode: str = "prefer",
        root_cert_path: str | None = None,
        client_cert_path: str | None = None,
        client_key_path: str | None = None,
    ):
        if mode not in (
            "disable", "allow", "prefer", "require", "verify-ca",
            "verify-full",
        ):
            raise ValueError(f"unknown TLS mode {mode!r}")
        self.mode = mode
        self.root_cert_path = root_cert_path
        self.client_cert_path = client_cert_path
        self.client_key_path = client_key_path
