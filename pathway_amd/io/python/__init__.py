"""pw.io.python — ConnectorSubject (reference io/python/__init__.py:49).

Streaming subjects land with the connector-runtime phase; the static-mode
path (run() called once, rows buffered) works now.
"""
from __future__ import annotations

import json
import threading
from typing import Any

from pathway_amd.internals import dtype as dt


class ConnectorSubject:
    """Subclass and implement run(); call self.next(**kwargs) /
    next_json / next_str / next_bytes; self.commit() to end a batch."""

    def __init__(self, datasource_name: str = "python", **kwargs):
        self._buffer: list[tuple[int, dict | None]] = []
        self._time = 0

    def run(self) -> None:
        raise NotImplementedError

    def next(self, **kwargs: Any) -> None:
        emit = getattr(self, "_emit", None)
        if emit is not None:
            emit(kwargs)
        else:
            self._buffer.append((self._time, kwargs))

    def _add(self, key: Any, message: dict) -> None:
        """Insert a row under an explicit key (reference ConnectorSubject._add)."""
        emit = getattr(self, "_emit_keyed", None)
        if emit is not None:
            emit(message, key, 1)
        else:
            self._buffer.append((self._time, message))

    def _remove(self, key: Any, message: dict) -> None:
        """Retract a previously added row (reference ConnectorSubject._remove):
        the key and values must match the insertion for the diffs to cancel."""
        emit = getattr(self, "_emit_keyed", None)
        if emit is not None:
            emit(message, key, -1)

    def next_json(self, message: dict) -> None:
        self.next(**message)

    def next_str(self, message: str) -> None:
        self.next(data=message)

    def next_bytes(self, message: bytes) -> None:
        self.next(data=message)

    def commit(self) -> None:
        self._time += 1

    def close(self) -> None:
        pass

    def on_stop(self) -> None:
        pass


def read(
    subject: ConnectorSubject,
    *,
    schema=None,
    autocommit_duration_ms: int | None = 1500,
    name: str | None = None,
    mode: str = "streaming",
    **kwargs: Any,
):
    """Streaming mode: subject.run() on a reader thread feeding the engine
    live (reference Connector::run input thread, mod.rs:660)."""
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import StreamingSource, spawn_reader
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]
    src = StreamingSource(names, dtypes, name=name)

    subject._emit = lambda rec: src.emit([rec.get(n) for n in names])

    def _emit_keyed(rec, key, diff):
        from pathway_amd.internals.api import BasePointer, Pointer, hash_values

        if key is None:
            p = None
        elif isinstance(key, BasePointer):
            p = key
        else:
            lo, hi = hash_values(list(key) if isinstance(key, (list, tuple)) else [key])
            p = Pointer(lo, hi)
        src.emit([rec.get(n) for n in names], key=p, diff=diff)

    subject._emit_keyed = _emit_keyed

    def run_subject():
        try:
            subject.run()
            subject.on_stop()
        finally:
            src.finish()

    spawn_reader(run_subject, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


ConnectorObserver = ConnectorSubject
