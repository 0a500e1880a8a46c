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
        self._buffer.append((self._time, kwargs))

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
    **kwargs: Any,
):
    from pathway_amd.debug import table_from_rows

    subject.run()
    subject.on_stop()
    names = schema.column_names()
    rows = []
    for t, rec in subject._buffer:
        if rec is None:
            continue
        rows.append(tuple(rec.get(n) for n in names) + (t, 1))
    return table_from_rows(schema, rows, is_stream=True)


ConnectorObserver = ConnectorSubject
