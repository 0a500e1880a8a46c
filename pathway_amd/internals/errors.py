"""Error logs (reference parse_graph.py:182-201 global error log)."""

from __future__ import annotations

_global_error_rows: list = []


def global_error_log():
    from pathway_amd.debug import table_from_rows
    from pathway_amd.internals.schema import schema_from_types

    schema = schema_from_types(message=str, trace=str)
    return table_from_rows(schema, list(_global_error_rows))


def record_error(message: str, trace: str = "") -> None:
    _global_error_rows.append((message, trace))
