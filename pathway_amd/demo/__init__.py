"""pw.demo — synthetic demo streams (reference demo/__init__.py:29-200)."""
from __future__ import annotations

from typing import Any, Callable

from pathway_amd.internals import dtype as dt


def generate_custom_stream(
    value_generators: dict[str, Callable[[int], Any]],
    *,
    schema,
    nb_rows: int | None = 100,
    autocommit_duration_ms: int = 1000,
    input_rate: float = 1.0,
    persistent_id: str | None = None,
    name: str | None = None,
):
    from pathway_amd.debug import table_from_rows

    names = schema.column_names()
    rows = []
    for i in range(nb_rows or 100):
        rows.append(tuple(value_generators[n](i) for n in names) + (i, 1))
    return table_from_rows(schema, rows, is_stream=True)


def range_stream(nb_rows: int = 30, offset: int = 0, **kwargs):
    from pathway_amd.internals.schema import schema_from_types

    schema = schema_from_types(value=float)
    return generate_custom_stream(
        {"value": lambda i: float(i + offset)}, schema=schema, nb_rows=nb_rows, **kwargs
    )


def noisy_linear_stream(nb_rows: int = 10, input_rate: float = 1.0, **kwargs):
    import random

    from pathway_amd.internals.schema import schema_from_types

    rng = random.Random(0)
    schema = schema_from_types(x=float, y=float)
    return generate_custom_stream(
        {"x": lambda i: float(i), "y": lambda i: i + rng.uniform(-1, 1)},
        schema=schema,
        nb_rows=nb_rows,
        **kwargs,
    )


def replay_csv(path: str, *, schema, input_rate: float = 1.0):
    from pathway_amd.io import csv as io_csv

    return io_csv.read(path, schema=schema, mode="static")


def replay_csv_with_time(path: str, *, schema, time_column: str, unit: str = "s", autocommit_ms: int = 100, speedup: float = 1):
    return replay_csv(path, schema=schema)
