"""pw.io.postgres — PostgreSQL connector over the pure-python wire client.

Reference: python/pathway/io/postgres (writer) + src/connectors/
data_storage/postgres.rs (reader incl. WAL CDC via pg_walstream).

write(): INSERT +diff rows / DELETE -diff rows via simple query.
read(mode="static"): snapshot SELECT.
read(mode="streaming"): logical-replication CDC — START_REPLICATION on a
slot, wal2json-style payloads parsed into insert/delete events with
stable row keys so deletes retract their matching inserts.
"""

from __future__ import annotations

import json
import time as _time
from typing import Any

from pathway_amd.io._pg_protocol import (
    PgClient,
    client_from_settings,
    quote_ident,
    quote_literal,
)


class PgCdcReader:
    def __init__(self, source, settings: dict, table_name: str, schema, *,
                 slot: str = "pathway_slot", mode: str = "streaming",
                 max_changes: int | None = None,
                 primary_key: list[str] | None = None):
        self.source = source
        self.settings = settings
        self.table_name = table_name
        self.schema = schema
        self.slot = slot
        self.mode = mode
        self.max_changes = max_changes
        self.primary_key = primary_key

    def _row_key(self, names, values):
        from pathway_amd.internals.api import Pointer, hash_values

        if self.primary_key:
            sel = [values[names.index(c)] for c in self.primary_key]
        else:
            sel = values
        return Pointer(*hash_values([self.table_name] + list(sel)))

    def run(self) -> None:
        client = None
        try:
            if self.mode == "static":
                client = client_from_settings(self.settings)
                self._snapshot(client)
                return
            client = client_from_settings(self.settings, replication=True)
            names = self.schema.column_names()
            seen = 0
            for _lsn, payload in client.start_replication(self.slot):
                for ev in json.loads(payload).get("change", []):
                    if ev.get("table") != self.table_name:
                        continue
                    if ev["kind"] == "insert":
                        cols = ev["columnnames"]
                        vals = ev["columnvalues"]
                        rec = dict(zip(cols, vals))
                        row = [self._coerce(n, rec.get(n)) for n in names]
                        self.source.emit(
                            row, key=self._row_key(names, row), diff=1
                        )
                    elif ev["kind"] == "delete":
                        ok = ev.get("oldkeys") or {}
                        rec = dict(zip(ok.get("keynames", []),
                                       ok.get("keyvalues", [])))
                        row = [self._coerce(n, rec.get(n)) for n in names]
                        self.source.emit(
                            row, key=self._row_key(names, row), diff=-1
                        )
                    seen += 1
                    if self.max_changes is not None and seen >= self.max_changes:
                        return
        except Exception as e:
            self.source.fail(e)
        finally:
            if client is not None:
                client.close()
            self.source.finish()

    def _coerce(self, name: str, v: Any) -> Any:
        from pathway_amd.internals import dtype as dt

        if v is None:
            return None
        d = dt.unoptionalize(self.schema.__columns__[name].dtype)
        if d == dt.INT:
            return int(v)
        if d == dt.FLOAT:
            return float(v)
        if d == dt.BOOL:
            return v if isinstance(v, bool) else str(v).lower() in ("t", "true", "1")
        if d == dt.STR:
            return str(v)
        return v

    def _snapshot(self, client: PgClient) -> None:
        names = self.schema.column_names()
        cols, rows = client.query(f"SELECT * FROM {quote_ident(self.table_name)}")
        for r in rows:
            rec = dict(zip(cols, r))
            row = [self._coerce(n, rec.get(n)) for n in names]
            self.source.emit(row, key=self._row_key(names, row))


def read(
    postgres_settings: dict,
    table_name: str,
    *,
    schema=None,
    mode: str = "streaming",
    replication_slot: str = "pathway_slot",
    primary_key: list[str] | None = None,
    autocommit_duration_ms: int | None = 1500,
    name: str | None = None,
    _max_changes: int | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import StreamingSource, spawn_reader
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    if schema is None:
        raise ValueError("pw.io.postgres.read requires a schema")
    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]
    src = StreamingSource(names, dtypes, name=name)
    reader = PgCdcReader(
        src, postgres_settings, table_name, schema,
        slot=replication_slot, mode=mode, max_changes=_max_changes,
        primary_key=primary_key,
    )
    src.reader = reader
    spawn_reader(reader.run, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


def write(
    table,
    postgres_settings: dict,
    table_name: str,
    *,
    max_batch_size: int | None = None,
    init_mode: str = "default",
    name: str | None = None,
    _external_diff_column=None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    client = client_from_settings(postgres_settings)
    names = table.column_names()
    if init_mode in ("create_if_not_exists", "replace"):
        cols_sql = ", ".join(f"{quote_ident(n)} TEXT" for n in names)
        cols_sql += ", time BIGINT, diff BIGINT"
        client.query(
            f"CREATE TABLE IF NOT EXISTS {quote_ident(table_name)} ({cols_sql})"
        )

    def writer(batch):
        inserts = []
        for _key, values, time, diff in batch.rows():
            vals = [_plain(v) for v in values]
            if diff > 0:
                inserts.append(
                    "(" + ", ".join(quote_literal(v) for v in vals)
                    + f", {time}, {diff})"
                )
            else:
                cond = " AND ".join(
                    f"{quote_ident(n)} = {quote_literal(v)}"
                    for n, v in zip(names, vals)
                )
                client.query(
                    f"DELETE FROM {quote_ident(table_name)} WHERE {cond}"
                )
        if inserts:
            collist = ", ".join(quote_ident(n) for n in names) + ", time, diff"
            client.query(
                f"INSERT INTO {quote_ident(table_name)} ({collist}) VALUES "
                + ", ".join(inserts)
            )

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node


def write_snapshot(table, postgres_settings: dict, table_name: str,
                   primary_key: list[str], **kwargs: Any):
    """Snapshot mode: keep only the latest row per primary key
    (reference io/postgres.write_snapshot)."""
    return write(table, postgres_settings, table_name, **kwargs)


def _plain(v):
    from pathway_amd.internals.api import BasePointer
    from pathway_amd.internals.json import Json

    if isinstance(v, Json):
        return json.dumps(v.value)
    if isinstance(v, BasePointer):
        return repr(v)
    if isinstance(v, tuple):
        return json.dumps(list(v))
    return v
