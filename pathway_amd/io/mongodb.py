"""pw.io.mongodb — MongoDB connector over the pure-python OP_MSG client.

Reference: python/pathway/io/mongodb + src/connectors/data_storage
mongodb writer.  read() supports static snapshots and a streaming mode
that tails the collection by ``_id`` order (new documents appear as
inserts — the offline analog of a change stream).  write() inserts
+diff rows and deletes -diff rows, so a retraction stream materializes
as the live document set.
"""

from __future__ import annotations

import time as _time
from typing import Any

from pathway_amd.io._mongo_protocol import MongoClient
from pathway_amd.io.formats.bson import ObjectId


class MongoReader:
    def __init__(self, source, connection_string: str, database: str,
                 collection: str, schema, *, mode: str = "streaming",
                 refresh_interval: float = 0.5, max_polls: int | None = None):
        self.source = source
        self.connection_string = connection_string
        self.database = database
        self.collection = collection
        self.schema = schema
        self.mode = mode
        self.refresh_interval = refresh_interval
        self.max_polls = max_polls
        self.last_id: ObjectId | None = None

    def run(self) -> None:
        client = None
        try:
            client = MongoClient(self.connection_string)
            polls = 0
            while True:
                self._poll(client)
                if self.mode == "static":
                    return
                polls += 1
                if self.max_polls is not None and polls >= self.max_polls:
                    return
                _time.sleep(self.refresh_interval)
        except Exception as e:
            self.source.fail(e)
        finally:
            if client is not None:
                client.close()
            self.source.finish()

    def _poll(self, client: MongoClient) -> None:
        filt: dict[str, Any] = {}
        if self.last_id is not None:
            filt = {"_id": {"$gt": self.last_id}}
        docs = client.find(self.database, self.collection, filt)
        names = self.schema.column_names()
        for doc in docs:
            oid = doc.get("_id")
            if isinstance(oid, ObjectId) and (
                self.last_id is None or oid > self.last_id
            ):
                self.last_id = oid
            self.source.emit([doc.get(n) for n in names])


def read(
    connection_string: str,
    *,
    database: str,
    collection: str,
    schema=None,
    mode: str = "streaming",
    refresh_interval: float = 0.5,
    autocommit_duration_ms: int | None = 1500,
    name: str | None = None,
    _max_polls: int | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import StreamingSource, spawn_reader
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    if schema is None:
        raise ValueError("pw.io.mongodb.read requires a schema")
    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]
    src = StreamingSource(names, dtypes, name=name)
    reader = MongoReader(
        src, connection_string, database, collection, schema,
        mode=mode, refresh_interval=refresh_interval, max_polls=_max_polls,
    )
    src.reader = reader
    spawn_reader(reader.run, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


def write(
    table,
    connection_string: str,
    *,
    database: str,
    collection: str,
    max_batch_size: int | None = None,
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    client = MongoClient(connection_string)
    names = table.column_names()

    def writer(batch):
        inserts = []
        for _key, values, time, diff in batch.rows():
            rec = dict(zip(names, [_plain(v) for v in values]))
            if diff > 0:
                rec["time"] = time
                rec["diff"] = diff
                inserts.append(rec)
            else:
                client.delete_many(database, collection, rec)
        if inserts:
            client.insert_many(database, collection, inserts)

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node


def _plain(v):
    from pathway_amd.internals.api import BasePointer
    from pathway_amd.internals.json import Json

    if isinstance(v, Json):
        return v.value
    if isinstance(v, BasePointer):
        return repr(v)
    if isinstance(v, tuple):
        return list(v)
    return v
