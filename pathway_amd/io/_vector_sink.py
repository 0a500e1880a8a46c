"""Shared machinery for vector-database sinks
(pinecone/qdrant/chroma/milvus/weaviate/leann).

Each sink receives a table whose rows carry a vector column plus
arbitrary metadata columns; +diff rows upsert and -diff rows delete by
the row's stable id (the engine key), mirroring the reference's vector
sink semantics (src/connectors/data_storage/{pinecone,qdrant,...}.rs).
"""

from __future__ import annotations

from typing import Any, Callable


def _plain(v):
    import numpy as np

    from pathway_amd.internals.api import BasePointer
    from pathway_amd.internals.json import Json

    if isinstance(v, Json):
        return v.value
    if isinstance(v, BasePointer):
        return repr(v)
    if isinstance(v, np.ndarray):
        return [float(x) for x in v.ravel()]
    if isinstance(v, tuple):
        return list(v)
    return v


def make_vector_writer(
    table,
    vector_column: str,
    *,
    upsert: Callable[[list[dict]], None],
    delete: Callable[[list[str]], None],
):
    """Build an OutputNode writer: rows -> [{id, vector, metadata}]."""
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    names = table.column_names()

    def writer(batch):
        ups: list[dict] = []
        dels: list[str] = []
        for key, values, time, diff in batch.rows():
            rec = dict(zip(names, [_plain(v) for v in values]))
            rid = repr(key)
            if diff > 0:
                vec = rec.pop(vector_column, None)
                ups.append({"id": rid, "vector": vec, "metadata": rec})
            else:
                dels.append(rid)
        if ups:
            upsert(ups)
        if dels:
            delete(dels)

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
