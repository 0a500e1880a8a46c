"""pw.io.elasticsearch — Elasticsearch sink over the _bulk REST API.

Reference: python/pathway/io/elasticsearch + src/connectors/data_storage/
elasticsearch.rs (931 LoC over elasticsearch-rs).  Emits standard
ndjson _bulk requests (index/delete actions; retractions delete by the
row's stable id), so it works against any ES/OpenSearch endpoint.
Tested against the capturing fake HTTP service.
"""

from __future__ import annotations

import json
from typing import Any

from pathway_amd.io import _rest


class ElasticSearchAuth:
    """Auth header factory (reference io/elasticsearch ElasticSearchAuth)."""

    def __init__(self, kind: str, **params: Any):
        self.kind = kind
        self.params = params

    @classmethod
    def basic(cls, username: str, password: str) -> "ElasticSearchAuth":
        return cls("basic", username=username, password=password)

    @classmethod
    def apikey(cls, api_key: str) -> "ElasticSearchAuth":
        return cls("apikey", api_key=api_key)

    @classmethod
    def bearer(cls, token: str) -> "ElasticSearchAuth":
        return cls("bearer", token=token)

    def headers(self) -> dict[str, str]:
        if self.kind == "basic":
            import base64

            tok = base64.b64encode(
                f"{self.params['username']}:{self.params['password']}".encode()
            ).decode()
            return {"Authorization": f"Basic {tok}"}
        if self.kind == "apikey":
            return {"Authorization": f"ApiKey {self.params['api_key']}"}
        if self.kind == "bearer":
            return {"Authorization": f"Bearer {self.params['token']}"}
        return {}


def write(
    table,
    host: str,
    auth: ElasticSearchAuth | None = None,
    index_name: str = "pathway",
    *,
    max_batch_size: int | None = None,
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    names = table.column_names()
    headers = auth.headers() if auth else {}

    def writer(batch):
        lines = []
        for key, values, time, diff in batch.rows():
            doc_id = repr(key)
            if diff > 0:
                rec = dict(zip(names, [_plain(v) for v in values]))
                rec["time"] = time
                rec["diff"] = diff
                lines.append(json.dumps(
                    {"index": {"_index": index_name, "_id": doc_id}}
                ))
                lines.append(json.dumps(rec, default=str))
            else:
                lines.append(json.dumps(
                    {"delete": {"_index": index_name, "_id": doc_id}}
                ))
        if not lines:
            return
        _rest.request(
            "POST", f"{host.rstrip('/')}/_bulk",
            raw_body=("\n".join(lines) + "\n").encode(),
            headers=headers, content_type="application/x-ndjson",
        )

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
