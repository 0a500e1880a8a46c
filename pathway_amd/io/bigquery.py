"""pw.io.bigquery — BigQuery sink over the tabledata.insertAll REST API.

Reference: src/connectors/data_storage/bigquery.rs (gcp-bigquery-client).
Posts standard streaming-insert requests; credentials are a bearer token
or a service-account object with a token attribute.
"""

from __future__ import annotations

from typing import Any

from pathway_amd.io import _rest

DEFAULT_BASE = "https://bigquery.googleapis.com/bigquery/v2"


def write(
    table,
    dataset_name: str,
    table_name: str,
    *,
    project_id: str | None = None,
    service_user_credentials_file: str | None = None,
    credentials: Any = None,
    base_url: str = DEFAULT_BASE,
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.json import Json
    from pathway_amd.internals.rungraph import G

    if project_id is None and service_user_credentials_file:
        import json as _json

        with open(service_user_credentials_file) as f:
            project_id = _json.load(f).get("project_id")
    headers = {}
    token = getattr(credentials, "token", None) or (
        credentials if isinstance(credentials, str) else None
    )
    if token:
        headers["Authorization"] = f"Bearer {token}"
    url = (f"{base_url}/projects/{project_id}/datasets/{dataset_name}"
           f"/tables/{table_name}/insertAll")
    names = table.column_names()

    def writer(batch):
        rows = []
        for key, values, time, diff in batch.rows():
            rec = {}
            for n, v in zip(names, values):
                rec[n] = v.value if isinstance(v, Json) else v
            rec["time"] = time
            rec["diff"] = diff
            rows.append({"insertId": f"{key!r}-{time}-{diff}", "json": rec})
        if rows:
            out = _rest.request(
                "POST", url, body={"kind": "bigquery#tableDataInsertAllRequest",
                                   "rows": rows},
                headers=headers,
            )
            if out and out.get("insertErrors"):
                raise RuntimeError(f"bigquery insert errors: {out['insertErrors']}")

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
