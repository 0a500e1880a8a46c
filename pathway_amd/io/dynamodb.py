"""pw.io.dynamodb — DynamoDB sink over the AWS JSON 1.0 HTTP API.

Reference: src/connectors/data_storage/dynamodb.rs (aws-sdk).  Emits
standard ``X-Amz-Target: DynamoDB_20120810.*`` requests (PutItem /
DeleteItem / CreateTable) with DynamoDB attribute-value encoding; +diff
rows upsert, -diff rows delete by primary key.
"""

from __future__ import annotations

from typing import Any

from pathway_amd.io import _rest

TARGET_PREFIX = "DynamoDB_20120810"


def _attr(v: Any) -> dict:
    if v is None:
        return {"NULL": True}
    if isinstance(v, bool):
        return {"BOOL": v}
    if isinstance(v, (int, float)):
        return {"N": str(v)}
    if isinstance(v, bytes):
        import base64

        return {"B": base64.b64encode(v).decode()}
    if isinstance(v, (list, tuple)):
        return {"L": [_attr(x) for x in v]}
    if isinstance(v, dict):
        return {"M": {k: _attr(x) for k, x in v.items()}}
    return {"S": str(v)}


class _Api:
    def __init__(self, endpoint: str, region: str = "us-east-1"):
        self.endpoint = endpoint or f"https://dynamodb.{region}.amazonaws.com"

    def call(self, op: str, body: dict) -> Any:
        return _rest.request(
            "POST", self.endpoint, body=body,
            headers={"X-Amz-Target": f"{TARGET_PREFIX}.{op}",
                     "Content-Type": "application/x-amz-json-1.0"},
            content_type="application/x-amz-json-1.0",
        )


def write(
    table,
    table_name: str,
    partition_key: str,
    sort_key: str | None = None,
    *,
    endpoint: str | None = None,
    region: str = "us-east-1",
    init_mode: str = "default",
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.json import Json
    from pathway_amd.internals.rungraph import G

    api = _Api(endpoint, region)
    names = table.column_names()
    if init_mode in ("create_if_not_exists", "replace"):
        keyschema = [{"AttributeName": partition_key, "KeyType": "HASH"}]
        attrs = [{"AttributeName": partition_key, "AttributeType": "S"}]
        if sort_key:
            keyschema.append({"AttributeName": sort_key, "KeyType": "RANGE"})
            attrs.append({"AttributeName": sort_key, "AttributeType": "S"})
        try:
            api.call("CreateTable", {
                "TableName": table_name, "KeySchema": keyschema,
                "AttributeDefinitions": attrs,
                "BillingMode": "PAY_PER_REQUEST",
            })
        except _rest.RestError:
            pass  # already exists

    def writer(batch):
        for _key, values, time, diff in batch.rows():
            rec = dict(zip(names, values))
            rec = {k: (v.value if isinstance(v, Json) else v)
                   for k, v in rec.items()}
            if diff > 0:
                item = {k: _attr(v) for k, v in rec.items()}
                item["time"] = _attr(time)
                item["diff"] = _attr(diff)
                api.call("PutItem", {"TableName": table_name, "Item": item})
            else:
                key = {partition_key: _attr(rec[partition_key])}
                if sort_key:
                    key[sort_key] = _attr(rec[sort_key])
                api.call("DeleteItem", {"TableName": table_name, "Key": key})

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
