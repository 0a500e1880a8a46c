"""pw.io.kinesis — Kinesis connector over the AWS JSON 1.1 HTTP API.

Reference: src/connectors/data_storage/kinesis.rs (654 LoC, aws-sdk).
write(): PutRecords batches.  read(): GetShardIterator + GetRecords
polling per shard with sequence-number offsets (seekable).
"""

from __future__ import annotations

import base64
import json as _json
import time as _time
from typing import Any

from pathway_amd.io import _rest

TARGET_PREFIX = "Kinesis_20131202"


class _Api:
    def __init__(self, endpoint: str | None, region: str = "us-east-1"):
        self.endpoint = endpoint or f"https://kinesis.{region}.amazonaws.com"

    def call(self, op: str, body: dict) -> Any:
        return _rest.request(
            "POST", self.endpoint, body=body,
            headers={"X-Amz-Target": f"{TARGET_PREFIX}.{op}",
                     "Content-Type": "application/x-amz-json-1.1"},
            content_type="application/x-amz-json-1.1",
        )


class KinesisReader:
    def __init__(self, source, api: _Api, stream_name: str, parse, *,
                 max_polls: int | None = None, poll_interval: float = 0.2):
        self.source = source
        self.api = api
        self.stream_name = stream_name
        self.parse = parse
        self.max_polls = max_polls
        self.poll_interval = poll_interval

    def run(self) -> None:
        try:
            desc = self.api.call(
                "DescribeStream", {"StreamName": self.stream_name}
            )
            shards = [s["ShardId"]
                      for s in desc["StreamDescription"]["Shards"]]
            iters = {}
            for sid in shards:
                out = self.api.call("GetShardIterator", {
                    "StreamName": self.stream_name, "ShardId": sid,
                    "ShardIteratorType": "TRIM_HORIZON",
                })
                iters[sid] = out["ShardIterator"]
            polls = 0
            while True:
                got = False
                for sid, it in list(iters.items()):
                    out = self.api.call("GetRecords", {"ShardIterator": it})
                    for rec in out.get("Records", []):
                        data = base64.b64decode(rec["Data"])
                        for values, diff in self.parse(data):
                            self.source.emit(values, diff=diff)
                        got = True
                    iters[sid] = out.get("NextShardIterator", it)
                polls += 1
                if self.max_polls is not None and polls >= self.max_polls:
                    return
                if not got:
                    _time.sleep(self.poll_interval)
        except Exception as e:
            self.source.fail(e)
        finally:
            self.source.finish()


def read(
    stream_name: str,
    *,
    schema=None,
    format: str = "raw",
    endpoint: str | None = None,
    region: str = "us-east-1",
    mode: str = "streaming",
    name: str | None = None,
    _max_polls: int | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import StreamingSource, spawn_reader
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.schema import schema_from_types
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    if schema is None:
        schema = schema_from_types(data=bytes if format == "raw" else str)
    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]

    def parse(payload: bytes):
        if format == "raw":
            return [([payload], 1)]
        if format == "plaintext":
            return [([payload.decode("utf-8", "replace")], 1)]
        if format == "json":
            rec = _json.loads(payload)
            return [([rec.get(n) for n in names], 1)]
        raise ValueError(f"unsupported kinesis format {format!r}")

    src = StreamingSource(names, dtypes, name=name)
    reader = KinesisReader(src, _Api(endpoint, region), stream_name, parse,
                           max_polls=_max_polls)
    src.reader = reader
    spawn_reader(reader.run, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


def write(
    table,
    stream_name: str,
    *,
    endpoint: str | None = None,
    region: str = "us-east-1",
    format: str = "json",
    partition_key: str | None = None,
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    api = _Api(endpoint, region)
    names = table.column_names()

    def writer(batch):
        records = []
        for key, values, time, diff in batch.rows():
            rec = dict(zip(names, values))
            rec["time"] = time
            rec["diff"] = diff
            pk = str(rec.get(partition_key)) if partition_key else repr(key)
            records.append({
                "Data": base64.b64encode(
                    _json.dumps(rec, default=str).encode()
                ).decode(),
                "PartitionKey": pk,
            })
        if records:
            api.call("PutRecords",
                     {"StreamName": stream_name, "Records": records})

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
