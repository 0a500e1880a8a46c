"""pw.io.debezium (reference io/debezium) — CDC envelopes over a kafka
transport; parses the debezium change-event format."""
from __future__ import annotations

from typing import Any


def parse_debezium_event(payload: dict, names: list[str]):
    """Returns (values, diff_events) from a debezium envelope
    (reference data_format debezium parser)."""
    p = payload.get("payload", payload)
    op = p.get("op")
    out = []
    if op in ("c", "r"):
        out.append(([p["after"].get(n) for n in names], 1))
    elif op == "u":
        out.append(([p["before"].get(n) for n in names], -1))
        out.append(([p["after"].get(n) for n in names], 1))
    elif op == "d":
        out.append(([p["before"].get(n) for n in names], -1))
    return out


def read(rdkafka_settings: dict, topic_name: str, *, schema=None, autocommit_duration_ms=1500, name=None, **kwargs):
    from pathway_amd.io._utils import require_client

    kafka = require_client("confluent_kafka", "debezium")
    import json as _json

    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import StreamingSource, spawn_reader
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]
    src = StreamingSource(names, dtypes, name=name)

    def reader():
        consumer = kafka.Consumer(rdkafka_settings)
        consumer.subscribe([topic_name])
        try:
            while True:
                msg = consumer.poll(0.2)
                if msg is None or msg.error():
                    continue
                for values, diff in parse_debezium_event(_json.loads(msg.value()), names):
                    src.emit(values, diff=diff)
        finally:
            src.finish()

    spawn_reader(reader, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())
