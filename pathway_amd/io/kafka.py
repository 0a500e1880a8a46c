"""pw.io.kafka (reference io/kafka, engine kafka.rs:777LoC).

API-parity surface; needs a kafka client (confluent_kafka/kafka-python),
not installed in this offline image — read/write raise at call time.
"""
from __future__ import annotations

from typing import Any

from pathway_amd.io._utils import require_client


def read(
    rdkafka_settings: dict,
    topic: str | list[str] | None = None,
    *,
    schema=None,
    format: str = "raw",
    autocommit_duration_ms: int | None = 1500,
    json_field_paths: dict | None = None,
    parallel_readers: int | None = None,
    persistent_id: str | None = None,
    name: str | None = None,
    mode: str = "streaming",
    with_metadata: bool = False,
    start_from_timestamp_ms: int | None = None,
    **kwargs: Any,
):
    kafka = require_client("confluent_kafka", "kafka")
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import StreamingSource, spawn_reader
    from pathway_amd.internals import dtype as dt
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.schema import schema_from_types
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe
    import json as _json

    if schema is None:
        schema = schema_from_types(data=bytes)
    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]
    src = StreamingSource(names, dtypes, name=name)
    topics = [topic] if isinstance(topic, str) else list(topic or [])

    def reader():
        consumer = kafka.Consumer(rdkafka_settings)
        consumer.subscribe(topics)
        try:
            while True:
                msg = consumer.poll(0.2)
                if msg is None:
                    continue
                if msg.error():
                    continue
                payload = msg.value()
                if format == "raw":
                    src.emit([payload])
                elif format == "json":
                    rec = _json.loads(payload)
                    src.emit([rec.get(n) for n in names])
                elif format == "plaintext":
                    src.emit([payload.decode()])
        finally:
            consumer.close()
            src.finish()

    spawn_reader(reader)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


def write(
    table,
    rdkafka_settings: dict,
    topic_name: str,
    *,
    format: str = "json",
    name: str | None = None,
    **kwargs: Any,
):
    kafka = require_client("confluent_kafka", "kafka")
    import json as _json

    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    producer = kafka.Producer(rdkafka_settings)
    names = table.column_names()

    def writer(batch):
        for key, values, time, diff in batch.rows():
            rec = dict(zip(names, values))
            rec["time"] = time
            rec["diff"] = diff
            producer.produce(topic_name, _json.dumps(rec, default=str).encode())
        producer.flush()

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node


def simple_read(server: str, topic: str, *, format: str = "raw", **kwargs):
    """One-call raw Kafka reader (reference io/kafka simple_read)."""
    return read(
        {"bootstrap.servers": server, "group.id": "pathway-simple", "auto.offset.reset": "beginning"},
        topic=topic,
        format=format,
        **kwargs,
    )
