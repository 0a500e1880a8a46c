"""pw.io.redpanda — Redpanda connector (reference io/redpanda).

Redpanda speaks the Kafka wire protocol; this module delegates to
pw.io.kafka's pure-python protocol client (RecordBatch v2, offsets,
seek) with identical signatures.
"""

from __future__ import annotations

from typing import Any

from pathway_amd.io import kafka as _kafka


def read(rdkafka_settings: dict, topic: str | list[str] | None = None,
         **kwargs: Any):
    return _kafka.read(rdkafka_settings, topic, **kwargs)


def write(table, rdkafka_settings: dict, topic_name: str, **kwargs: Any):
    return _kafka.write(table, rdkafka_settings, topic_name, **kwargs)


def simple_read(server: str, topic: str, **kwargs: Any):
    return _kafka.simple_read(server, topic, **kwargs)
