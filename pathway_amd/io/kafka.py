"""pw.io.kafka — Kafka connector over the pure-python wire client.

Reference semantics: python/pathway/io/kafka + src/connectors/data_storage/
kafka.rs (rdkafka reader/writer with per-partition offset tracking and
seek on recovery).  This build speaks the Kafka binary protocol directly
(io/_kafka_protocol.py) — no client library — and is exercised end-to-end
against an in-process fake broker in tests/test_kafka_connector.py.

Supported read formats: raw, plaintext, json, dsv, avro (Confluent
schema-registry framing), debezium (CDC envelope -> retraction stream).
Write formats: json, raw, plaintext, dsv, avro.
"""

from __future__ import annotations

import json as _json
import threading
import time as _time
from typing import Any

from pathway_amd.io._kafka_protocol import KafkaClient


def _bootstrap_of(rdkafka_settings: dict) -> str:
    bs = rdkafka_settings.get("bootstrap.servers")
    if not bs:
        raise ValueError("rdkafka_settings must contain 'bootstrap.servers'")
    return bs


class KafkaReader:
    """Reader thread body: polls partition leaders, tracks offsets.

    Offsets are exposed as {(topic, partition): next_offset} — the
    persistence layer stores them as the source's OffsetAntichain and
    passes them back via start_from_offsets on recovery (reference
    connectors/mod.rs:319,567 seek path).
    """

    def __init__(
        self,
        source,
        settings: dict,
        topics: list[str],
        *,
        parse,  # (key: bytes|None, value: bytes|None) -> list[(values, diff)]
        mode: str = "streaming",
        start_from_timestamp_ms: int | None = None,
        start_from_offsets: dict[tuple[str, int], int] | None = None,
        poll_interval: float = 0.05,
        max_polls: int | None = None,
    ):
        self.source = source
        self.settings = settings
        self.topics = topics
        self.parse = parse
        self.mode = mode
        self.start_from_timestamp_ms = start_from_timestamp_ms
        self.offsets: dict[tuple[str, int], int] = dict(start_from_offsets or {})
        self._stop = threading.Event()
        self.poll_interval = poll_interval
        self.max_polls = max_polls  # tests: finish after N poll loops

    def stop(self) -> None:
        self._stop.set()

    def run(self) -> None:
        client = None
        try:
            client = KafkaClient(_bootstrap_of(self.settings))
            self._run(client)
        except Exception as e:  # surface connector errors loudly
            self.source.fail(e)
        finally:
            if client is not None:
                client.close()
            self.source.finish()

    def _run(self, client: KafkaClient) -> None:
        reset = str(
            self.settings.get("auto.offset.reset", "beginning")
        ).lower()
        parts: list[tuple[str, int]] = []
        from pathway_amd import parallel as par

        comm = par.get_comm()
        world = comm.world if comm is not None else 1
        rank = comm.rank if comm is not None else 0
        for t in self.topics:
            for p in client.partitions(t):
                # multi-worker partition assignment (reference kafka.rs:
                # each partition is consumed by exactly one worker)
                if world > 1 and p % world != rank:
                    continue
                parts.append((t, p))
                if (t, p) not in self.offsets:
                    if self.start_from_timestamp_ms is not None:
                        self.offsets[(t, p)] = client.list_offsets(
                            t, p, self.start_from_timestamp_ms
                        )
                    elif reset in ("beginning", "earliest", "smallest"):
                        self.offsets[(t, p)] = client.list_offsets(t, p, -2)
                    else:
                        self.offsets[(t, p)] = client.list_offsets(t, p, -1)
        # static mode: drain up to the high watermark observed at start
        static_hw: dict[tuple[str, int], int] = {}
        if self.mode == "static":
            for t, p in parts:
                static_hw[(t, p)] = client.list_offsets(t, p, -1)
        polls = 0
        while not self._stop.is_set():
            if self.max_polls is not None:
                polls += 1
                if polls > self.max_polls:
                    return
            progress = False
            done = True
            for t, p in parts:
                off = self.offsets[(t, p)]
                if self.mode == "static" and off >= static_hw[(t, p)]:
                    continue
                hw, recs = client.fetch(t, p, off, max_wait_ms=100)
                for roff, key, value, ts in recs:
                    if self.mode == "static" and roff >= static_hw[(t, p)]:
                        break
                    for parsed in self.parse(key, value):
                        values, diff = parsed[0], parsed[1]
                        row_key = parsed[2] if len(parsed) > 2 else None
                        self.source.emit(values, key=row_key, diff=diff)
                    self.offsets[(t, p)] = roff + 1
                if recs:
                    progress = True
                if self.mode == "static" and self.offsets[(t, p)] < static_hw[(t, p)]:
                    done = False
            if self.mode == "static":
                if done:
                    return
            elif not progress:
                _time.sleep(self.poll_interval)


def _build_parser(format: str, schema, names, registry_client=None,
                  primary_key=None, csv_settings=None):
    from pathway_amd.internals import dtype as dt
    from pathway_amd.internals.json import Json

    def coerce(rec: dict) -> list:
        row = []
        for n in names:
            v = rec.get(n)
            d = dt.unoptionalize(schema.__columns__[n].dtype) if schema else None
            if isinstance(v, (dict, list)) and d == dt.JSON:
                v = Json(v)
            row.append(v)
        return row

    if format == "raw":
        return lambda key, value: [([value], 1)]
    if format == "plaintext":
        return lambda key, value: [([value.decode("utf-8", "replace")], 1)]
    if format == "json":
        def parse_json(key, value):
            rec = _json.loads(value)
            return [(coerce(rec), 1)]
        return parse_json
    if format == "dsv":
        delim = getattr(csv_settings, "delimiter", ",") if csv_settings else ","
        def parse_dsv(key, value):
            fields = value.decode().rstrip("\r\n").split(delim)
            rec = dict(zip(names, fields))
            row = []
            for n in names:
                v = rec.get(n)
                d = dt.unoptionalize(schema.__columns__[n].dtype) if schema else dt.STR
                if v is not None and d in (dt.INT,):
                    v = int(v)
                elif v is not None and d in (dt.FLOAT,):
                    v = float(v)
                elif v is not None and d in (dt.BOOL,):
                    v = v.lower() in ("true", "1")
                row.append(v)
            return [(row, 1)]
        return parse_dsv
    if format == "avro":
        from pathway_amd.io.formats import avro as _avro

        def parse_avro(key, value):
            schema_id, payload = _avro.confluent_decode(value)
            sch = registry_client.get_schema(schema_id)
            rec = _avro.decode_bytes(payload, sch)
            return [(coerce(rec), 1)]
        return parse_avro
    if format == "debezium":
        from pathway_amd.internals.api import Pointer, hash_values
        from pathway_amd.io.formats import debezium as _dbz

        def parse_dbz(key, value):
            events = _dbz.parse_message(value, key, primary_key=primary_key)
            out = []
            for e in events:
                # stable row identity from the message/primary key so the
                # -before event retracts the matching earlier +insert
                ptr = Pointer(*hash_values(list(e.key))) if e.key else None
                out.append((coerce(e.values), e.diff, ptr))
            return out
        return parse_dbz
    raise ValueError(f"unsupported kafka format {format!r}")


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
    start_from_offsets: dict | None = None,
    schema_registry_settings: Any = None,
    primary_key: list[str] | None = None,
    csv_settings: Any = None,
    max_backlog_size: int | None = None,
    topic_names: list[str] | None = None,
    _max_polls: int | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import StreamingSource, spawn_reader
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.schema import schema_from_types
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    if topic is None and topic_names:
        topic = topic_names
    topics = [topic] if isinstance(topic, str) else list(topic or [])
    if not topics:
        raise ValueError("pw.io.kafka.read needs a topic")

    if schema is None:
        if format in ("raw",):
            schema = schema_from_types(data=bytes)
        elif format == "plaintext":
            schema = schema_from_types(data=str)
        else:
            raise ValueError(f"format {format!r} requires a schema")
    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]

    registry_client = None
    if format == "avro":
        from pathway_amd.io.formats.registry import SchemaRegistryClient

        if schema_registry_settings is None:
            raise ValueError("avro format requires schema_registry_settings")
        url = (
            schema_registry_settings
            if isinstance(schema_registry_settings, str)
            else schema_registry_settings.get("url")
        )
        registry_client = SchemaRegistryClient(url)

    parse = _build_parser(
        format, schema, names, registry_client, primary_key, csv_settings
    )

    src = StreamingSource(names, dtypes, name=name, maxsize=max_backlog_size)
    reader = KafkaReader(
        src,
        rdkafka_settings,
        topics,
        parse=parse,
        mode=mode,
        start_from_timestamp_ms=start_from_timestamp_ms,
        start_from_offsets=start_from_offsets,
        max_polls=_max_polls,
    )
    src.reader = reader  # offsets exposed for persistence metadata
    spawn_reader(reader.run, src, sharded=True)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


def write(
    table,
    rdkafka_settings: dict,
    topic_name: str,
    *,
    format: str = "json",
    delimiter: str = ",",
    key=None,
    value=None,
    schema_registry_settings: Any = None,
    subject: str | None = None,
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    client = KafkaClient(_bootstrap_of(rdkafka_settings))
    names = table.column_names()
    state = {"rr": 0, "parts": None, "avro": None}

    def _partition() -> int:
        if state["parts"] is None:
            state["parts"] = client.partitions(topic_name) or [0]
        parts = state["parts"]
        state["rr"] = (state["rr"] + 1) % len(parts)
        return parts[state["rr"]]

    def _avro_ctx():
        if state["avro"] is None:
            from pathway_amd.io.formats.registry import SchemaRegistryClient

            url = (
                schema_registry_settings
                if isinstance(schema_registry_settings, str)
                else schema_registry_settings.get("url")
            )
            reg = SchemaRegistryClient(url)
            fieldschema = []
            from pathway_amd.internals import dtype as dt

            tmap = {dt.INT: "long", dt.FLOAT: "double", dt.BOOL: "boolean",
                    dt.STR: "string", dt.BYTES: "bytes"}
            for n in names:
                d = dt.unoptionalize(table._dtypes[n]) if hasattr(table, "_dtypes") else dt.STR
                fieldschema.append(
                    {"name": n, "type": ["null", tmap.get(d, "string")]}
                )
            fieldschema.append({"name": "time", "type": "long"})
            fieldschema.append({"name": "diff", "type": "long"})
            sch = {"type": "record", "name": "PwRow", "fields": fieldschema}
            sid = reg.register(subject or f"{topic_name}-value", sch)
            state["avro"] = (sch, sid)
        return state["avro"]

    def fmt_row(values, time, diff) -> bytes:
        if format == "json":
            rec = dict(zip(names, [_plain(v) for v in values]))
            rec["time"] = time
            rec["diff"] = diff
            return _json.dumps(rec, default=str).encode()
        if format in ("raw", "plaintext"):
            v = values[0]
            if isinstance(v, bytes):
                return v
            return str(v).encode()
        if format == "dsv":
            return delimiter.join(str(_plain(v)) for v in values).encode()
        if format == "avro":
            from pathway_amd.io.formats import avro as _avro

            sch, sid = _avro_ctx()
            rec = dict(zip(names, [_plain(v) for v in values]))
            rec["time"] = time
            rec["diff"] = diff
            return _avro.confluent_encode(rec, sch, sid)
        raise ValueError(f"unsupported kafka write format {format!r}")

    def writer(batch):
        records = []
        for key_, values, time, diff in batch.rows():
            records.append((None, fmt_row(values, time, diff)))
        if records:
            client.produce(topic_name, _partition(), records)

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


def simple_read(server: str, topic: str, *, format: str = "raw", **kwargs):
    """One-call raw Kafka reader (reference io/kafka simple_read)."""
    return read(
        {"bootstrap.servers": server, "group.id": "pathway-simple",
         "auto.offset.reset": "beginning"},
        topic=topic,
        format=format,
        **kwargs,
    )
