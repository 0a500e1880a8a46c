"""Kafka connector: wire-protocol client + reader/writer against an
in-process fake broker (tests/fakes/fake_kafka.py) — exercises the real
framing, RecordBatch v2 encode/decode, CRC-32C, offsets and seek.

Reference behavior: src/connectors/data_storage/kafka.rs.
"""

import json
import time

import pytest

import pathway_amd as pw
from pathway_amd.io._kafka_protocol import (
    KafkaClient,
    crc32c,
    decode_record_batches,
    encode_record_batch,
)
from tests.fakes.fake_kafka import FakeKafkaBroker


@pytest.fixture()
def broker():
    b = FakeKafkaBroker(num_partitions=2).start()
    yield b
    b.stop()


def test_crc32c_known_vectors():
    # RFC 3720 test vector: 32 bytes of zeros -> 0x8A9136AA
    assert crc32c(b"\x00" * 32) == 0x8A9136AA
    assert crc32c(b"123456789") == 0xE3069283


def test_record_batch_roundtrip():
    recs = [(b"k1", b"v1"), (None, b"v2"), (b"k3", None)]
    batch = encode_record_batch(42, recs, timestamp_ms=1700000000000)
    got = decode_record_batches(batch)
    assert [(o, k, v) for o, k, v, _ in got] == [
        (42, b"k1", b"v1"),
        (43, None, b"v2"),
        (44, b"k3", None),
    ]
    assert all(ts == 1700000000000 for _, _, _, ts in got)


def test_client_produce_fetch_offsets(broker):
    c = KafkaClient(broker.bootstrap)
    assert c.partitions("t1") == [0, 1]
    base = c.produce("t1", 0, [(None, b"a"), (None, b"b")])
    assert base == 0
    base2 = c.produce("t1", 0, [(b"k", b"c")])
    assert base2 == 2
    hw, recs = c.fetch("t1", 0, 0)
    assert hw == 3
    assert [v for _, _, v, _ in recs] == [b"a", b"b", b"c"]
    # fetch from mid-offset (seek)
    _, recs2 = c.fetch("t1", 0, 2)
    assert [v for _, _, v, _ in recs2] == [b"c"]
    assert c.list_offsets("t1", 0, -2) == 0
    assert c.list_offsets("t1", 0, -1) == 3
    c.close()


def _run_static_read(**kw):
    from pathway_amd.internals.rungraph import G

    G.clear()
    t = pw.io.kafka.read(**kw)
    return pw.debug.table_to_pandas(t)


def test_kafka_read_static_json(broker):
    for i in range(6):
        part = i % 2
        broker.seed(
            "rows", part, [(None, json.dumps({"k": i, "v": f"s{i}"}).encode())]
        )

    class S(pw.Schema):
        k: int
        v: str

    df = _run_static_read(
        rdkafka_settings={"bootstrap.servers": broker.bootstrap,
                          "auto.offset.reset": "beginning"},
        topic="rows",
        schema=S,
        format="json",
        mode="static",
    )
    assert sorted(zip(df["k"], df["v"])) == [(i, f"s{i}") for i in range(6)]


def test_kafka_read_seek_from_offsets(broker):
    for i in range(4):
        broker.seed("seekt", 0, [(None, json.dumps({"k": i}).encode())])

    class S(pw.Schema):
        k: int

    df = _run_static_read(
        rdkafka_settings={"bootstrap.servers": broker.bootstrap},
        topic="seekt",
        schema=S,
        format="json",
        mode="static",
        start_from_offsets={("seekt", 0): 2, ("seekt", 1): 0},
    )
    assert sorted(df["k"]) == [2, 3]


def test_kafka_read_debezium_retracts(broker):
    from pathway_amd.io.formats import debezium as dbz

    broker.seed("cdc", 0, [
        (None, dbz.format_message(None, {"id": 1, "v": "a"})),
        (None, dbz.format_message(None, {"id": 2, "v": "b"})),
        (None, dbz.format_message({"id": 1, "v": "a"}, {"id": 1, "v": "a2"})),
        (None, dbz.format_message({"id": 2, "v": "b"}, None)),
    ])

    class S(pw.Schema):
        id: int
        v: str

    from pathway_amd.internals.rungraph import G

    G.clear()
    t = pw.io.kafka.read(
        rdkafka_settings={"bootstrap.servers": broker.bootstrap},
        topic="cdc",
        schema=S,
        format="debezium",
        mode="static",
        primary_key=["id"],
    )
    keys, cols = pw.debug.table_to_dicts(t)
    # id=2 deleted; id=1 upserted (the -before/+after pair leaves one row)
    rows = sorted((cols["id"][k], cols["v"][k]) for k in keys)
    assert rows == [(1, "a2")]


def test_kafka_write_json_and_read_back(broker):
    from pathway_amd.internals.rungraph import G

    G.clear()
    t = pw.debug.table_from_markdown(
        """
        a | b
        1 | x
        2 | y
        """
    )
    pw.io.kafka.write(
        t, {"bootstrap.servers": broker.bootstrap}, "out_topic", format="json"
    )
    pw.run()
    vals = [json.loads(v) for v in broker.all_values("out_topic")]
    assert sorted((r["a"], r["b"]) for r in vals) == [(1, "x"), (2, "y")]
    assert all(r["diff"] == 1 for r in vals)


def test_kafka_write_avro_with_registry(broker):
    from tests.fakes.fake_registry import FakeSchemaRegistry
    from pathway_amd.io.formats import avro as _avro
    from pathway_amd.io.formats.registry import SchemaRegistryClient

    reg = FakeSchemaRegistry().start()
    try:
        from pathway_amd.internals.rungraph import G

        G.clear()
        t = pw.debug.table_from_markdown(
            """
            a | b
            5 | p
            """
        )
        pw.io.kafka.write(
            t, {"bootstrap.servers": broker.bootstrap}, "avro_topic",
            format="avro", schema_registry_settings=reg.url,
        )
        pw.run()
        [wire] = broker.all_values("avro_topic")
        sid, payload = _avro.confluent_decode(wire)
        sch = SchemaRegistryClient(reg.url).get_schema(sid)
        rec = _avro.decode_bytes(payload, sch)
        assert rec["a"] == 5 and rec["b"] == "p" and rec["diff"] == 1
    finally:
        reg.stop()


def test_kafka_streaming_live_append(broker):
    """Streaming mode: rows produced after the read starts arrive."""
    class S(pw.Schema):
        k: int

    from pathway_amd.internals.rungraph import G

    G.clear()
    broker.seed("live", 0, [(None, json.dumps({"k": 0}).encode())])
    t = pw.io.kafka.read(
        rdkafka_settings={"bootstrap.servers": broker.bootstrap},
        topic="live",
        schema=S,
        format="json",
        mode="streaming",
        _max_polls=12,
    )
    seen = []
    pw.io.subscribe(t, lambda key, row, time, is_addition: seen.append(row["k"]))

    import threading

    def late_producer():
        time.sleep(0.3)
        c = KafkaClient(broker.bootstrap)
        c.produce("live", 0, [(None, json.dumps({"k": 1}).encode())])
        c.close()

    threading.Thread(target=late_producer, daemon=True).start()
    pw.run()
    assert sorted(seen) == [0, 1]
