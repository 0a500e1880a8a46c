"""Data-format codecs: avro, bson, debezium, schema registry.

Reference semantics: src/connectors/data_format/{avro,bson,debezium}.rs.
"""

import datetime
import io
import json

import pytest

from pathway_amd.io.formats import avro, bson, debezium
from pathway_amd.io.formats.registry import SchemaRegistryClient

RECORD_SCHEMA = {
    "type": "record",
    "name": "Row",
    "fields": [
        {"name": "id", "type": "long"},
        {"name": "name", "type": "string"},
        {"name": "score", "type": "double"},
        {"name": "active", "type": "boolean"},
        {"name": "tags", "type": {"type": "array", "items": "string"}},
        {"name": "attrs", "type": {"type": "map", "values": "long"}},
        {"name": "maybe", "type": ["null", "string"]},
        {"name": "blob", "type": "bytes"},
    ],
}

ROW = {
    "id": 123456789012345,
    "name": "żółć utf8 ✓",
    "score": -1.5,
    "active": True,
    "tags": ["a", "b", "c"],
    "attrs": {"x": 1, "y": -2},
    "maybe": None,
    "blob": b"\x00\x01\xff",
}


def test_avro_roundtrip_record():
    data = avro.encode_bytes(ROW, RECORD_SCHEMA)
    back = avro.decode_bytes(data, RECORD_SCHEMA)
    assert back == ROW


def test_avro_union_and_negative_varints():
    schema = ["null", "long", "string"]
    for v in (None, 0, -1, 1, -(2**40), 2**40, "s"):
        assert avro.decode_bytes(avro.encode_bytes(v, schema), schema) == v


def test_avro_zigzag_known_bytes():
    # spec examples: 0->00, -1->01, 1->02, -2->03, 2->04
    for v, b in [(0, b"\x00"), (-1, b"\x01"), (1, b"\x02"), (-2, b"\x03"), (2, b"\x04")]:
        assert avro.encode_bytes(v, "long") == b


def test_avro_enum_fixed_nested():
    schema = {
        "type": "record",
        "name": "N",
        "fields": [
            {"name": "color", "type": {"type": "enum", "name": "C", "symbols": ["R", "G", "B"]}},
            {"name": "mac", "type": {"type": "fixed", "name": "F", "size": 4}},
            {"name": "child", "type": ["null", "N"]},
        ],
    }
    v = {"color": "G", "mac": b"\x01\x02\x03\x04",
         "child": {"color": "B", "mac": b"\xff\xff\xff\xff", "child": None}}
    assert avro.decode_bytes(avro.encode_bytes(v, schema), schema) == v


@pytest.mark.parametrize("codec", ["null", "deflate"])
def test_avro_container_file(tmp_path, codec):
    p = tmp_path / "rows.avro"
    with open(p, "wb") as f:
        w = avro.ContainerWriter(f, RECORD_SCHEMA, codec=codec)
        rows = []
        for i in range(250):
            r = dict(ROW, id=i, maybe=("x" if i % 2 else None))
            rows.append(r)
            w.append(r)
        w.close()
    with open(p, "rb") as f:
        back = list(avro.read_container(f))
    assert back == rows


def test_confluent_framing_and_registry():
    from tests.fakes.fake_registry import FakeSchemaRegistry

    reg = FakeSchemaRegistry().start()
    try:
        client = SchemaRegistryClient(reg.url)
        sid = client.register("rows-value", RECORD_SCHEMA)
        assert client.latest("rows-value") == (sid, RECORD_SCHEMA)
        wire = avro.confluent_encode(ROW, RECORD_SCHEMA, sid)
        got_id, payload = avro.confluent_decode(wire)
        assert got_id == sid
        schema = SchemaRegistryClient(reg.url).get_schema(got_id)
        assert avro.decode_bytes(payload, schema) == ROW
    finally:
        reg.stop()


def test_bson_roundtrip():
    doc = {
        "i32": 42,
        "i64": 2**40,
        "neg": -7,
        "f": 3.25,
        "s": "héllo",
        "b": True,
        "none": None,
        "bin": b"\x00\xff",
        "sub": {"a": 1, "b": [1, "two", None]},
        "arr": [1.5, {"x": 1}],
        "oid": bson.ObjectId(),
        "dt": datetime.datetime(2024, 5, 1, 12, 0, tzinfo=datetime.timezone.utc),
    }
    data = bson.encode(doc)
    back = bson.decode(data)
    assert back == doc


def test_bson_decode_all_stream():
    docs = [{"n": i} for i in range(5)]
    blob = b"".join(bson.encode(d) for d in docs)
    assert bson.decode_all(blob) == docs


def test_bson_objectid_unique_and_hex():
    a, b = bson.ObjectId(), bson.ObjectId()
    assert a != b
    assert bson.ObjectId(str(a)) == a


def test_debezium_ops():
    ins = debezium.parse_message(debezium.format_message(None, {"id": 1, "v": "a"}))
    assert [(e.values, e.diff) for e in ins] == [({"id": 1, "v": "a"}, 1)]

    upd = debezium.parse_message(
        debezium.format_message({"id": 1, "v": "a"}, {"id": 1, "v": "b"})
    )
    assert [(e.values, e.diff) for e in upd] == [
        ({"id": 1, "v": "a"}, -1),
        ({"id": 1, "v": "b"}, 1),
    ]

    dele = debezium.parse_message(debezium.format_message({"id": 1, "v": "b"}, None))
    assert [(e.values, e.diff) for e in dele] == [({"id": 1, "v": "b"}, -1)]


def test_debezium_connect_envelope_and_keys():
    msg = json.dumps(
        {"schema": {"type": "struct"},
         "payload": {"before": None, "after": {"id": 9, "v": "z"}, "op": "r",
                     "source": {}, "ts_ms": 1700000000000}}
    ).encode()
    key = json.dumps({"schema": {}, "payload": {"id": 9}}).encode()
    evs = debezium.parse_message(msg, key)
    assert evs[0].key == (9,)
    assert evs[0].diff == 1
    assert evs[0].ts_ms == 1700000000000

    evs2 = debezium.parse_message(msg, primary_key=["id"])
    assert evs2[0].key == (9,)


def test_debezium_tombstone_ignored():
    assert debezium.parse_message(None) == []
    assert debezium.parse_message(b"") == []
