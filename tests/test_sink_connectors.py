"""HTTP/TCP sink+broker connectors against protocol-level fakes:
elasticsearch (_bulk ndjson), logstash, clickhouse (JSONEachRow),
questdb (ILP/TCP), nats (text protocol), mqtt (3.1.1 binary)."""

import json
import socketserver
import threading
import time

import pytest

import pathway_amd as pw
from pathway_amd.internals.rungraph import G
from tests.fakes.fake_http import FakeHTTPService


@pytest.fixture()
def http():
    s = FakeHTTPService().start()
    yield s
    s.stop()


def _t():
    G.clear()
    return pw.debug.table_from_markdown(
        """
        a | b
        1 | x
        2 | y
        """
    )


def test_elasticsearch_bulk_write(http):
    t = _t()
    auth = pw.io.elasticsearch.ElasticSearchAuth.basic("u", "p")
    pw.io.elasticsearch.write(t, http.url, auth, index_name="idx")
    pw.run()
    [req] = [r for r in http.requests if r.path == "/_bulk"]
    assert req.headers.get("Authorization", "").startswith("Basic ")
    lines = req.ndjson()
    actions = [l for l in lines if "index" in l]
    docs = [l for l in lines if "a" in l]
    assert len(actions) == 2 and len(docs) == 2
    assert all(a["index"]["_index"] == "idx" for a in actions)
    assert sorted((d["a"], d["b"]) for d in docs) == [(1, "x"), (2, "y")]


def test_elasticsearch_retraction_deletes(http):
    G.clear()
    t = pw.debug.table_from_markdown(
        """
        id | a | __time__ | __diff__
        7  | 1 | 2        | 1
        7  | 1 | 4        | -1
        """
    )
    pw.io.elasticsearch.write(t, http.url, None, index_name="idx")
    pw.run()
    lines = []
    for r in http.requests:
        if r.path == "/_bulk":
            lines += r.ndjson()
    idx = [l for l in lines if "index" in l]
    dele = [l for l in lines if "delete" in l]
    assert len(idx) == 1 and len(dele) == 1
    assert idx[0]["index"]["_id"] == dele[0]["delete"]["_id"]


def test_logstash_write(http):
    t = _t()
    pw.io.logstash.write(t, http.url)
    pw.run()
    recs = [r.json() for r in http.requests]
    assert sorted((r["a"], r["b"]) for r in recs) == [(1, "x"), (2, "y")]


def test_clickhouse_write(http):
    t = _t()
    pw.io.clickhouse.write(
        t, {"host": http.url, "database": "db", "user": "u"}, "tbl"
    )
    pw.run()
    [req] = http.requests
    assert "INSERT+INTO+tbl+FORMAT+JSONEachRow" in req.path or \
        "INSERT%20INTO%20tbl%20FORMAT%20JSONEachRow" in req.path
    assert "database=db" in req.path
    recs = req.ndjson()
    assert sorted((r["a"], r["b"]) for r in recs) == [(1, "x"), (2, "y")]


def test_questdb_ilp_write():
    lines = []
    done = threading.Event()

    class Handler(socketserver.StreamRequestHandler):
        def handle(self):
            for raw in self.rfile:
                lines.append(raw.decode().rstrip("\n"))
            done.set()

    class Srv(socketserver.ThreadingTCPServer):
        allow_reuse_address = True
        daemon_threads = True

    srv = Srv(("127.0.0.1", 0), Handler)
    th = threading.Thread(target=srv.serve_forever, daemon=True)
    th.start()
    try:
        t = _t()
        pw.io.questdb.write(
            t, {"host": "127.0.0.1", "port": srv.server_address[1]}, "metrics"
        )
        pw.run()
        done.wait(5)
        assert len(lines) == 2
        assert all(l.startswith("metrics ") for l in lines)
        assert any("a=1i" in l and 'b="x"' in l for l in lines)
        assert all("diff=1i" in l for l in lines)
    finally:
        srv.shutdown()
        srv.server_close()


def test_nats_roundtrip():
    from tests.fakes.fake_nats import FakeNats
    from pathway_amd.io.nats import NatsClient

    srv = FakeNats().start()
    try:
        # raw client pub/sub
        sub = NatsClient(srv.uri)
        sub.subscribe("s1")
        pub = NatsClient(srv.uri)
        pub.publish("s1", b"hello")
        subject, payload = sub.next_message()
        assert (subject, payload) == ("s1", b"hello")
        pub.close()
        sub.close()

        # table write -> read back
        t = _t()
        pw.io.nats.write(t, srv.uri, "rows", format="json")
        pw.run()
        deadline = time.time() + 5
        while time.time() < deadline and sum(
            1 for s, _ in srv.published if s == "rows"
        ) < 2:
            time.sleep(0.05)
        recs = [json.loads(p) for s, p in srv.published if s == "rows"]
        assert sorted((r["a"], r["b"]) for r in recs) == [(1, "x"), (2, "y")]

        # streaming read
        G.clear()
        from pathway_amd.internals.schema import schema_from_types

        tbl = pw.io.nats.read(
            srv.uri, "live", schema=schema_from_types(k=int), format="json",
            _max_messages=2,
        )

        def later():
            # NATS pub/sub is at-most-once: wait for the reader's SUB to
            # land before publishing (real brokers drop subscriber-less
            # messages the same way)
            deadline = time.time() + 10
            while time.time() < deadline and not srv.subs.get("live"):
                time.sleep(0.01)
            c = NatsClient(srv.uri)
            c.publish("live", json.dumps({"k": 5}).encode())
            c.publish("live", json.dumps({"k": 7}).encode())
            c.close()

        threading.Thread(target=later, daemon=True).start()
        keys, cols = pw.debug.table_to_dicts(tbl)
        assert sorted(cols["k"].values()) == [5, 7]
    finally:
        srv.stop()


def test_mqtt_roundtrip():
    from tests.fakes.fake_mqtt import FakeMqtt
    from pathway_amd.io.mqtt import MqttClient

    srv = FakeMqtt().start()
    try:
        sub = MqttClient(srv.uri, client_id="sub1")
        sub.subscribe("top")
        pub = MqttClient(srv.uri, client_id="pub1")
        pub.publish("top", b"payload")
        topic, payload = sub.next_message()
        assert (topic, payload) == ("top", b"payload")
        pub.close()
        sub.close()

        t = _t()
        pw.io.mqtt.write(t, srv.uri, "rows", format="json")
        pw.run()
        deadline = time.time() + 5
        while time.time() < deadline and sum(
            1 for s, _ in srv.published if s == "rows"
        ) < 2:
            time.sleep(0.05)
        recs = [json.loads(p) for s, p in srv.published if s == "rows"]
        assert sorted((r["a"], r["b"]) for r in recs) == [(1, "x"), (2, "y")]

        G.clear()
        from pathway_amd.internals.schema import schema_from_types

        tbl = pw.io.mqtt.read(
            srv.uri, "live", schema=schema_from_types(k=int), format="json",
            _max_messages=2,
        )

        def later():
            deadline = time.time() + 10
            while time.time() < deadline and not srv.subs.get("live"):
                time.sleep(0.01)
            c = MqttClient(srv.uri, client_id="late")
            c.publish("live", json.dumps({"k": 1}).encode())
            c.publish("live", json.dumps({"k": 2}).encode())
            c.close()

        threading.Thread(target=later, daemon=True).start()
        keys, cols = pw.debug.table_to_dicts(tbl)
        assert sorted(cols["k"].values()) == [1, 2]
    finally:
        srv.stop()
