"""Azure Blob, Airbyte-protocol, and PyFilesystem connectors."""

import json
import os
import sys
import textwrap

import pytest

import pathway_amd as pw
from pathway_amd.internals.rungraph import G
from pathway_amd.internals.schema import schema_from_types


def test_azure_blob_roundtrip():
    from pathway_amd.io.azure import AzureBlobClient
    from tests.fakes.fake_azure import FakeAzureBlob

    srv = FakeAzureBlob().start()
    try:
        c = AzureBlobClient(srv.url, "cont", sas_token="sv=2024&sig=x")
        c.put_blob("in/a.txt", b"one\ntwo\n")
        c.put_blob("in/b.txt", b"three\n")
        assert c.get_blob("in/a.txt") == b"one\ntwo\n"
        assert [n for n, _ in c.list_blobs("in/")] == ["in/a.txt", "in/b.txt"]

        G.clear()
        t = pw.io.azure.read(
            "in/", account_url=srv.url, container="cont",
            format="plaintext", mode="static",
        )
        keys, cols = pw.debug.table_to_dicts(t)
        assert sorted(cols["data"].values()) == ["one", "three", "two"]

        G.clear()
        t2 = pw.debug.table_from_markdown(
            """
            a
            5
            """
        )
        pw.io.azure.write(t2, "out/", account_url=srv.url, container="cont")
        pw.run()
        outs = [n for n, _ in c.list_blobs("out/")]
        assert outs
        rec = json.loads(c.get_blob(outs[0]).decode().splitlines()[0])
        assert rec["a"] == 5
    finally:
        srv.stop()


def test_airbyte_exec_source(tmp_path):
    # a minimal Airbyte source speaking the real protocol on stdout
    src_py = tmp_path / "fake_source.py"
    src_py.write_text(textwrap.dedent("""
        import json, sys
        args = sys.argv[1:]
        cfg_path = args[args.index("--config") + 1]
        cfg = json.load(open(cfg_path))
        state = None
        if "--state" in args:
            state = json.load(open(args[args.index("--state") + 1]))
        start = (state or {}).get("cursor", 0)
        for i in range(start, start + cfg.get("count", 3)):
            print(json.dumps({"type": "RECORD", "record": {
                "stream": "items", "data": {"i": i}, "emitted_at": 0}}))
        print(json.dumps({"type": "STATE", "state": {"cursor": start + cfg.get("count", 3)}}))
    """))
    config = {
        "source": {
            "exec": [sys.executable, str(src_py)],
            "config": {"count": 3},
            "streams": ["items"],
        }
    }
    G.clear()
    t = pw.io.airbyte.read(config, mode="streaming",
                           refresh_interval_ms=100, _max_runs=2)
    res = t.groupby().reduce(c=pw.reducers.count())
    keys, cols = pw.debug.table_to_dicts(res)
    # two runs: 0..2 then (state cursor=3) 3..5 -> 6 unique records
    assert list(cols["c"].values()) == [6]


def test_airbyte_yaml_config(tmp_path):
    src_py = tmp_path / "s.py"
    src_py.write_text(
        'import json\n'
        'print(json.dumps({"type": "RECORD", "record": '
        '{"stream": "s1", "data": {"x": 1}}}))\n'
    )
    cfg = tmp_path / "conf.yaml"
    cfg.write_text(
        f"source:\n  exec: {sys.executable} {src_py}\n  config: {{}}\n"
        f"  streams: [s1]\n"
    )
    G.clear()
    t = pw.io.airbyte.read(str(cfg), mode="static")
    keys, cols = pw.debug.table_to_dicts(t)
    assert list(cols["stream"].values()) == ["s1"]
    [data] = list(cols["data"].values())
    assert data.value == {"x": 1}


def test_pyfilesystem_read(tmp_path):
    d = tmp_path / "data"
    d.mkdir()
    (d / "x.bin").write_bytes(b"\x01\x02")
    (d / "y.bin").write_bytes(b"\x03")
    G.clear()
    t = pw.io.pyfilesystem.read(f"osfs://{d}", format="binary", mode="static")
    keys, cols = pw.debug.table_to_dicts(t)
    assert sorted(cols["data"].values()) == [b"\x01\x02", b"\x03"]


def test_rabbitmq_roundtrip():
    import json
    import threading
    import time

    from pathway_amd.io._amqp_protocol import AmqpClient
    from tests.fakes.fake_rabbitmq import FakeRabbit

    srv = FakeRabbit().start()
    try:
        # raw client pub/consume
        sub = AmqpClient(port=srv.port)
        sub.queue_declare("q1")
        sub.consume("q1")
        pub = AmqpClient(port=srv.port)
        pub.queue_declare("q1")
        pub.publish("q1", b"hello")
        rk, body = sub.next_delivery()
        assert (rk, body) == ("q1", b"hello")
        pub.close()
        sub.close()

        # table write -> broker
        G.clear()
        t = pw.debug.table_from_markdown(
            """
            a | b
            1 | x
            2 | y
            """
        )
        pw.io.rabbitmq.write(t, f"amqp://127.0.0.1:{srv.port}", "rows")
        pw.run()
        deadline = time.time() + 5
        while time.time() < deadline and sum(
            1 for k, _ in srv.published if k == "rows"
        ) < 2:
            time.sleep(0.05)
        recs = [json.loads(b) for k, b in srv.published if k == "rows"]
        assert sorted((r["a"], r["b"]) for r in recs) == [(1, "x"), (2, "y")]

        # streaming read
        G.clear()
        tbl = pw.io.rabbitmq.read(
            f"amqp://127.0.0.1:{srv.port}", "live",
            schema=schema_from_types(k=int), format="json", _max_messages=2,
        )

        def later():
            time.sleep(0.3)
            c = AmqpClient(port=srv.port)
            c.publish("live", json.dumps({"k": 5}).encode())
            c.publish("live", json.dumps({"k": 7}).encode())
            c.close()

        threading.Thread(target=later, daemon=True).start()
        keys, cols = pw.debug.table_to_dicts(tbl)
        assert sorted(cols["k"].values()) == [5, 7]
    finally:
        srv.stop()


def test_pulsar_roundtrip():
    import json
    import threading
    import time

    pytest.importorskip("aiohttp")
    from pathway_amd.io.pulsar import PulsarWsProducer
    from tests.fakes.fake_pulsar import FakePulsar

    srv = FakePulsar().start()
    try:
        # producer protocol
        p = PulsarWsProducer(srv.url, "t1")
        p.send(b"hello")
        p.close()
        tpath = "persistent/public/default/t1"
        assert srv.messages.get(tpath) == [b"hello"]

        # table write
        G.clear()
        t = pw.debug.table_from_markdown(
            """
            a | b
            1 | x
            2 | y
            """
        )
        pw.io.pulsar.write(t, srv.url, "rows")
        pw.run()
        recs = [json.loads(m) for m in
                srv.messages.get("persistent/public/default/rows", [])]
        assert sorted((r["a"], r["b"]) for r in recs) == [(1, "x"), (2, "y")]

        # streaming read (backlog + live)
        G.clear()
        tbl = pw.io.pulsar.read(
            srv.url, "live", schema=schema_from_types(k=int), format="json",
            subscription="sub1", _max_messages=2,
        )

        def later():
            time.sleep(0.3)
            p2 = PulsarWsProducer(srv.url, "live")
            p2.send(json.dumps({"k": 5}).encode())
            p2.send(json.dumps({"k": 7}).encode())
            p2.close()

        threading.Thread(target=later, daemon=True).start()
        keys, cols = pw.debug.table_to_dicts(tbl)
        assert sorted(cols["k"].values()) == [5, 7]
    finally:
        srv.stop()
