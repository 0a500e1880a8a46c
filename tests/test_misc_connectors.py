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
