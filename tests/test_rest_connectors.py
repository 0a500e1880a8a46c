"""REST-API connectors (dynamodb, kinesis, bigquery, pubsub, slack,
vector sinks, gdrive) — request-shape tests against the capturing fake
HTTP service."""

import base64
import json

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


def test_dynamodb_put_and_delete(http):
    G.clear()
    t = pw.debug.table_from_markdown(
        """
        id | a | __time__ | __diff__
        7  | 1 | 2        | 1
        7  | 1 | 4        | -1
        """
    )
    pw.io.dynamodb.write(
        t, "tbl", partition_key="a", endpoint=http.url,
        init_mode="create_if_not_exists",
    )
    pw.run()
    targets = [r.headers.get("X-Amz-Target") for r in http.requests]
    assert "DynamoDB_20120810.CreateTable" in targets
    puts = [r for r in http.requests
            if r.headers.get("X-Amz-Target", "").endswith("PutItem")]
    dels = [r for r in http.requests
            if r.headers.get("X-Amz-Target", "").endswith("DeleteItem")]
    assert len(puts) == 1 and len(dels) == 1
    item = puts[0].json()["Item"]
    assert item["a"] == {"N": "1"}
    assert dels[0].json()["Key"]["a"] == {"N": "1"}


def test_kinesis_write_and_read(http):
    t = _t()
    pw.io.kinesis.write(t, "stream1", endpoint=http.url, format="json")
    pw.run()
    [req] = [r for r in http.requests
             if r.headers.get("X-Amz-Target", "").endswith("PutRecords")]
    body = req.json()
    assert body["StreamName"] == "stream1"
    recs = [json.loads(base64.b64decode(r["Data"]))
            for r in body["Records"]]
    assert sorted((r["a"], r["b"]) for r in recs) == [(1, "x"), (2, "y")]

    # read: fake must reply shards + records
    payload = base64.b64encode(json.dumps({"k": 3}).encode()).decode()
    http.replies = {}
    state = {"n": 0}

    def reply_wrap():
        # call order: DescribeStream, GetShardIterator, GetRecords...
        state["n"] += 1
        return {"StreamDescription": {"Shards": [{"ShardId": "s0"}]},
                "ShardIterator": "it0",
                "Records": ([{"Data": payload}] if state["n"] == 3 else []),
                "NextShardIterator": "it1"}

    http.replies[""] = (200, reply_wrap)
    from pathway_amd.internals.schema import schema_from_types

    G.clear()
    tbl = pw.io.kinesis.read(
        "stream1", schema=schema_from_types(k=int), format="json",
        endpoint=http.url, _max_polls=2,
    )
    keys, cols = pw.debug.table_to_dicts(tbl)
    assert list(cols["k"].values()) == [3]


def test_bigquery_insert_all(http):
    t = _t()
    pw.io.bigquery.write(
        t, "ds", "tbl", project_id="proj", base_url=http.url,
        credentials="tok123",
    )
    pw.run()
    [req] = http.requests
    assert req.path == "/projects/proj/datasets/ds/tables/tbl/insertAll"
    assert req.headers.get("Authorization") == "Bearer tok123"
    rows = [r["json"] for r in req.json()["rows"]]
    assert sorted((r["a"], r["b"]) for r in rows) == [(1, "x"), (2, "y")]


def test_pubsub_publish_and_pull(http):
    t = _t()
    pw.io.pubsub.write(t, "proj", "topic1", base_url=http.url, format="json")
    pw.run()
    [req] = http.requests
    assert req.path == "/projects/proj/topics/topic1:publish"
    msgs = [json.loads(base64.b64decode(m["data"]))
            for m in req.json()["messages"]]
    assert sorted((m["a"], m["b"]) for m in msgs) == [(1, "x"), (2, "y")]

    state = {"n": 0}

    def reply():
        state["n"] += 1
        if state["n"] == 1:
            return {"receivedMessages": [
                {"ackId": "a1", "message": {
                    "data": base64.b64encode(
                        json.dumps({"k": 9}).encode()).decode()}}
            ]}
        return {}

    http.replies[""] = (200, reply)
    from pathway_amd.internals.schema import schema_from_types

    G.clear()
    tbl = pw.io.pubsub.read(
        "proj", "sub1", schema=schema_from_types(k=int), format="json",
        base_url=http.url, _max_polls=2,
    )
    keys, cols = pw.debug.table_to_dicts(tbl)
    assert list(cols["k"].values()) == [9]
    acks = [r for r in http.requests if r.path.endswith(":acknowledge")]
    assert acks and acks[0].json()["ackIds"] == ["a1"]


def test_slack_alerts(http):
    G.clear()
    t = pw.debug.table_from_markdown(
        """
        msg
        alert_one
        """
    )
    pw.io.slack.send_alerts(t, "C123", "xoxb-tok", base_url=http.url)
    pw.run()
    [req] = http.requests
    assert req.path == "/chat.postMessage"
    assert req.headers.get("Authorization") == "Bearer xoxb-tok"
    assert req.json() == {"channel": "C123", "text": "alert_one"}


def _vec_table():
    G.clear()
    import pathway_amd as pw_

    t = pw_.debug.table_from_markdown(
        """
        doc
        alpha
        """
    )
    return t.select(
        doc=pw.this.doc,
        vector=pw.apply(lambda d: (1.0, 2.0, 3.0), pw.this.doc),
    )


def test_qdrant_sink(http):
    t = _vec_table()
    pw.io.qdrant.write(t, http.url, "coll1", api_key="k")
    pw.run()
    [req] = http.requests
    assert req.path == "/collections/coll1/points"
    pt = req.json()["points"][0]
    assert pt["vector"] == [1.0, 2.0, 3.0]
    assert pt["payload"]["doc"] == "alpha"


def test_pinecone_sink(http):
    t = _vec_table()
    pw.io.pinecone.write(t, http.url, "key")
    pw.run()
    [req] = http.requests
    assert req.path == "/vectors/upsert"
    v = req.json()["vectors"][0]
    assert v["values"] == [1.0, 2.0, 3.0]
    assert v["metadata"]["doc"] == "alpha"


def test_chroma_milvus_weaviate_leann_sinks(http):
    for mod, args, path_frag in [
        (pw.io.chroma, (http.url, "cid"), "/api/v1/collections/cid/upsert"),
        (pw.io.milvus, (http.url, "coll"), "/v2/vectordb/entities/upsert"),
        (pw.io.weaviate, (http.url, "Cls"), "/v1/batch/objects"),
        (pw.io.leann, (http.url, "idx"), "/indexes/idx/documents"),
    ]:
        http.requests.clear()
        t = _vec_table()
        mod.write(t, *args)
        pw.run()
        assert any(r.path == path_frag for r in http.requests), (
            mod.__name__, [r.path for r in http.requests])


def test_gdrive_read(http):
    files = {"files": [
        {"id": "f1", "name": "a.txt", "md5Checksum": "m1",
         "modifiedTime": "t1", "mimeType": "text/plain"},
    ]}
    http.replies["/files?"] = (200, files)
    http.replies["/files/f1"] = (200, {"content": "hello"})
    G.clear()
    t = pw.io.gdrive.read(
        "folder1", mode="static", format="binary",
        credentials="tok", base_url=http.url,
    )
    keys, cols = pw.debug.table_to_dicts(t)
    [data] = list(cols["data"].values())
    assert json.loads(data) == {"content": "hello"}
    listed = [r for r in http.requests if r.path.startswith("/files?")]
    assert listed and "Bearer tok" == listed[0].headers.get("Authorization")
