"""MongoDB connector: OP_MSG wire client + read/write against the
in-process fake server (real framing + BSON paths)."""

import threading
import time

import pytest

import pathway_amd as pw
from pathway_amd.internals.rungraph import G
from pathway_amd.internals.schema import schema_from_types
from pathway_amd.io._mongo_protocol import MongoClient
from tests.fakes.fake_mongo import FakeMongo


@pytest.fixture()
def mongo():
    m = FakeMongo().start()
    yield m
    m.stop()


def test_client_insert_find_delete(mongo):
    c = MongoClient(mongo.uri)
    assert c.ping()
    n = c.insert_many("db", "c1", [{"a": 1}, {"a": 2}, {"a": 2}])
    assert n == 3
    docs = c.find("db", "c1", {"a": 2})
    assert len(docs) == 2
    assert all(d["a"] == 2 for d in docs)
    assert all("_id" in d for d in docs)
    assert c.delete_many("db", "c1", {"a": 2}) == 2
    assert len(c.find("db", "c1")) == 1
    c.close()


def test_mongodb_read_static(mongo):
    c = MongoClient(mongo.uri)
    c.insert_many("db", "rows", [{"k": i, "v": f"s{i}"} for i in range(4)])
    G.clear()
    t = pw.io.mongodb.read(
        mongo.uri, database="db", collection="rows",
        schema=schema_from_types(k=int, v=str), mode="static",
    )
    keys, cols = pw.debug.table_to_dicts(t)
    assert sorted((cols["k"][k], cols["v"][k]) for k in keys) == [
        (i, f"s{i}") for i in range(4)
    ]
    c.close()


def test_mongodb_streaming_tail(mongo):
    c = MongoClient(mongo.uri)
    c.insert_many("db", "live", [{"k": 0}])

    def later():
        time.sleep(0.3)
        c2 = MongoClient(mongo.uri)
        c2.insert_many("db", "live", [{"k": 1}, {"k": 2}])
        c2.close()

    th = threading.Thread(target=later)
    th.start()
    G.clear()
    t = pw.io.mongodb.read(
        mongo.uri, database="db", collection="live",
        schema=schema_from_types(k=int), mode="streaming",
        refresh_interval=0.1, _max_polls=12,
    )
    res = t.groupby().reduce(s=pw.reducers.sum(pw.this.k), c=pw.reducers.count())
    keys, cols = pw.debug.table_to_dicts(res)
    th.join()
    assert list(cols["s"].values()) == [3]
    assert list(cols["c"].values()) == [3]
    c.close()


def test_mongodb_write_with_retractions(mongo):
    G.clear()
    t = pw.debug.table_from_markdown(
        """
        a | b | __time__ | __diff__
        1 | x | 2        | 1
        2 | y | 2        | 1
        2 | y | 4        | -1
        """
    )
    pw.io.mongodb.write(t, mongo.uri, database="db", collection="out")
    pw.run()
    c = MongoClient(mongo.uri)
    docs = c.find("db", "out")
    assert [(d["a"], d["b"]) for d in docs] == [(1, "x")]
    c.close()
