"""S3/MinIO connector: SigV4 client + object-store reader/writer against
the in-process fake S3 endpoint (real HTTP + ListObjectsV2 XML paths)."""

import json
import threading
import time

import pytest

import pathway_amd as pw
from pathway_amd.internals.rungraph import G
from pathway_amd.internals.schema import schema_from_types
from pathway_amd.io._s3_client import S3Client
from pathway_amd.io.s3 import AwsS3Settings
from tests.fakes.fake_s3 import FakeS3


@pytest.fixture()
def s3srv():
    s = FakeS3().start()
    yield s
    s.stop()


def _settings(s3srv, bucket="b1"):
    return AwsS3Settings(
        bucket_name=bucket, access_key="ak", secret_access_key="sk",
        endpoint=s3srv.endpoint,
    )


def test_client_put_get_list_delete(s3srv):
    c = S3Client(s3srv.endpoint, access_key="ak", secret_key="sk")
    c.put_object("b", "x/a.txt", b"alpha")
    c.put_object("b", "x/b.txt", b"beta")
    c.put_object("b", "y/c.txt", b"gamma")
    assert c.get_object("b", "x/a.txt") == b"alpha"
    assert c.get_object("b", "missing") is None
    keys = [o.key for o in c.list_objects("b", "x/")]
    assert keys == ["x/a.txt", "x/b.txt"]
    c.copy_object("b", "x/a.txt", "x/a2.txt")
    assert c.get_object("b", "x/a2.txt") == b"alpha"
    c.delete_object("b", "x/a.txt")
    assert c.get_object("b", "x/a.txt") is None
    assert c.head_object("b", "x/b.txt") is not None


def test_s3_read_static_json(s3srv):
    c = S3Client(s3srv.endpoint)
    rows = [{"k": i, "v": f"s{i}"} for i in range(4)]
    c.put_object("b1", "data/part0.jsonl",
                 "\n".join(json.dumps(r) for r in rows[:2]).encode())
    c.put_object("b1", "data/part1.jsonl",
                 "\n".join(json.dumps(r) for r in rows[2:]).encode())

    G.clear()
    t = pw.io.s3.read(
        "data/", aws_s3_settings=_settings(s3srv), format="json",
        schema=schema_from_types(k=int, v=str), mode="static",
    )
    keys, cols = pw.debug.table_to_dicts(t)
    assert sorted((cols["k"][k], cols["v"][k]) for k in keys) == [
        (i, f"s{i}") for i in range(4)
    ]


def test_s3_read_streaming_new_and_deleted_objects(s3srv):
    c = S3Client(s3srv.endpoint)
    c.put_object("b1", "in/a.txt", b"one\ntwo\n")

    def later():
        time.sleep(0.3)
        c.put_object("b1", "in/b.txt", b"three\n")
        time.sleep(0.3)
        c.delete_object("b1", "in/a.txt")

    th = threading.Thread(target=later)
    th.start()
    G.clear()
    t = pw.io.s3.read(
        "in/", aws_s3_settings=_settings(s3srv), format="plaintext",
        mode="streaming", refresh_interval=0.1, _max_polls=12,
    )
    res = t.groupby().reduce(c=pw.reducers.count())
    keys, cols = pw.debug.table_to_dicts(res)
    th.join()
    # a.txt rows retracted after deletion; only b.txt's row remains
    assert list(cols["c"].values()) == [1]


def test_s3_write_and_read_back(s3srv):
    G.clear()
    t = pw.debug.table_from_markdown(
        """
        a | b
        1 | x
        2 | y
        """
    )
    pw.io.s3.write(t, "out/", aws_s3_settings=_settings(s3srv), format="json")
    pw.run()
    c = S3Client(s3srv.endpoint)
    objs = c.list_objects("b1", "out/")
    assert objs
    recs = []
    for o in objs:
        for line in c.get_object("b1", o.key).decode().splitlines():
            recs.append(json.loads(line))
    assert sorted((r["a"], r["b"]) for r in recs) == [(1, "x"), (2, "y")]


def test_minio_wrapper(s3srv):
    from pathway_amd.io.minio import MinIOSettings

    c = S3Client(s3srv.endpoint)
    c.put_object("mb", "p/x.txt", b"hello\n")
    G.clear()
    t = pw.io.minio.read(
        "p/",
        MinIOSettings(
            endpoint=s3srv.endpoint, bucket_name="mb",
            access_key="a", secret_access_key="s",
        ),
        format="plaintext",
        mode="static",
    )
    keys, cols = pw.debug.table_to_dicts(t)
    assert list(cols["data"].values()) == ["hello"]


def test_sigv4_signature_shape(s3srv):
    # the Authorization header is built per AWS SigV4; verify its shape
    import urllib.parse

    c = S3Client(s3srv.endpoint, access_key="AKID", secret_key="SECRET",
                 region="eu-west-1")
    headers = c._sign("GET", "/b/k", {}, {}, b"")
    auth = headers["Authorization"]
    assert auth.startswith("AWS4-HMAC-SHA256 Credential=AKID/")
    assert "/eu-west-1/s3/aws4_request" in auth
    assert "SignedHeaders=" in auth and "Signature=" in auth
