"""pw.io.s3 — S3/MinIO-compatible object-store connector.

Reference: python/pathway/io/s3 + src/connectors/data_storage S3 scanner.
Implemented over the pure-python SigV4 REST client (io/_s3_client.py);
tested end-to-end against the in-process fake S3 endpoint
(tests/fakes/fake_s3.py) — same HTTP/XML paths as a real endpoint.
"""

from __future__ import annotations

from typing import Any

from pathway_amd.io._object_store import ObjectStoreReader, ObjectStoreWriter
from pathway_amd.io._s3_client import S3Client


class AwsS3Settings:
    """Credentials + addressing for S3-compatible services
    (reference io/s3 AwsS3Settings)."""

    def __init__(
        self,
        bucket_name: str | None = None,
        *,
        access_key: str | None = None,
        secret_access_key: str | None = None,
        with_iam: bool = False,
        region: str | None = None,
        endpoint: str | None = None,
        **kw: Any,
    ):
        self.bucket_name = bucket_name
        self.access_key = access_key
        self.secret_access_key = secret_access_key
        self.with_iam = with_iam
        self.region = region or "us-east-1"
        self.endpoint = endpoint

    def create_client(self) -> S3Client:
        endpoint = self.endpoint or f"https://s3.{self.region}.amazonaws.com"
        return S3Client(
            endpoint,
            access_key=self.access_key or "",
            secret_key=self.secret_access_key or "",
            region=self.region,
        )


class DigitalOceanS3Settings(AwsS3Settings):
    """DigitalOcean Spaces credentials (reference io/s3)."""


class WasabiS3Settings(AwsS3Settings):
    """Wasabi credentials (reference io/s3)."""


class _BucketStore:
    """ObjectStore protocol over (S3Client, bucket)."""

    def __init__(self, client: S3Client, bucket: str):
        self.client = client
        self.bucket = bucket

    def list(self, prefix: str):
        return [(o.key, o.etag) for o in self.client.list_objects(self.bucket, prefix)]

    def get(self, key: str):
        return self.client.get_object(self.bucket, key)

    def put(self, key: str, data: bytes):
        self.client.put_object(self.bucket, key, data)

    def delete(self, key: str):
        self.client.delete_object(self.bucket, key)


def _split_path(path: str, settings: AwsS3Settings | None) -> tuple[str, str]:
    """'s3://bucket/prefix' or plain 'prefix' -> (bucket, prefix)."""
    if path.startswith("s3://"):
        rest = path[5:]
        bucket, _, prefix = rest.partition("/")
        return bucket, prefix
    bucket = settings.bucket_name if settings else None
    if not bucket:
        raise ValueError("no bucket: pass s3://bucket/... or AwsS3Settings(bucket_name=...)")
    return bucket, path.lstrip("/")


def read(
    path: str,
    *,
    aws_s3_settings: AwsS3Settings | None = None,
    format: str = "plaintext",
    schema=None,
    mode: str = "streaming",
    with_metadata: bool = False,
    autocommit_duration_ms: int | None = 1500,
    name: str | None = None,
    refresh_interval: float = 0.5,
    _max_polls: int | None = None,
    downloader_threads_count: int | None = None,
    persistent_id: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import StreamingSource, spawn_reader
    from pathway_amd.internals import dtype as dt
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.schema import schema_from_types
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    settings = aws_s3_settings or AwsS3Settings()
    bucket, prefix = _split_path(path, settings)
    if schema is None:
        if format == "plaintext":
            schema = schema_from_types(data=str)
        elif format == "binary":
            schema = schema_from_types(data=bytes)
        else:
            raise ValueError(f"format {format!r} requires a schema")
    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]
    if with_metadata:
        names = names + ["_metadata"]
        dtypes = dtypes + [dt.JSON]

    src = StreamingSource(names, dtypes, name=name)
    store = _BucketStore(settings.create_client(), bucket)
    reader = ObjectStoreReader(
        src, store, prefix, format, schema,
        mode=mode, refresh_interval=refresh_interval, max_polls=_max_polls,
        with_metadata=with_metadata,
    )
    src.reader = reader
    spawn_reader(reader.run, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())


def write(
    table,
    path: str,
    *,
    aws_s3_settings: AwsS3Settings | None = None,
    format: str = "json",
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    settings = aws_s3_settings or AwsS3Settings()
    bucket, prefix = _split_path(path, settings)
    store = _BucketStore(settings.create_client(), bucket)
    writer = ObjectStoreWriter(store, prefix, format)
    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node


def read_from_digital_ocean(path, do_s3_settings, format, **kwargs):
    return read(path, aws_s3_settings=do_s3_settings, format=format, **kwargs)


def read_from_wasabi(path, wasabi_s3_settings, format, **kwargs):
    return read(path, aws_s3_settings=wasabi_s3_settings, format=format, **kwargs)
