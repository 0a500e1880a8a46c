"""pw.io.azure — Azure Blob Storage connector over the Blob REST API.

Reference: python/pathway/io/azure + src/connectors/data_storage (Azure
scanner) / persistence azure backend.  Speaks the Blob service REST
dialect directly (List Blobs XML, Put/Get/Delete Blob); auth via SAS
token (query-string) or anonymous — the in-process fake
(tests/fakes/fake_azure.py) exercises the same request/XML paths.
"""

from __future__ import annotations

import urllib.error
import urllib.parse
import urllib.request
import xml.etree.ElementTree as ET
from typing import Any

from pathway_amd.io._object_store import ObjectStoreReader, ObjectStoreWriter


class AzureBlobClient:
    def __init__(self, account_url: str, container: str, *,
                 sas_token: str | None = None, timeout: float = 30.0):
        self.base = account_url.rstrip("/")
        self.container = container
        self.sas = (sas_token or "").lstrip("?")
        self.timeout = timeout

    def _url(self, blob: str = "", params: dict[str, str] | None = None) -> str:
        url = f"{self.base}/{self.container}"
        if blob:
            url += f"/{urllib.parse.quote(blob)}"
        qs = urllib.parse.urlencode(params or {})
        parts = [p for p in (qs, self.sas) if p]
        return url + ("?" + "&".join(parts) if parts else "")

    def _request(self, method: str, url: str, body: bytes | None = None,
                 headers: dict | None = None):
        req = urllib.request.Request(url, data=body, method=method,
                                     headers=headers or {})
        try:
            with urllib.request.urlopen(req, timeout=self.timeout) as resp:
                return resp.status, resp.read()
        except urllib.error.HTTPError as e:
            return e.code, e.read()

    def list_blobs(self, prefix: str = "") -> list[tuple[str, str]]:
        status, body = self._request(
            "GET", self._url(params={"restype": "container", "comp": "list",
                                     "prefix": prefix})
        )
        if status >= 300:
            return []
        root = ET.fromstring(body)
        out = []
        for b in root.iter("Blob"):
            name = b.findtext("Name")
            etag = b.findtext("Properties/Etag") or ""
            out.append((name, etag))
        return out

    def get_blob(self, name: str) -> bytes | None:
        status, body = self._request("GET", self._url(name))
        return None if status >= 300 else body

    def put_blob(self, name: str, data: bytes) -> None:
        status, body = self._request(
            "PUT", self._url(name), body=data,
            headers={"x-ms-blob-type": "BlockBlob"},
        )
        if status >= 300:
            raise RuntimeError(f"azure put failed {status}: {body[:200]!r}")

    def delete_blob(self, name: str) -> None:
        self._request("DELETE", self._url(name))

    # ObjectStore protocol
    def list(self, prefix: str):
        return self.list_blobs(prefix)

    def get(self, key: str):
        return self.get_blob(key)

    def put(self, key: str, data: bytes):
        self.put_blob(key, data)

    def delete(self, key: str):
        self.delete_blob(key)


def read(
    path: str,
    *,
    account_url: str,
    container: str,
    sas_token: str | None = None,
    format: str = "plaintext",
    schema=None,
    mode: str = "streaming",
    with_metadata: bool = False,
    refresh_interval: float = 0.5,
    name: str | None = None,
    _max_polls: int | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import StreamingSource, spawn_reader
    from pathway_amd.internals import dtype as dt
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.schema import schema_from_types
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

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
    client = AzureBlobClient(account_url, container, sas_token=sas_token)
    src = StreamingSource(names, dtypes, name=name)
    reader = ObjectStoreReader(
        src, client, path.lstrip("/"), format, schema,
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
    account_url: str,
    container: str,
    sas_token: str | None = None,
    format: str = "json",
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    client = AzureBlobClient(account_url, container, sas_token=sas_token)
    writer = ObjectStoreWriter(client, path.lstrip("/"), format)
    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
