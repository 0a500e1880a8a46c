"""pw.io.gdrive — Google Drive reader over the Drive v3 REST API.

Reference: python/pathway/io/gdrive (google-api-python-client).  Uses
files.list with a parent query + files.get?alt=media downloads; objects
are tracked by md5Checksum/modifiedTime so edited files retract their
old rows and deleted files are retracted (same semantics as the fs and
object-store connectors).
"""

from __future__ import annotations

import time as _time
import urllib.parse
from typing import Any

from pathway_amd.io import _rest

DEFAULT_BASE = "https://www.googleapis.com/drive/v3"


class GDriveClient:
    def __init__(self, credentials: Any = None, base_url: str = DEFAULT_BASE):
        self.base = base_url.rstrip("/")
        token = getattr(credentials, "token", None) or (
            credentials if isinstance(credentials, str) else None
        )
        self.headers = {"Authorization": f"Bearer {token}"} if token else {}

    def list_files(self, folder_id: str) -> list[dict]:
        q = urllib.parse.quote(f"'{folder_id}' in parents and trashed = false")
        fields = urllib.parse.quote(
            "files(id, name, md5Checksum, modifiedTime, mimeType, size)"
        )
        out = _rest.request(
            "GET", f"{self.base}/files?q={q}&fields={fields}",
            headers=self.headers,
        ) or {}
        return out.get("files", [])

    def download(self, file_id: str) -> bytes:
        out = _rest.request(
            "GET", f"{self.base}/files/{file_id}?alt=media",
            headers=self.headers,
        )
        if isinstance(out, bytes):
            return out
        import json as _json

        return _json.dumps(out).encode()


class _DriveStore:
    """ObjectStore protocol over a Drive folder (read-only)."""

    def __init__(self, client: GDriveClient, folder_id: str):
        self.client = client
        self.folder_id = folder_id
        self._ids: dict[str, str] = {}

    def list(self, prefix: str):
        files = self.client.list_files(self.folder_id)
        out = []
        for f in files:
            self._ids[f["name"]] = f["id"]
            etag = f.get("md5Checksum") or f.get("modifiedTime") or f["id"]
            out.append((f["name"], etag))
        return out

    def get(self, key: str):
        fid = self._ids.get(key)
        return self.client.download(fid) if fid else None

    def put(self, key: str, data: bytes):
        raise NotImplementedError("gdrive connector is read-only")

    def delete(self, key: str):
        raise NotImplementedError("gdrive connector is read-only")


def read(
    object_id: str,
    *,
    mode: str = "streaming",
    format: str = "binary",
    schema=None,
    object_size_limit: int | None = None,
    service_user_credentials_file: str | None = None,
    credentials: Any = None,
    with_metadata: bool = False,
    refresh_interval: float = 30.0,
    base_url: str = DEFAULT_BASE,
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
    from pathway_amd.io._object_store import ObjectStoreReader

    if schema is None:
        schema = schema_from_types(
            data=bytes if format == "binary" else str
        )
    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]
    if with_metadata:
        names = names + ["_metadata"]
        dtypes = dtypes + [dt.JSON]

    client = GDriveClient(credentials, base_url)
    store = _DriveStore(client, object_id)
    src = StreamingSource(names, dtypes, name=name)
    reader = ObjectStoreReader(
        src, store, "", format, schema,
        mode=mode, refresh_interval=refresh_interval, max_polls=_max_polls,
        with_metadata=with_metadata,
    )
    src.reader = reader
    spawn_reader(reader.run, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())
