"""pw.io.pyfilesystem — read from a PyFilesystem-style FS object.

Reference: python/pathway/io/pyfilesystem (fs library).  Accepts either
a PyFilesystem2-compatible object (duck-typed: listdir/readbytes/
getinfo) or an ``osfs://`` / plain directory path, which is served by
the built-in local implementation — the fs client library is not
required.
"""

from __future__ import annotations

import os
import time as _time
from typing import Any

from pathway_amd.io._object_store import ObjectStoreReader


class _LocalFS:
    """Minimal PyFilesystem-compatible view over a local directory."""

    def __init__(self, root: str):
        self.root = root

    def listdir(self, path: str = "/") -> list[str]:
        d = os.path.join(self.root, path.lstrip("/"))
        return sorted(os.listdir(d)) if os.path.isdir(d) else []

    def readbytes(self, path: str) -> bytes:
        with open(os.path.join(self.root, path.lstrip("/")), "rb") as f:
            return f.read()

    def getinfo(self, path: str, namespaces=None):
        st = os.stat(os.path.join(self.root, path.lstrip("/")))

        class Info:
            size = st.st_size
            modified = st.st_mtime

        return Info()


class _FSStore:
    """ObjectStore protocol over a PyFilesystem-like object."""

    def __init__(self, fs: Any, path: str = "/"):
        self.fs = fs
        self.path = path

    def list(self, prefix: str):
        out = []
        for name in self.fs.listdir(self.path):
            if prefix and not name.startswith(prefix):
                continue
            try:
                info = self.fs.getinfo(f"{self.path.rstrip('/')}/{name}")
                etag = f"{getattr(info, 'size', 0)}-{getattr(info, 'modified', 0)}"
            except Exception:
                etag = ""
            out.append((name, etag))
        return out

    def get(self, key: str):
        try:
            return self.fs.readbytes(f"{self.path.rstrip('/')}/{key}")
        except Exception:
            return None

    def put(self, key: str, data: bytes):
        raise NotImplementedError

    def delete(self, key: str):
        raise NotImplementedError


def read(
    source: Any,
    path: str = "/",
    *,
    format: str = "binary",
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

    if isinstance(source, str):
        root = source[len("osfs://"):] if source.startswith("osfs://") else source
        source = _LocalFS(root)
    if schema is None:
        schema = schema_from_types(
            data=bytes if format == "binary" else str
        )
    names = schema.column_names()
    dtypes = [schema.__columns__[n].dtype for n in names]
    if with_metadata:
        names = names + ["_metadata"]
        dtypes = dtypes + [dt.JSON]
    store = _FSStore(source, path)
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
