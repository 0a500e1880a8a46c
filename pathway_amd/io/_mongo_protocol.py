"""Pure-python MongoDB wire-protocol client (OP_MSG over TCP).

Replaces the reference's mongodb crate (src/connectors/data_storage
mongodb writer) with a from-scratch OP_MSG implementation over the BSON
codec in io/formats/bson.py.  Commands used: hello, insert, find,
getMore, delete, drop, ping — enough for the connector's read/write
paths.  Exercised against the in-process fake server
(tests/fakes/fake_mongo.py) speaking the same protocol.
"""

from __future__ import annotations

import socket
import struct
import threading
import urllib.parse
from typing import Any

from pathway_amd.io.formats import bson

OP_MSG = 2013


class MongoError(RuntimeError):
    pass


class MongoClient:
    def __init__(self, connection_string: str = "mongodb://127.0.0.1:27017",
                 timeout: float = 30.0):
        u = urllib.parse.urlparse(connection_string)
        host = u.hostname or "127.0.0.1"
        port = u.port or 27017
        self.sock = socket.create_connection((host, port), timeout=timeout)
        self.request_id = 0
        self.lock = threading.Lock()
        self.hello = self.command("admin", {"hello": 1})

    def close(self) -> None:
        try:
            self.sock.close()
        except OSError:
            pass

    def _recv_exact(self, n: int) -> bytes:
        buf = b""
        while len(buf) < n:
            chunk = self.sock.recv(n - len(buf))
            if not chunk:
                raise MongoError("server closed connection")
            buf += chunk
        return buf

    def command(self, db: str, cmd: dict[str, Any]) -> dict[str, Any]:
        doc = dict(cmd)
        doc["$db"] = db
        body = struct.pack("<I", 0) + b"\x00" + bson.encode(doc)  # flags + kind0
        with self.lock:
            self.request_id += 1
            rid = self.request_id
            header = struct.pack("<iiii", 16 + len(body), rid, 0, OP_MSG)
            self.sock.sendall(header + body)
            (length,) = struct.unpack("<i", self._recv_exact(4))
            rest = self._recv_exact(length - 4)
        _rid, response_to, opcode = struct.unpack_from("<iii", rest, 0)
        if opcode != OP_MSG:
            raise MongoError(f"unexpected opcode {opcode}")
        # skip flagBits(4) + section kind byte(1)
        reply = bson.decode(rest[12 + 5 :])
        if reply.get("ok") != 1 and reply.get("ok") != 1.0:
            raise MongoError(f"command failed: {reply}")
        return reply

    # -- convenience operations --

    def insert_many(self, db: str, coll: str, docs: list[dict]) -> int:
        if not docs:
            return 0
        for d in docs:
            d.setdefault("_id", bson.ObjectId())
        reply = self.command(db, {"insert": coll, "documents": docs})
        return int(reply.get("n", 0))

    def find(self, db: str, coll: str, filter: dict | None = None,
             *, sort: dict | None = None, batch_size: int = 1000) -> list[dict]:
        cmd: dict[str, Any] = {"find": coll, "filter": filter or {},
                               "batchSize": batch_size}
        if sort:
            cmd["sort"] = sort
        reply = self.command(db, cmd)
        cursor = reply["cursor"]
        out = list(cursor["firstBatch"])
        cid = cursor["id"]
        while cid:
            reply = self.command(db, {"getMore": cid, "collection": coll,
                                      "batchSize": batch_size})
            cursor = reply["cursor"]
            out.extend(cursor["nextBatch"])
            cid = cursor["id"]
        return out

    def delete_many(self, db: str, coll: str, filter: dict) -> int:
        reply = self.command(
            db, {"delete": coll, "deletes": [{"q": filter, "limit": 0}]}
        )
        return int(reply.get("n", 0))

    def drop(self, db: str, coll: str) -> None:
        try:
            self.command(db, {"drop": coll})
        except MongoError:
            pass

    def ping(self) -> bool:
        return self.command("admin", {"ping": 1}).get("ok") in (1, 1.0)
