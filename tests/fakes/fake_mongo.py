"""In-process fake MongoDB server (OP_MSG subset, in-memory collections)."""

from __future__ import annotations

import socketserver
import struct
import threading

from pathway_amd.io.formats import bson

OP_MSG = 2013


class FakeMongo:
    def __init__(self):
        #: (db, coll) -> list[dict]
        self.collections: dict[tuple[str, str], list[dict]] = {}
        self.lock = threading.Lock()
        store = self

        class Handler(socketserver.BaseRequestHandler):
            def handle(self):
                sock = self.request
                try:
                    while True:
                        head = self._recv(sock, 4)
                        if head is None:
                            return
                        (length,) = struct.unpack("<i", head)
                        rest = self._recv(sock, length - 4)
                        if rest is None:
                            return
                        rid, _resp_to, opcode = struct.unpack_from("<iii", rest, 0)
                        if opcode != OP_MSG:
                            return
                        cmd = bson.decode(rest[12 + 5 :])
                        reply = store.handle_command(cmd)
                        body = struct.pack("<I", 0) + b"\x00" + bson.encode(reply)
                        header = struct.pack("<iiii", 16 + len(body), 1, rid, OP_MSG)
                        sock.sendall(header + body)
                except (ConnectionResetError, BrokenPipeError, OSError):
                    return

            @staticmethod
            def _recv(sock, n):
                buf = b""
                while len(buf) < n:
                    try:
                        chunk = sock.recv(n - len(buf))
                    except OSError:
                        return None
                    if not chunk:
                        return None
                    buf += chunk
                return buf

        class Server(socketserver.ThreadingTCPServer):
            allow_reuse_address = True
            daemon_threads = True

        self.server = Server(("127.0.0.1", 0), Handler)
        self.thread = threading.Thread(target=self.server.serve_forever, daemon=True)

    @property
    def uri(self) -> str:
        return f"mongodb://127.0.0.1:{self.server.server_address[1]}"

    def start(self) -> "FakeMongo":
        self.thread.start()
        return self

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()

    # -- command dispatch --

    def handle_command(self, cmd: dict) -> dict:
        db = cmd.get("$db", "test")
        if "hello" in cmd or "ismaster" in cmd:
            return {"ok": 1, "isWritablePrimary": True, "maxWireVersion": 17}
        if "ping" in cmd:
            return {"ok": 1}
        if "insert" in cmd:
            coll = cmd["insert"]
            docs = cmd.get("documents", [])
            with self.lock:
                self.collections.setdefault((db, coll), []).extend(docs)
            return {"ok": 1, "n": len(docs)}
        if "find" in cmd:
            coll = cmd["find"]
            filt = cmd.get("filter") or {}
            with self.lock:
                docs = [d for d in self.collections.get((db, coll), [])
                        if self._matches(d, filt)]
            return {"ok": 1, "cursor": {"firstBatch": docs, "id": 0,
                                        "ns": f"{db}.{coll}"}}
        if "delete" in cmd:
            coll = cmd["delete"]
            n = 0
            with self.lock:
                docs = self.collections.get((db, coll), [])
                for spec in cmd.get("deletes", []):
                    filt = spec.get("q") or {}
                    keep = [d for d in docs if not self._matches(d, filt)]
                    n += len(docs) - len(keep)
                    docs[:] = keep
            return {"ok": 1, "n": n}
        if "drop" in cmd:
            with self.lock:
                self.collections.pop((db, cmd["drop"]), None)
            return {"ok": 1}
        return {"ok": 0, "errmsg": f"unsupported command {list(cmd)[0]}"}

    @staticmethod
    def _matches(doc: dict, filt: dict) -> bool:
        for k, v in filt.items():
            if isinstance(v, dict) and "$gt" in v:
                dv = doc.get(k)
                if dv is None or not dv > v["$gt"]:
                    return False
            elif doc.get(k) != v:
                return False
        return True
