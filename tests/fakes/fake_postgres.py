"""In-process fake PostgreSQL server (protocol v3 subset).

Supports: trust-auth startup, simple queries over an in-memory table
store (CREATE TABLE / INSERT / DELETE / SELECT * [WHERE col=val]), and
logical replication (START_REPLICATION → CopyBoth streaming of
wal2json-style XLogData payloads fed via FakePostgres.emit_change).
"""

from __future__ import annotations

import json
import re
import socketserver
import struct
import threading
import time


class FakePostgres:
    def __init__(self):
        #: table -> (columns, rows)
        self.tables: dict[str, tuple[list[str], list[list]]] = {}
        self.lock = threading.Lock()
        #: replication log: list of (lsn, payload bytes)
        self.wal: list[tuple[int, bytes]] = []
        self.next_lsn = 1000
        store = self

        class Handler(socketserver.BaseRequestHandler):
            def handle(self):
                sock = self.request
                try:
                    # startup
                    head = self._recv(sock, 4)
                    if head is None:
                        return
                    (length,) = struct.unpack(">i", head)
                    body = self._recv(sock, length - 4)
                    (proto,) = struct.unpack_from(">i", body, 0)
                    if proto == 80877103:  # SSLRequest
                        sock.sendall(b"N")
                        head = self._recv(sock, 4)
                        (length,) = struct.unpack(">i", head)
                        body = self._recv(sock, length - 4)
                    params = {}
                    kv = body[4:].split(b"\x00")
                    for i in range(0, len(kv) - 1, 2):
                        if kv[i]:
                            params[kv[i].decode()] = kv[i + 1].decode()
                    self.replication = params.get("replication") == "database"
                    self._send(sock, b"R", struct.pack(">i", 0))  # AuthOk
                    self._send(sock, b"S", b"server_version\x0015.0\x00")
                    self._send(sock, b"K", struct.pack(">ii", 1, 2))
                    self._send(sock, b"Z", b"I")
                    while True:
                        mh = self._recv(sock, 5)
                        if mh is None:
                            return
                        mtype = mh[:1]
                        (mlen,) = struct.unpack(">i", mh[1:])
                        data = self._recv(sock, mlen - 4) if mlen > 4 else b""
                        if mtype == b"X":
                            return
                        if mtype == b"Q":
                            sql = data.rstrip(b"\x00").decode()
                            if sql.upper().startswith("START_REPLICATION"):
                                self._replicate(sock, sql)
                                return
                            self._query(sock, sql)
                except (ConnectionResetError, BrokenPipeError, OSError):
                    return

            # -- plumbing --

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

            @staticmethod
            def _send(sock, mtype: bytes, body: bytes):
                sock.sendall(mtype + struct.pack(">i", 4 + len(body)) + body)

            def _error(self, sock, msg: str):
                body = b"SERROR\x00C42601\x00M" + msg.encode() + b"\x00\x00"
                self._send(sock, b"E", body)
                self._send(sock, b"Z", b"I")

            def _complete(self, sock, tag: str):
                self._send(sock, b"C", tag.encode() + b"\x00")
                self._send(sock, b"Z", b"I")

            # -- SQL (minimal) --

            def _query(self, sock, sql: str):
                s = sql.strip().rstrip(";")
                up = s.upper()
                try:
                    if up.startswith("CREATE TABLE"):
                        m = re.match(r"CREATE TABLE (?:IF NOT EXISTS )?(\S+)\s*\((.*)\)",
                                     s, re.I | re.S)
                        name = m.group(1).strip('"')
                        cols = [c.strip().split()[0].strip('"')
                                for c in m.group(2).split(",")]
                        with store.lock:
                            store.tables.setdefault(name, (cols, []))
                        return self._complete(sock, "CREATE TABLE")
                    if up.startswith("INSERT INTO"):
                        m = re.match(
                            r"INSERT INTO (\S+)\s*\(([^)]*)\)\s*VALUES\s*(.*)",
                            s, re.I | re.S)
                        name = m.group(1).strip('"')
                        cols = [c.strip().strip('"') for c in m.group(2).split(",")]
                        tuples = re.findall(r"\(([^)]*)\)", m.group(3))
                        with store.lock:
                            tcols, rows = store.tables.setdefault(name, (cols, []))
                            for tup in tuples:
                                vals = [store._parse_value(v) for v in
                                        store._split_values(tup)]
                                rec = dict(zip(cols, vals))
                                rows.append([rec.get(c) for c in tcols])
                                store.record_change(name, "insert", tcols,
                                                    [rec.get(c) for c in tcols])
                        return self._complete(sock, f"INSERT 0 {len(tuples)}")
                    if up.startswith("DELETE FROM"):
                        m = re.match(r"DELETE FROM (\S+)(?:\s+WHERE\s+(.*))?", s, re.I | re.S)
                        name = m.group(1).strip('"')
                        cond = m.group(2)
                        with store.lock:
                            tcols, rows = store.tables.get(name, ([], []))
                            keep, removed = [], []
                            for r in rows:
                                if cond is None or store._match(tcols, r, cond):
                                    removed.append(r)
                                else:
                                    keep.append(r)
                            rows[:] = keep
                            for r in removed:
                                store.record_change(name, "delete", tcols, r)
                        return self._complete(sock, f"DELETE {len(removed)}")
                    if up.startswith("SELECT"):
                        m = re.match(r"SELECT \* FROM (\S+)(?:\s+WHERE\s+(.*))?", s, re.I | re.S)
                        if not m:
                            return self._error(sock, f"unsupported SELECT: {sql}")
                        name = m.group(1).strip('"')
                        cond = m.group(2)
                        with store.lock:
                            tcols, rows = store.tables.get(name, ([], []))
                            out = [r for r in rows
                                   if cond is None or store._match(tcols, r, cond)]
                        # RowDescription
                        desc = struct.pack(">h", len(tcols))
                        for c in tcols:
                            desc += c.encode() + b"\x00"
                            desc += struct.pack(">ihihih", 0, 0, 25, -1, -1, 0)
                        self._send(sock, b"T", desc)
                        for r in out:
                            dr = struct.pack(">h", len(r))
                            for v in r:
                                if v is None:
                                    dr += struct.pack(">i", -1)
                                else:
                                    b = str(v).encode()
                                    dr += struct.pack(">i", len(b)) + b
                            self._send(sock, b"D", dr)
                        return self._complete(sock, f"SELECT {len(out)}")
                    if up.startswith("CREATE_REPLICATION_SLOT"):
                        return self._complete(sock, "CREATE_REPLICATION_SLOT")
                    return self._error(sock, f"unsupported statement: {sql}")
                except Exception as e:  # pragma: no cover
                    return self._error(sock, f"{type(e).__name__}: {e}")

            # -- replication --

            def _replicate(self, sock, sql: str):
                self._send(sock, b"W", struct.pack(">bh", 0, 0))
                sent = 0
                try:
                    while True:
                        with store.lock:
                            wal = list(store.wal)
                        while sent < len(wal):
                            lsn, payload = wal[sent]
                            msg = b"w" + struct.pack(
                                ">qqq", lsn, lsn, int(time.time() * 1e6)
                            ) + payload
                            self._send(sock, b"d", msg)
                            sent += 1
                        time.sleep(0.05)
                except (BrokenPipeError, ConnectionResetError, OSError):
                    return

        class Server(socketserver.ThreadingTCPServer):
            allow_reuse_address = True
            daemon_threads = True

        self.server = Server(("127.0.0.1", 0), Handler)
        self.thread = threading.Thread(target=self.server.serve_forever, daemon=True)

    # -- helpers --

    @staticmethod
    def _split_values(tup: str) -> list[str]:
        out, cur, ins = [], "", False
        i = 0
        while i < len(tup):
            ch = tup[i]
            if ins:
                if ch == "'" and i + 1 < len(tup) and tup[i + 1] == "'":
                    cur += "'"
                    i += 1
                elif ch == "'":
                    ins = False
                else:
                    cur += ch
            else:
                if ch == "'":
                    ins = True
                    cur += "\x01"  # mark as string
                elif ch == ",":
                    out.append(cur.strip())
                    cur = ""
                else:
                    cur += ch
            i += 1
        out.append(cur.strip())
        return out

    @staticmethod
    def _parse_value(v: str):
        if v.startswith("\x01"):
            return v[1:]
        if v.upper() == "NULL":
            return None
        if v.upper() in ("TRUE", "FALSE"):
            return v.upper() == "TRUE"
        try:
            return int(v)
        except ValueError:
            try:
                return float(v)
            except ValueError:
                return v

    @staticmethod
    def _match(cols: list[str], row: list, cond: str) -> bool:
        ok = True
        for clause in cond.split(" AND "):
            m = re.match(r"\s*\"?(\w+)\"?\s*=\s*(.*)", clause.strip())
            if not m:
                return False
            col, val = m.group(1), m.group(2).strip()
            val = FakePostgres._parse_value(
                "\x01" + val[1:-1].replace("''", "'") if val.startswith("'") else val
            )
            try:
                got = row[cols.index(col)]
            except ValueError:
                return False
            if str(got) != str(val):
                ok = False
        return ok

    def record_change(self, table: str, kind: str, cols: list[str], row: list) -> None:
        """Append a wal2json-style change record to the WAL stream."""
        payload = json.dumps({
            "change": [{
                "kind": kind,
                "table": table,
                "columnnames": cols,
                "columnvalues": row if kind == "insert" else None,
                "oldkeys": (
                    {"keynames": cols, "keyvalues": row}
                    if kind == "delete" else None
                ),
            }]
        }).encode()
        self.wal.append((self.next_lsn, payload))
        self.next_lsn += 8

    # -- lifecycle --

    @property
    def port(self) -> int:
        return self.server.server_address[1]

    def start(self) -> "FakePostgres":
        self.thread.start()
        return self

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()
