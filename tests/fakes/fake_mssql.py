"""In-process fake SQL Server (TDS subset: prelogin, login7, SQLBatch
with NVARCHAR resultsets)."""

from __future__ import annotations

import re
import socketserver
import struct
import threading

PKT_RESPONSE = 0x04
TOK_COLMETADATA = 0x81
TOK_LOGINACK = 0xAD
TOK_ROW = 0xD1
TOK_DONE = 0xFD
TOK_ERROR = 0xAA


class FakeMSSQL:
    def __init__(self):
        self.tables: dict[str, tuple[list[str], list[list]]] = {}
        self.lock = threading.Lock()
        store = self

        class Handler(socketserver.BaseRequestHandler):
            def handle(self):
                sock = self.request
                try:
                    # prelogin
                    if self._read_message(sock) is None:
                        return
                    self._respond(sock, b"\xff")  # empty prelogin response
                    # login7
                    if self._read_message(sock) is None:
                        return
                    ack = b"\x01" + struct.pack("<I", 0x74000004)
                    name = "fake".encode("utf-16-le")
                    ack += bytes([len("fake")]) + name + b"\x00\x00\x00\x00"
                    loginack = bytes([TOK_LOGINACK]) + struct.pack("<H", len(ack)) + ack
                    done = bytes([TOK_DONE]) + struct.pack("<HHQ", 0, 0, 0)
                    self._respond(sock, loginack + done)
                    while True:
                        msg = self._read_message(sock)
                        if msg is None:
                            return
                        sql = msg[22:].decode("utf-16-le", "ignore")
                        self._query(sock, sql)
                except (ConnectionResetError, BrokenPipeError, OSError):
                    return

            # -- framing --

            @staticmethod
            def _recv_exact(sock, n):
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

            def _read_message(self, sock):
                out = b""
                while True:
                    head = self._recv_exact(sock, 8)
                    if head is None:
                        return None
                    _t, status, length = struct.unpack(">BBH", head[:4])
                    body = self._recv_exact(sock, length - 8)
                    if body is None:
                        return None
                    out += body
                    if status & 0x01:
                        return out

            @staticmethod
            def _respond(sock, payload: bytes):
                head = struct.pack(">BBHHBB", PKT_RESPONSE, 0x01,
                                   8 + len(payload), 0, 0, 0)
                sock.sendall(head + payload)

            def _error(self, sock, msg: str):
                m = msg.encode("utf-16-le")
                body = struct.pack("<IBB", 50000, 1, 16)
                body += struct.pack("<H", len(msg)) + m
                body += b"\x00" + b"\x00\x00" + b"\x00\x00\x00\x00"
                tok = bytes([TOK_ERROR]) + struct.pack("<H", len(body)) + body
                done = bytes([TOK_DONE]) + struct.pack("<HHQ", 0x2, 0, 0)
                self._respond(sock, tok + done)

            def _ok(self, sock, nrows: int = 0):
                done = bytes([TOK_DONE]) + struct.pack("<HHQ", 0x10, 0, nrows)
                self._respond(sock, done)

            def _resultset(self, sock, cols, rows):
                out = bytes([TOK_COLMETADATA]) + struct.pack("<H", len(cols))
                for c in cols:
                    out += struct.pack("<IH", 0, 0)  # usertype, flags
                    out += b"\xe7" + struct.pack("<H", 8000)  # NVARCHAR(max len)
                    out += b"\x09\x04\xd0\x00\x34"  # collation
                    cn = c.encode("utf-16-le")
                    out += bytes([len(c)]) + cn
                for r in rows:
                    out += bytes([TOK_ROW])
                    for v in r:
                        if v is None:
                            out += struct.pack("<H", 0xFFFF)
                        else:
                            vb = str(v).encode("utf-16-le")
                            out += struct.pack("<H", len(vb)) + vb
                out += bytes([TOK_DONE]) + struct.pack("<HHQ", 0x10, 0, len(rows))
                self._respond(sock, out)

            # -- SQL subset --

            def _query(self, sock, sql: str):
                s = sql.strip().rstrip(";").strip("\x00").strip()
                up = s.upper()
                try:
                    if up.startswith("CREATE TABLE"):
                        m = re.match(
                            r"CREATE TABLE (?:IF NOT EXISTS )?\[?(\w+)\]?\s*\((.*)\)",
                            s, re.I | re.S)
                        name = m.group(1)
                        cols = [c.strip().split()[0].strip("[]")
                                for c in m.group(2).split(",")]
                        with store.lock:
                            store.tables.setdefault(name, (cols, []))
                        return self._ok(sock)
                    if up.startswith("INSERT INTO"):
                        m = re.match(
                            r"INSERT INTO \[?(\w+)\]?\s*\(([^)]*)\)\s*VALUES\s*(.*)",
                            s, re.I | re.S)
                        name = m.group(1)
                        cols = [c.strip().strip("[]") for c in m.group(2).split(",")]
                        tuples = re.findall(r"\(([^)]*)\)", m.group(3))
                        with store.lock:
                            tcols, rows = store.tables.setdefault(name, (cols, []))
                            for tup in tuples:
                                vals = [store._parse_value(v)
                                        for v in store._split_values(tup)]
                                rec = dict(zip(cols, vals))
                                rows.append([rec.get(c) for c in tcols])
                        return self._ok(sock, len(tuples))
                    if up.startswith("DELETE FROM"):
                        m = re.match(r"DELETE FROM \[?(\w+)\]?(?:\s+WHERE\s+(.*))?",
                                     s, re.I | re.S)
                        name, cond = m.group(1), m.group(2)
                        with store.lock:
                            tcols, rows = store.tables.get(name, ([], []))
                            keep = [r for r in rows
                                    if not store._match(tcols, r, cond)]
                            removed = len(rows) - len(keep)
                            rows[:] = keep
                        return self._ok(sock, removed)
                    if up.startswith("SELECT"):
                        m = re.match(r"SELECT \* FROM \[?(\w+)\]?(?:\s+WHERE\s+(.*))?",
                                     s, re.I | re.S)
                        if not m:
                            return self._error(sock, f"unsupported SELECT {s}")
                        name, cond = m.group(1), m.group(2)
                        with store.lock:
                            tcols, rows = store.tables.get(name, ([], []))
                            out = [r for r in rows
                                   if cond is None or store._match(tcols, r, cond)]
                        return self._resultset(sock, tcols, out)
                    return self._error(sock, f"unsupported statement {s}")
                except Exception as e:
                    return self._error(sock, f"{type(e).__name__}: {e}")

        class Server(socketserver.ThreadingTCPServer):
            allow_reuse_address = True
            daemon_threads = True

        self.server = Server(("127.0.0.1", 0), Handler)
        self.thread = threading.Thread(target=self.server.serve_forever, daemon=True)

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
                    cur += "\x01"
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
        v = v.strip()
        if v.startswith("N\x01") or v.startswith("\x01"):
            return v.lstrip("N").lstrip("\x01")
        if v.startswith("N'"):
            return v[2:-1]
        if v.upper() == "NULL":
            return None
        try:
            return int(v)
        except ValueError:
            try:
                return float(v)
            except ValueError:
                return v.lstrip("N\x01")

    @staticmethod
    def _match(cols, row, cond) -> bool:
        if cond is None:
            return True
        for clause in re.split(r"\s+AND\s+", cond, flags=re.I):
            m = re.match(r"\s*\[?(\w+)\]?\s*(>|=)\s*(.*)", clause.strip())
            if not m:
                return False
            col, op, val = m.groups()
            val = val.strip()
            if val.startswith("N'"):
                val = val[2:-1].replace("''", "'")
            elif val.startswith("'"):
                val = val[1:-1].replace("''", "'")
            try:
                got = row[cols.index(col)]
            except ValueError:
                return False
            if op == "=":
                if str(got) != str(val):
                    return False
            else:
                try:
                    if not float(got) > float(val):
                        return False
                except (TypeError, ValueError):
                    if not str(got) > str(val):
                        return False
        return True

    @property
    def port(self) -> int:
        return self.server.server_address[1]

    def start(self) -> "FakeMSSQL":
        self.thread.start()
        return self

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()
