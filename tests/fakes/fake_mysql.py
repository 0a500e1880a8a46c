"""In-process fake MySQL server (protocol v10 + COM_QUERY subset)."""

from __future__ import annotations

import re
import socketserver
import struct
import threading

from pathway_amd.io._mysql_protocol import write_lenenc


class FakeMySQL:
    def __init__(self):
        #: table -> (columns, rows)
        self.tables: dict[str, tuple[list[str], list[list]]] = {}
        self.lock = threading.Lock()
        store = self

        class Handler(socketserver.BaseRequestHandler):
            def handle(self):
                sock = self.request
                self.seq = 0
                try:
                    # initial handshake
                    greet = (
                        b"\x0a" + b"8.0.0-fake\x00"
                        + struct.pack("<I", 1)
                        + b"12345678" + b"\x00"
                        + struct.pack("<H", 0xFFFF)  # caps low
                        + b"\x21" + struct.pack("<H", 2)
                        + struct.pack("<H", 0xC000)  # caps high
                        + bytes([21]) + b"\x00" * 10
                        + b"901234567890\x00"
                        + b"mysql_native_password\x00"
                    )
                    self._send(sock, greet)
                    self._recv_packet(sock)  # handshake response (trust all)
                    self._send(sock, b"\x00\x00\x00\x02\x00\x00\x00")  # OK
                    while True:
                        self.seq = 0
                        pkt = self._recv_packet(sock)
                        if pkt is None or pkt[:1] == b"\x01":  # COM_QUIT
                            return
                        if pkt[:1] == b"\x03":
                            self._query(sock, pkt[1:].decode())
                except (ConnectionResetError, BrokenPipeError, OSError):
                    return

            def _recv_packet(self, sock):
                head = self._recv_exact(sock, 4)
                if head is None:
                    return None
                length = int.from_bytes(head[:3], "little")
                self.seq = head[3] + 1
                return self._recv_exact(sock, length)

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

            def _send(self, sock, payload: bytes):
                sock.sendall(
                    len(payload).to_bytes(3, "little") + bytes([self.seq])
                    + payload
                )
                self.seq += 1

            def _err(self, sock, msg: str):
                self._send(sock, b"\xff" + struct.pack("<H", 1064)
                           + b"#42000" + msg.encode())

            def _ok(self, sock):
                self._send(sock, b"\x00\x00\x00\x02\x00\x00\x00")

            def _query(self, sock, sql: str):
                s = sql.strip().rstrip(";")
                up = s.upper()
                try:
                    if up.startswith("CREATE TABLE"):
                        m = re.match(
                            r"CREATE TABLE (?:IF NOT EXISTS )?`?(\w+)`?\s*\((.*)\)",
                            s, re.I | re.S)
                        name = m.group(1)
                        cols = [c.strip().split()[0].strip("`")
                                for c in m.group(2).split(",")]
                        with store.lock:
                            store.tables.setdefault(name, (cols, []))
                        return self._ok(sock)
                    if up.startswith("INSERT INTO"):
                        m = re.match(
                            r"INSERT INTO `?(\w+)`?\s*\(([^)]*)\)\s*VALUES\s*(.*)",
                            s, re.I | re.S)
                        name = m.group(1)
                        cols = [c.strip().strip("`") for c in m.group(2).split(",")]
                        tuples = re.findall(r"\(([^)]*)\)", m.group(3))
                        with store.lock:
                            tcols, rows = store.tables.setdefault(name, (cols, []))
                            for tup in tuples:
                                vals = [store._parse_value(v)
                                        for v in store._split_values(tup)]
                                rec = dict(zip(cols, vals))
                                rows.append([rec.get(c) for c in tcols])
                        return self._ok(sock)
                    if up.startswith("DELETE FROM"):
                        m = re.match(r"DELETE FROM `?(\w+)`?(?:\s+WHERE\s+(.*))?",
                                     s, re.I | re.S)
                        name = m.group(1)
                        cond = m.group(2)
                        with store.lock:
                            tcols, rows = store.tables.get(name, ([], []))
                            keep = [r for r in rows
                                    if not store._match(tcols, r, cond)]
                            rows[:] = keep
                        return self._ok(sock)
                    if up.startswith("SELECT"):
                        m = re.match(
                            r"SELECT \* FROM `?(\w+)`?"
                            r"(?:\s+WHERE\s+(.*?))?(?:\s+ORDER BY\s+\S+)?$",
                            s, re.I | re.S)
                        if not m:
                            return self._err(sock, f"unsupported SELECT {s}")
                        name = m.group(1)
                        cond = m.group(2)
                        with store.lock:
                            tcols, rows = store.tables.get(name, ([], []))
                            out = [r for r in rows
                                   if cond is None or store._match(tcols, r, cond)]
                        self._send(sock, write_lenenc(len(tcols)))
                        for c in tcols:
                            cd = b""
                            for val in (b"def", b"db", b"t", b"t", c.encode(),
                                        c.encode()):
                                cd += write_lenenc(len(val)) + val
                            cd += b"\x0c" + struct.pack("<HIBHB", 33, 255, 253, 0, 0)
                            cd += b"\x00\x00"
                            self._send(sock, cd)
                        self._send(sock, b"\xfe\x00\x00\x02\x00")  # EOF
                        for r in out:
                            rp = b""
                            for v in r:
                                if v is None:
                                    rp += b"\xfb"
                                else:
                                    vb = str(v).encode()
                                    rp += write_lenenc(len(vb)) + vb
                            self._send(sock, rp)
                        self._send(sock, b"\xfe\x00\x00\x02\x00")  # EOF
                        return
                    return self._err(sock, f"unsupported statement {s}")
                except Exception as e:
                    return self._err(sock, f"{type(e).__name__}: {e}")

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
                if ch == "\\" and i + 1 < len(tup):
                    cur += tup[i + 1]
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
        if v.startswith("\x01"):
            return v[1:]
        if v.upper() == "NULL":
            return None
        try:
            return int(v)
        except ValueError:
            try:
                return float(v)
            except ValueError:
                return v

    @staticmethod
    def _match(cols, row, cond) -> bool:
        if cond is None:
            return True
        for clause in re.split(r"\s+AND\s+", cond, flags=re.I):
            m = re.match(r"\s*`?(\w+)`?\s*(>|=)\s*(.*)", clause.strip())
            if not m:
                return False
            col, op, val = m.groups()
            val = val.strip()
            if val.startswith("'"):
                val = val[1:-1]
            try:
                got = row[cols.index(col)]
            except ValueError:
                return False
            if op == "=":
                if str(got) != str(val):
                    return False
            else:  # >
                try:
                    if not (float(got) > float(val)):
                        return False
                except (TypeError, ValueError):
                    if not (str(got) > str(val)):
                        return False
        return True

    @property
    def port(self) -> int:
        return self.server.server_address[1]

    def start(self) -> "FakeMySQL":
        self.thread.start()
        return self

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()
