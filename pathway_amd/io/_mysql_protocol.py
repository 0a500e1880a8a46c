"""Pure-python MySQL client protocol (v10 handshake + COM_QUERY).

Replaces the reference's mysql_async dependency (src/connectors/
data_storage/mysql.rs, 2,118 LoC) with a from-scratch implementation of
the MySQL client/server protocol: 3-byte-length packets,
HandshakeResponse41 with mysql_native_password, COM_QUERY text result
sets (length-encoded integers/strings).  Works against real servers;
exercised against tests/fakes/fake_mysql.py speaking the same bytes.
"""

from __future__ import annotations

import hashlib
import socket
import struct
import threading
from typing import Any

CLIENT_LONG_PASSWORD = 0x1
CLIENT_PROTOCOL_41 = 0x200
CLIENT_SECURE_CONNECTION = 0x8000
CLIENT_PLUGIN_AUTH = 0x80000
CLIENT_CONNECT_WITH_DB = 0x8


class MySQLError(RuntimeError):
    pass


def native_password_auth(password: str, nonce: bytes) -> bytes:
    """SHA1(pass) XOR SHA1(nonce + SHA1(SHA1(pass)))."""
    if not password:
        return b""
    p1 = hashlib.sha1(password.encode()).digest()
    p2 = hashlib.sha1(p1).digest()
    p3 = hashlib.sha1(nonce + p2).digest()
    return bytes(a ^ b for a, b in zip(p1, p3))


def read_lenenc(data: bytes, i: int) -> tuple[int | None, int]:
    b = data[i]
    if b < 0xFB:
        return b, i + 1
    if b == 0xFB:
        return None, i + 1  # NULL
    if b == 0xFC:
        return struct.unpack_from("<H", data, i + 1)[0], i + 3
    if b == 0xFD:
        return int.from_bytes(data[i + 1 : i + 4], "little"), i + 4
    return struct.unpack_from("<Q", data, i + 1)[0], i + 9


def write_lenenc(n: int) -> bytes:
    if n < 0xFB:
        return bytes([n])
    if n < 1 << 16:
        return b"\xfc" + struct.pack("<H", n)
    if n < 1 << 24:
        return b"\xfd" + n.to_bytes(3, "little")
    return b"\xfe" + struct.pack("<Q", n)


class MySQLClient:
    def __init__(self, host: str = "127.0.0.1", port: int = 3306, *,
                 user: str = "root", password: str = "",
                 database: str = "", timeout: float = 30.0):
        self.sock = socket.create_connection((host, port), timeout=timeout)
        self.lock = threading.Lock()
        self.seq = 0
        self._handshake(user, password, database)

    # -- packets --

    def _recv_exact(self, n: int) -> bytes:
        buf = b""
        while len(buf) < n:
            chunk = self.sock.recv(n - len(buf))
            if not chunk:
                raise MySQLError("server closed connection")
            buf += chunk
        return buf

    def _read_packet(self) -> bytes:
        head = self._recv_exact(4)
        length = int.from_bytes(head[:3], "little")
        self.seq = head[3] + 1
        return self._recv_exact(length)

    def _send_packet(self, payload: bytes) -> None:
        head = len(payload).to_bytes(3, "little") + bytes([self.seq])
        self.seq += 1
        self.sock.sendall(head + payload)

    # -- handshake --

    def _handshake(self, user: str, password: str, database: str) -> None:
        greet = self._read_packet()
        if greet[0] == 0xFF:
            raise MySQLError(self._parse_err(greet))
        if greet[0] != 10:
            raise MySQLError(f"unsupported protocol version {greet[0]}")
        i = 1
        z = greet.index(b"\x00", i)
        i = z + 1  # server version
        i += 4  # thread id
        nonce = greet[i : i + 8]
        i += 8 + 1  # auth data part 1 + filler
        i += 2 + 1 + 2 + 2  # caps low, charset, status, caps high
        if len(greet) > i:
            auth_len = greet[i]
            i += 1 + 10  # auth len + reserved
            extra = max(13, auth_len - 8) if auth_len else 13
            part2 = greet[i : i + extra].rstrip(b"\x00")
            nonce = nonce + part2
        caps = (CLIENT_LONG_PASSWORD | CLIENT_PROTOCOL_41
                | CLIENT_SECURE_CONNECTION | CLIENT_PLUGIN_AUTH)
        if database:
            caps |= CLIENT_CONNECT_WITH_DB
        auth = native_password_auth(password, nonce[:20])
        resp = struct.pack("<IIB23x", caps, 1 << 24, 33)
        resp += user.encode() + b"\x00"
        resp += bytes([len(auth)]) + auth
        if database:
            resp += database.encode() + b"\x00"
        resp += b"mysql_native_password\x00"
        self._send_packet(resp)
        ok = self._read_packet()
        if ok[0] == 0xFF:
            raise MySQLError(self._parse_err(ok))

    @staticmethod
    def _parse_err(pkt: bytes) -> str:
        (code,) = struct.unpack_from("<H", pkt, 1)
        msg = pkt[9:].decode("utf-8", "replace")
        return f"mysql error {code}: {msg}"

    # -- queries --

    def query(self, sql: str) -> tuple[list[str], list[list[str | None]]]:
        """COM_QUERY with a text resultset -> (columns, rows)."""
        with self.lock:
            self.seq = 0
            self._send_packet(b"\x03" + sql.encode())
            first = self._read_packet()
            if first[0] == 0xFF:
                raise MySQLError(self._parse_err(first))
            if first[0] == 0x00:
                return [], []  # OK packet (no resultset)
            ncols, _ = read_lenenc(first, 0)
            columns = []
            for _ in range(ncols):
                cd = self._read_packet()
                # column def: catalog, schema, table, org_table, name, ...
                i = 0
                vals = []
                for _f in range(5):
                    ln, i = read_lenenc(cd, i)
                    vals.append(cd[i : i + (ln or 0)])
                    i += ln or 0
                columns.append(vals[4].decode())
            pkt = self._read_packet()
            if pkt[0] == 0xFE and len(pkt) < 9:
                pkt = self._read_packet()  # EOF after columns
            rows: list[list[str | None]] = []
            while True:
                if pkt[0] == 0xFE and len(pkt) < 9:
                    break  # EOF
                if pkt[0] == 0xFF:
                    raise MySQLError(self._parse_err(pkt))
                i = 0
                row: list[str | None] = []
                for _ in range(ncols):
                    ln, i = read_lenenc(pkt, i)
                    if ln is None:
                        row.append(None)
                    else:
                        row.append(pkt[i : i + ln].decode())
                        i += ln
                rows.append(row)
                pkt = self._read_packet()
            return columns, rows

    def close(self) -> None:
        try:
            with self.lock:
                self.seq = 0
                self._send_packet(b"\x01")  # COM_QUIT
            self.sock.close()
        except OSError:
            pass


def quote_literal(v: Any) -> str:
    if v is None:
        return "NULL"
    if isinstance(v, bool):
        return "1" if v else "0"
    if isinstance(v, (int, float)):
        return str(v)
    s = str(v).replace("\\", "\\\\").replace("'", "\\'")
    return f"'{s}'"


def quote_ident(name: str) -> str:
    return "`" + name.replace("`", "``") + "`"
