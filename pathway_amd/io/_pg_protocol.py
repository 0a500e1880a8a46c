"""Pure-python PostgreSQL wire-protocol (v3) client.

Replaces the reference's postgres crate + pg_walstream CDC
(src/connectors/data_storage/postgres.rs, 4,547 LoC) with a from-scratch
implementation of the frontend/backend protocol: startup (trust auth),
simple query, and logical streaming replication (START_REPLICATION →
CopyBoth → XLogData frames carrying wal2json-style payloads).
Exercised against tests/fakes/fake_postgres.py, which speaks the same
protocol bytes.
"""

from __future__ import annotations

import socket
import struct
import threading
from typing import Any, Iterator


class PgError(RuntimeError):
    pass


class PgClient:
    def __init__(self, host: str = "127.0.0.1", port: int = 5432, *,
                 user: str = "postgres", database: str = "postgres",
                 password: str | None = None, replication: bool = False,
                 timeout: float = 30.0):
        self.sock = socket.create_connection((host, port), timeout=timeout)
        self.lock = threading.Lock()
        params = {"user": user, "database": database,
                  "client_encoding": "UTF8"}
        if replication:
            params["replication"] = "database"
        body = b""
        for k, v in params.items():
            body += k.encode() + b"\x00" + v.encode() + b"\x00"
        body += b"\x00"
        payload = struct.pack(">ii", 8 + len(body), 196608) + body
        self.sock.sendall(payload)
        # consume messages until ReadyForQuery
        while True:
            mtype, data = self._recv_message()
            if mtype == b"R":
                (code,) = struct.unpack_from(">i", data, 0)
                if code == 0:
                    continue
                if code == 3:  # cleartext password
                    pw = (password or "").encode() + b"\x00"
                    self._send(b"p", pw)
                    continue
                raise PgError(f"unsupported auth method {code}")
            if mtype == b"Z":
                return
            if mtype == b"E":
                raise PgError(self._parse_error(data))
            # S (parameter status), K (backend key data): ignore

    # -- low-level --

    def _send(self, mtype: bytes, body: bytes) -> None:
        self.sock.sendall(mtype + struct.pack(">i", 4 + len(body)) + body)

    def _recv_exact(self, n: int) -> bytes:
        buf = b""
        while len(buf) < n:
            chunk = self.sock.recv(n - len(buf))
            if not chunk:
                raise PgError("server closed connection")
            buf += chunk
        return buf

    def _recv_message(self) -> tuple[bytes, bytes]:
        head = self._recv_exact(5)
        mtype = head[:1]
        (length,) = struct.unpack(">i", head[1:])
        data = self._recv_exact(length - 4) if length > 4 else b""
        return mtype, data

    @staticmethod
    def _parse_error(data: bytes) -> str:
        parts = {}
        i = 0
        while i < len(data) and data[i] != 0:
            code = chr(data[i])
            z = data.index(b"\x00", i + 1)
            parts[code] = data[i + 1 : z].decode("utf-8", "replace")
            i = z + 1
        return parts.get("M", repr(parts))

    def close(self) -> None:
        try:
            self._send(b"X", b"")
            self.sock.close()
        except OSError:
            pass

    # -- simple query --

    def query(self, sql: str) -> tuple[list[str], list[list[str | None]]]:
        """Simple-query protocol: returns (column names, text rows)."""
        with self.lock:
            self._send(b"Q", sql.encode() + b"\x00")
            columns: list[str] = []
            rows: list[list[str | None]] = []
            error: str | None = None
            while True:
                mtype, data = self._recv_message()
                if mtype == b"T":  # RowDescription
                    (nfields,) = struct.unpack_from(">h", data, 0)
                    i = 2
                    columns = []
                    for _ in range(nfields):
                        z = data.index(b"\x00", i)
                        columns.append(data[i:z].decode())
                        i = z + 1 + 18  # name + table oid(4) attnum(2) type oid(4) typlen(2) atttypmod(4) format(2)
                elif mtype == b"D":  # DataRow
                    (nfields,) = struct.unpack_from(">h", data, 0)
                    i = 2
                    row: list[str | None] = []
                    for _ in range(nfields):
                        (flen,) = struct.unpack_from(">i", data, i)
                        i += 4
                        if flen < 0:
                            row.append(None)
                        else:
                            row.append(data[i : i + flen].decode())
                            i += flen
                    rows.append(row)
                elif mtype == b"C":  # CommandComplete
                    pass
                elif mtype == b"E":
                    error = self._parse_error(data)
                elif mtype == b"Z":
                    if error:
                        raise PgError(error)
                    return columns, rows
                # N (notice), S: ignore

    # -- logical replication --

    def start_replication(self, slot: str, *, options: dict[str, str] | None = None,
                          start_lsn: str = "0/0") -> Iterator[tuple[int, bytes]]:
        """START_REPLICATION ... LOGICAL: yields (wal_end_lsn, payload).

        The connection must have been opened with replication=True.
        Yields until the server ends the copy stream; the caller can
        close() the socket to stop.
        """
        opts = ""
        if options:
            opts = " (" + ", ".join(f"\"{k}\" '{v}'" for k, v in options.items()) + ")"
        sql = f"START_REPLICATION SLOT {slot} LOGICAL {start_lsn}{opts}"
        self._send(b"Q", sql.encode() + b"\x00")
        mtype, data = self._recv_message()
        if mtype == b"E":
            raise PgError(self._parse_error(data))
        if mtype != b"W":  # CopyBothResponse
            raise PgError(f"expected CopyBothResponse, got {mtype}")
        while True:
            mtype, data = self._recv_message()
            if mtype == b"d":  # CopyData
                kind = data[:1]
                if kind == b"w":  # XLogData
                    _start, wal_end, _ts = struct.unpack_from(">qqq", data, 1)
                    payload = data[25:]
                    yield wal_end, payload
                elif kind == b"k":  # keepalive
                    wal_end, _ts, reply = struct.unpack_from(">qqb", data, 1)
                    if reply:
                        # standby status update: all positions = wal_end
                        msg = b"r" + struct.pack(
                            ">qqqqb", wal_end, wal_end, wal_end, 0, 0
                        )
                        self._send(b"d", msg)
            elif mtype in (b"c", b"C"):  # CopyDone / CommandComplete
                return
            elif mtype == b"Z":
                return
            elif mtype == b"E":
                raise PgError(self._parse_error(data))


def client_from_settings(settings: dict[str, Any], *, replication: bool = False) -> PgClient:
    return PgClient(
        host=settings.get("host", "127.0.0.1"),
        port=int(settings.get("port", 5432)),
        user=settings.get("user", "postgres"),
        database=settings.get("dbname", settings.get("database", "postgres")),
        password=settings.get("password"),
        replication=replication,
    )


def quote_literal(v: Any) -> str:
    if v is None:
        return "NULL"
    if isinstance(v, bool):
        return "TRUE" if v else "FALSE"
    if isinstance(v, (int, float)):
        return str(v)
    if isinstance(v, bytes):
        return "'\\x" + v.hex() + "'"
    s = str(v).replace("'", "''")
    return f"'{s}'"


def quote_ident(name: str) -> str:
    return '"' + name.replace('"', '""') + '"'
