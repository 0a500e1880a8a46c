"""BSON encode/decode (reference src/connectors/data_format/bson.rs).

Pure-python implementation of the BSON spec subset the reference maps to
engine values: double, string, document, array, binary, ObjectId, bool,
UTC datetime (ms), null, int32, int64, decimal128 passed through as bytes.
Used by the MongoDB wire-protocol client (io/mongodb.py) and the bson
parser/formatter.
"""

from __future__ import annotations

import datetime
import os
import struct
import threading
import time
from typing import Any

T_DOUBLE = 0x01
T_STRING = 0x02
T_DOC = 0x03
T_ARRAY = 0x04
T_BINARY = 0x05
T_OBJECTID = 0x07
T_BOOL = 0x08
T_DATETIME = 0x09
T_NULL = 0x0A
T_REGEX = 0x0B
T_INT32 = 0x10
T_TIMESTAMP = 0x11
T_INT64 = 0x12


class ObjectId:
    """12-byte MongoDB ObjectId: 4B unix time + 5B random + 3B counter."""

    _counter = int.from_bytes(os.urandom(3), "big")
    _random = os.urandom(5)
    _lock = threading.Lock()

    __slots__ = ("binary",)

    def __init__(self, binary: bytes | str | None = None):
        if binary is None:
            with ObjectId._lock:
                ObjectId._counter = (ObjectId._counter + 1) & 0xFFFFFF
                counter = ObjectId._counter
            self.binary = (
                struct.pack(">I", int(time.time()))
                + ObjectId._random
                + counter.to_bytes(3, "big")
            )
        elif isinstance(binary, str):
            self.binary = bytes.fromhex(binary)
        else:
            self.binary = bytes(binary)
        if len(self.binary) != 12:
            raise ValueError("ObjectId must be 12 bytes")

    def __eq__(self, other: object) -> bool:
        return isinstance(other, ObjectId) and self.binary == other.binary

    def __lt__(self, other: "ObjectId") -> bool:
        return self.binary < other.binary

    def __gt__(self, other: "ObjectId") -> bool:
        return self.binary > other.binary

    def __hash__(self) -> int:
        return hash(self.binary)

    def __repr__(self) -> str:
        return f"ObjectId({self.binary.hex()!r})"

    def __str__(self) -> str:
        return self.binary.hex()


class Binary(bytes):
    """BSON binary with a subtype (default 0)."""

    subtype = 0


def _cstring(s: str) -> bytes:
    b = s.encode("utf-8")
    if b"\x00" in b:
        raise ValueError("BSON keys cannot contain NUL")
    return b + b"\x00"


def encode(doc: dict[str, Any]) -> bytes:
    body = bytearray()
    for k, v in doc.items():
        _encode_element(body, k, v)
    return struct.pack("<i", len(body) + 5) + bytes(body) + b"\x00"


def _encode_element(out: bytearray, key: str, v: Any) -> None:
    name = _cstring(key)
    if v is None:
        out += bytes([T_NULL]) + name
    elif isinstance(v, bool):
        out += bytes([T_BOOL]) + name + (b"\x01" if v else b"\x00")
    elif isinstance(v, ObjectId):
        out += bytes([T_OBJECTID]) + name + v.binary
    elif isinstance(v, int):
        if -(2**31) <= v < 2**31:
            out += bytes([T_INT32]) + name + struct.pack("<i", v)
        else:
            out += bytes([T_INT64]) + name + struct.pack("<q", v)
    elif isinstance(v, float):
        out += bytes([T_DOUBLE]) + name + struct.pack("<d", v)
    elif isinstance(v, str):
        b = v.encode("utf-8") + b"\x00"
        out += bytes([T_STRING]) + name + struct.pack("<i", len(b)) + b
    elif isinstance(v, (bytes, bytearray)):
        sub = getattr(v, "subtype", 0)
        out += (
            bytes([T_BINARY])
            + name
            + struct.pack("<i", len(v))
            + bytes([sub])
            + bytes(v)
        )
    elif isinstance(v, datetime.datetime):
        ms = int(v.timestamp() * 1000)
        out += bytes([T_DATETIME]) + name + struct.pack("<q", ms)
    elif isinstance(v, dict):
        out += bytes([T_DOC]) + name + encode(v)
    elif isinstance(v, (list, tuple)):
        out += bytes([T_ARRAY]) + name + encode(
            {str(i): x for i, x in enumerate(v)}
        )
    else:
        raise TypeError(f"cannot BSON-encode {type(v).__name__}")


def decode(data: bytes, offset: int = 0) -> dict[str, Any]:
    doc, _ = _decode_doc(data, offset)
    return doc


def decode_all(data: bytes) -> list[dict[str, Any]]:
    out = []
    i = 0
    while i < len(data):
        doc, i = _decode_doc(data, i)
        out.append(doc)
    return out


def _decode_doc(data: bytes, i: int) -> tuple[dict[str, Any], int]:
    (doclen,) = struct.unpack_from("<i", data, i)
    end = i + doclen
    i += 4
    out: dict[str, Any] = {}
    while data[i] != 0:
        t = data[i]
        i += 1
        z = data.index(b"\x00", i)
        key = data[i:z].decode("utf-8")
        i = z + 1
        if t == T_NULL:
            out[key] = None
        elif t == T_BOOL:
            out[key] = data[i] != 0
            i += 1
        elif t == T_INT32:
            (out[key],) = struct.unpack_from("<i", data, i)
            i += 4
        elif t in (T_INT64, T_TIMESTAMP):
            (out[key],) = struct.unpack_from("<q", data, i)
            i += 8
        elif t == T_DOUBLE:
            (out[key],) = struct.unpack_from("<d", data, i)
            i += 8
        elif t == T_STRING:
            (slen,) = struct.unpack_from("<i", data, i)
            i += 4
            out[key] = data[i : i + slen - 1].decode("utf-8")
            i += slen
        elif t == T_BINARY:
            (blen,) = struct.unpack_from("<i", data, i)
            i += 5  # len + subtype byte
            out[key] = bytes(data[i : i + blen])
            i += blen
        elif t == T_OBJECTID:
            out[key] = ObjectId(data[i : i + 12])
            i += 12
        elif t == T_DATETIME:
            (ms,) = struct.unpack_from("<q", data, i)
            i += 8
            out[key] = datetime.datetime.fromtimestamp(
                ms / 1000.0, tz=datetime.timezone.utc
            )
        elif t == T_DOC:
            out[key], i = _decode_doc(data, i)
        elif t == T_ARRAY:
            sub, i = _decode_doc(data, i)
            out[key] = [sub[str(j)] for j in range(len(sub))]
        else:
            raise ValueError(f"unsupported BSON element type 0x{t:02x}")
    return out, end
