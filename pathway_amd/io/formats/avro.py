"""Avro binary codec + Object Container Files + Confluent wire framing.

Reference semantics: src/connectors/data_format/avro.rs (parser/formatter
over apache-avro) — this is an original pure-python implementation of the
Avro 1.11 binary spec subset the reference exercises: primitives, records,
enums, arrays, maps, unions, fixed, and the logical types the reference
maps to engine values (timestamp-millis/micros -> DateTimeUtc,
local-timestamp -> DateTimeNaive, decimal unsupported -> bytes).

Schemas are plain parsed-JSON structures (dict/list/str), as produced by
``json.loads`` of an Avro schema document.
"""

from __future__ import annotations

import io
import json
import os
import struct
import zlib
from typing import Any, BinaryIO

MAGIC = b"Obj\x01"

PRIMITIVES = {"null", "boolean", "int", "long", "float", "double", "bytes", "string"}


# ---------------------------------------------------------------------------
# varint / zigzag
# ---------------------------------------------------------------------------

def _zigzag_encode(n: int) -> int:
    return (n << 1) ^ (n >> 63) if n >= 0 else ((-n) << 1) - 1


def write_long(buf: bytearray, n: int) -> None:
    z = (n << 1) ^ (n >> 63)
    z &= (1 << 64) - 1
    while True:
        b = z & 0x7F
        z >>= 7
        if z:
            buf.append(b | 0x80)
        else:
            buf.append(b)
            return


def read_long(r: BinaryIO) -> int:
    shift = 0
    acc = 0
    while True:
        byte = r.read(1)
        if not byte:
            raise EOFError("truncated avro varint")
        b = byte[0]
        acc |= (b & 0x7F) << shift
        if not b & 0x80:
            break
        shift += 7
    return (acc >> 1) ^ -(acc & 1)


# ---------------------------------------------------------------------------
# schema helpers
# ---------------------------------------------------------------------------

def _named(schema: Any) -> str | None:
    if isinstance(schema, dict):
        return schema.get("name")
    return None


def resolve_names(schema: Any, names: dict[str, Any] | None = None) -> dict[str, Any]:
    """Collect named types (records/enums/fixed) for name references."""
    if names is None:
        names = {}
    if isinstance(schema, dict):
        t = schema.get("type")
        if t in ("record", "enum", "fixed") and "name" in schema:
            names[schema["name"]] = schema
            full = schema.get("namespace")
            if full:
                names[f"{full}.{schema['name']}"] = schema
        if t == "record":
            for f in schema.get("fields", []):
                resolve_names(f.get("type"), names)
        elif t == "array":
            resolve_names(schema.get("items"), names)
        elif t == "map":
            resolve_names(schema.get("values"), names)
    elif isinstance(schema, list):
        for s in schema:
            resolve_names(s, names)
    return names


def _schema_type(schema: Any, names: dict[str, Any]) -> Any:
    if isinstance(schema, str):
        if schema in PRIMITIVES:
            return schema
        if schema in names:
            return names[schema]
        raise ValueError(f"unknown avro type name {schema!r}")
    return schema


# ---------------------------------------------------------------------------
# encode
# ---------------------------------------------------------------------------

def encode(value: Any, schema: Any, buf: bytearray, names: dict[str, Any] | None = None) -> None:
    if names is None:
        names = resolve_names(schema)
    schema = _schema_type(schema, names)
    if isinstance(schema, str):
        t = schema
        if t == "null":
            return
        if t == "boolean":
            buf.append(1 if value else 0)
        elif t in ("int", "long"):
            write_long(buf, int(value))
        elif t == "float":
            buf += struct.pack("<f", float(value))
        elif t == "double":
            buf += struct.pack("<d", float(value))
        elif t == "bytes":
            write_long(buf, len(value))
            buf += bytes(value)
        elif t == "string":
            b = value.encode("utf-8")
            write_long(buf, len(b))
            buf += b
        else:
            raise ValueError(f"bad primitive {t}")
        return
    if isinstance(schema, list):  # union
        for i, branch in enumerate(schema):
            if _matches(value, branch, names):
                write_long(buf, i)
                encode(value, branch, buf, names)
                return
        raise ValueError(f"value {value!r} matches no union branch {schema!r}")
    t = schema["type"]
    if t == "record":
        for f in schema["fields"]:
            encode(value[f["name"]], f["type"], buf, names)
    elif t == "enum":
        buf_idx = schema["symbols"].index(value)
        write_long(buf, buf_idx)
    elif t == "array":
        items = list(value)
        if items:
            write_long(buf, len(items))
            for it in items:
                encode(it, schema["items"], buf, names)
        write_long(buf, 0)
    elif t == "map":
        if value:
            write_long(buf, len(value))
            for k, v in value.items():
                kb = k.encode("utf-8")
                write_long(buf, len(kb))
                buf += kb
                encode(v, schema["values"], buf, names)
        write_long(buf, 0)
    elif t == "fixed":
        if len(value) != schema["size"]:
            raise ValueError("fixed size mismatch")
        buf += bytes(value)
    elif t in PRIMITIVES:
        encode(value, t, buf, names)
    else:
        raise ValueError(f"unsupported avro schema {schema!r}")


def _matches(value: Any, schema: Any, names: dict[str, Any]) -> bool:
    schema = _schema_type(schema, names)
    t = schema if isinstance(schema, str) else schema.get("type")
    if t == "null":
        return value is None
    if value is None:
        return False
    if t == "boolean":
        return isinstance(value, bool)
    if t in ("int", "long"):
        return isinstance(value, int) and not isinstance(value, bool)
    if t in ("float", "double"):
        return isinstance(value, (int, float)) and not isinstance(value, bool)
    if t == "string":
        return isinstance(value, str)
    if t in ("bytes", "fixed"):
        return isinstance(value, (bytes, bytearray))
    if t == "record":
        return isinstance(value, dict)
    if t == "enum":
        return isinstance(value, str)
    if t == "array":
        return isinstance(value, (list, tuple))
    if t == "map":
        return isinstance(value, dict)
    return True


def encode_bytes(value: Any, schema: Any) -> bytes:
    buf = bytearray()
    encode(value, schema, buf)
    return bytes(buf)


# ---------------------------------------------------------------------------
# decode
# ---------------------------------------------------------------------------

def decode(r: BinaryIO, schema: Any, names: dict[str, Any] | None = None) -> Any:
    if names is None:
        names = resolve_names(schema)
    schema = _schema_type(schema, names)
    if isinstance(schema, str):
        t = schema
        if t == "null":
            return None
        if t == "boolean":
            return r.read(1)[0] != 0
        if t in ("int", "long"):
            return read_long(r)
        if t == "float":
            return struct.unpack("<f", r.read(4))[0]
        if t == "double":
            return struct.unpack("<d", r.read(8))[0]
        if t == "bytes":
            n = read_long(r)
            return r.read(n)
        if t == "string":
            n = read_long(r)
            return r.read(n).decode("utf-8")
        raise ValueError(f"bad primitive {t}")
    if isinstance(schema, list):
        idx = read_long(r)
        return decode(r, schema[idx], names)
    t = schema["type"]
    if t == "record":
        return {f["name"]: decode(r, f["type"], names) for f in schema["fields"]}
    if t == "enum":
        return schema["symbols"][read_long(r)]
    if t == "array":
        out = []
        while True:
            n = read_long(r)
            if n == 0:
                break
            if n < 0:
                read_long(r)  # block byte size, unused
                n = -n
            for _ in range(n):
                out.append(decode(r, schema["items"], names))
        return out
    if t == "map":
        out = {}
        while True:
            n = read_long(r)
            if n == 0:
                break
            if n < 0:
                read_long(r)
                n = -n
            for _ in range(n):
                klen = read_long(r)
                k = r.read(klen).decode("utf-8")
                out[k] = decode(r, schema["values"], names)
        return out
    if t == "fixed":
        return r.read(schema["size"])
    if t in PRIMITIVES:
        return decode(r, t, names)
    raise ValueError(f"unsupported avro schema {schema!r}")


def decode_bytes(data: bytes, schema: Any) -> Any:
    return decode(io.BytesIO(data), schema)


# ---------------------------------------------------------------------------
# Object Container Files (the `.avro` file format)
# ---------------------------------------------------------------------------

class ContainerWriter:
    """Avro Object Container File writer (deflate or null codec)."""

    def __init__(self, f: BinaryIO, schema: Any, codec: str = "null",
                 sync_marker: bytes | None = None):
        self.f = f
        self.schema = schema
        self.names = resolve_names(schema)
        self.codec = codec
        self.sync = sync_marker or os.urandom(16)
        self._block: bytearray = bytearray()
        self._count = 0
        header = bytearray(MAGIC)
        meta = {
            "avro.schema": json.dumps(schema).encode(),
            "avro.codec": codec.encode(),
        }
        write_long(header, len(meta))
        for k, v in meta.items():
            kb = k.encode()
            write_long(header, len(kb))
            header += kb
            write_long(header, len(v))
            header += v
        write_long(header, 0)
        header += self.sync
        f.write(bytes(header))

    def append(self, value: Any) -> None:
        encode(value, self.schema, self._block, self.names)
        self._count += 1
        if len(self._block) > 64 * 1024:
            self.flush_block()

    def flush_block(self) -> None:
        if not self._count:
            return
        data = bytes(self._block)
        if self.codec == "deflate":
            data = zlib.compress(data)[2:-4]  # raw deflate per avro spec
        out = bytearray()
        write_long(out, self._count)
        write_long(out, len(data))
        out += data
        out += self.sync
        self.f.write(bytes(out))
        self._block = bytearray()
        self._count = 0

    def close(self) -> None:
        self.flush_block()
        self.f.flush()


def read_container(f: BinaryIO):
    """Yield values from an Avro Object Container File."""
    if f.read(4) != MAGIC:
        raise ValueError("not an avro container file")
    meta: dict[str, bytes] = {}
    while True:
        n = read_long(f)
        if n == 0:
            break
        if n < 0:
            read_long(f)
            n = -n
        for _ in range(n):
            klen = read_long(f)
            k = f.read(klen).decode()
            vlen = read_long(f)
            meta[k] = f.read(vlen)
    schema = json.loads(meta["avro.schema"])
    codec = meta.get("avro.codec", b"null").decode()
    names = resolve_names(schema)
    sync = f.read(16)
    while True:
        try:
            count = read_long(f)
        except EOFError:
            return
        size = read_long(f)
        data = f.read(size)
        if codec == "deflate":
            data = zlib.decompress(data, -15)
        elif codec != "null":
            raise ValueError(f"unsupported avro codec {codec}")
        r = io.BytesIO(data)
        for _ in range(count):
            yield decode(r, schema, names)
        if f.read(16) != sync:
            raise ValueError("avro sync marker mismatch")


# ---------------------------------------------------------------------------
# Confluent schema-registry wire framing (magic 0 + schema id + payload)
# ---------------------------------------------------------------------------

def confluent_encode(value: Any, schema: Any, schema_id: int) -> bytes:
    return b"\x00" + struct.pack(">I", schema_id) + encode_bytes(value, schema)


def confluent_decode(data: bytes) -> tuple[int, bytes]:
    """Split Confluent framing -> (schema_id, avro payload)."""
    if not data or data[0] != 0:
        raise ValueError("not Confluent-framed avro (magic byte != 0)")
    (schema_id,) = struct.unpack_from(">I", data, 1)
    return schema_id, data[5:]
