"""Value model: 128-bit keys (pointers), host-side hashing, DataRow capture.

Mirrors the reference key/value model (/root/reference/src/engine/value.rs:38-118):
  * Key = 128 bits = hash of the row's defining values
  * shard = low 16 bits of the first key word (SHARD_MASK)
  * derived keys for reindex/join/concat outputs use per-operation salts

The hash is xxhash64 evaluated at two seeds over a canonical tagged byte
serialization of the value.  The identical algorithm is implemented in
C++/HIP (pathway_amd/ops/csrc/xxhash.h) so host-built static tables and the
device hash kernel produce identical keys; tests/test_hash.py checks the
Python and C++ host paths agree, and the gpu-marked test checks the device
kernel agrees too.
"""

from __future__ import annotations

import struct
from dataclasses import dataclass
from typing import Any, Iterable

import numpy as np

MASK64 = (1 << 64) - 1
SHARD_BITS = 16
SHARD_MASK = (1 << SHARD_BITS) - 1

_PRIME1 = 0x9E3779B185EBCA87
_PRIME2 = 0xC2B2AE3D27D4EB4F
_PRIME3 = 0x165667B19E3779F9
_PRIME4 = 0x85EBCA77C2B2AE63
_PRIME5 = 0x27D4EB2F165667C5

SEED_LO = 0
SEED_HI = 0x9E3779B185EBCA87


def _rotl(x: int, r: int) -> int:
    return ((x << r) | (x >> (64 - r))) & MASK64


def _round(acc: int, inp: int) -> int:
    acc = (acc + inp * _PRIME2) & MASK64
    acc = _rotl(acc, 31)
    return (acc * _PRIME1) & MASK64


def _merge_round(acc: int, val: int) -> int:
    val = _round(0, val)
    acc ^= val
    return (acc * _PRIME1 + _PRIME4) & MASK64


def xxh64(data: bytes, seed: int = 0) -> int:
    """Reference xxhash64 — bit-exact with the C++/HIP implementation."""
    n = len(data)
    i = 0
    if n >= 32:
        v1 = (seed + _PRIME1 + _PRIME2) & MASK64
        v2 = (seed + _PRIME2) & MASK64
        v3 = seed & MASK64
        v4 = (seed - _PRIME1) & MASK64
        while i + 32 <= n:
            (a, b, c, d) = struct.unpack_from("<QQQQ", data, i)
            v1 = _round(v1, a)
            v2 = _round(v2, b)
            v3 = _round(v3, c)
            v4 = _round(v4, d)
            i += 32
        h = (_rotl(v1, 1) + _rotl(v2, 7) + _rotl(v3, 12) + _rotl(v4, 18)) & MASK64
        h = _merge_round(h, v1)
        h = _merge_round(h, v2)
        h = _merge_round(h, v3)
        h = _merge_round(h, v4)
    else:
        h = (seed + _PRIME5) & MASK64
    h = (h + n) & MASK64
    while i + 8 <= n:
        (k,) = struct.unpack_from("<Q", data, i)
        h ^= _round(0, k)
        h = (_rotl(h, 27) * _PRIME1 + _PRIME4) & MASK64
        i += 8
    while i + 4 <= n:
        (k,) = struct.unpack_from("<I", data, i)
        h ^= (k * _PRIME1) & MASK64
        h = (_rotl(h, 23) * _PRIME2 + _PRIME3) & MASK64
        i += 4
    while i < n:
        h ^= (data[i] * _PRIME5) & MASK64
        h = (_rotl(h, 11) * _PRIME1) & MASK64
        i += 1
    h ^= h >> 33
    h = (h * _PRIME2) & MASK64
    h ^= h >> 29
    h = (h * _PRIME3) & MASK64
    h ^= h >> 32
    return h


def hash128(data: bytes) -> tuple[int, int]:
    """128-bit hash: xxh64 at two fixed seeds (lo, hi)."""
    return xxh64(data, SEED_LO), xxh64(data, SEED_HI)


# --- canonical value serialization (tags shared with csrc/xxhash.h) ---
TAG_NONE = 0
TAG_BOOL = 1
TAG_INT = 2
TAG_FLOAT = 3
TAG_POINTER = 4
TAG_STR = 5
TAG_BYTES = 6
TAG_TUPLE = 7
TAG_DT_NAIVE = 8
TAG_DT_UTC = 9
TAG_DURATION = 10
TAG_JSON = 11
TAG_ARRAY = 12
TAG_PYOBJ = 13
TAG_ERROR = 14
TAG_PENDING = 15


def serialize_value(value: Any) -> bytes:
    """Canonical tagged bytes of a value, used for key hashing.

    The tag is emitted as a full little-endian u64 word so that fixed-width
    values serialize to whole 8-byte words — this keeps the vectorized torch
    hash and the HIP device hash kernel trivially word-aligned.
    """
    from pathway_amd.internals.json import Json

    def w(tag: int) -> bytes:
        return struct.pack("<Q", tag)

    if value is None:
        return w(TAG_NONE)
    if value is ERROR:
        return w(TAG_ERROR)
    if value is PENDING:
        return w(TAG_PENDING)
    if isinstance(value, BasePointer):
        return w(TAG_POINTER) + struct.pack("<QQ", value.lo, value.hi)
    if isinstance(value, (bool, np.bool_)):
        return w(TAG_BOOL) + struct.pack("<Q", 1 if value else 0)
    if isinstance(value, (int, np.integer)):
        return w(TAG_INT) + struct.pack("<q", int(value))
    if isinstance(value, (float, np.floating)):
        f = float(value)
        # Normalize -0.0 -> 0.0 and NaN -> one canonical bit pattern so
        # values that compare equal hash equal (reference value.rs:614-625).
        if f == 0.0:
            f = 0.0
        elif f != f:
            return w(TAG_FLOAT) + struct.pack("<Q", 0x7FF8000000000000)
        return w(TAG_FLOAT) + struct.pack("<d", f)
    if isinstance(value, str):
        return w(TAG_STR) + value.encode("utf-8")
    if isinstance(value, bytes):
        return w(TAG_BYTES) + value
    if isinstance(value, Json):
        return w(TAG_JSON) + value.dumps().encode("utf-8")
    if isinstance(value, np.ndarray):
        # Shape + dtype kind are part of the key so that (2,2) vs (4,) or
        # int64 vs float64 arrays with identical raw bytes do not collide
        # (reference value.rs ArrayD hash_into hashes shape then elements).
        header = struct.pack("<QQ", ord(value.dtype.kind), value.ndim)
        header += b"".join(struct.pack("<Q", d) for d in value.shape)
        return w(TAG_ARRAY) + header + value.tobytes()
    if isinstance(value, (tuple, list)):
        out = bytearray(w(TAG_TUPLE))
        for v in value:
            lo, hi = hash128(serialize_value(v))
            out += struct.pack("<QQ", lo, hi)
        return bytes(out)
    import datetime

    if isinstance(value, datetime.timedelta):
        # Exact integer-ns serialization (never float seconds): the device
        # TensorColumn path hashes exact int64 ns, and the two must agree
        # for pointer identity / joins on ns-precision values.
        import pandas as pd

        return w(TAG_DURATION) + struct.pack("<q", int(pd.Timedelta(value).value))
    if isinstance(value, datetime.datetime):
        import pandas as pd

        tag = TAG_DT_UTC if value.tzinfo is not None else TAG_DT_NAIVE
        return w(tag) + struct.pack("<q", int(pd.Timestamp(value).value))
    # arbitrary python object: hash of repr as last resort
    return w(TAG_PYOBJ) + repr(value).encode("utf-8")


def hash_values(values: Iterable[Any]) -> tuple[int, int]:
    """Key for a row of values: hash of concatenated per-value 128-bit hashes.

    Reference: values_to_key / ShardPolicy.generate_key (value.rs:96-118).
    """
    buf = bytearray()
    for v in values:
        lo, hi = hash128(serialize_value(v))
        buf += struct.pack("<QQ", lo, hi)
    return hash128(bytes(buf))


def derive_key(salt: int, parts: Iterable[tuple[int, int]]) -> tuple[int, int]:
    """Derived key for reindex/join/concat outputs: salted hash of input keys."""
    buf = bytearray(struct.pack("<Q", salt & MASK64))
    for lo, hi in parts:
        buf += struct.pack("<QQ", lo & MASK64, hi & MASK64)
    return hash128(bytes(buf))


def _to_signed(x: int) -> int:
    x &= MASK64
    return x - (1 << 64) if x >= (1 << 63) else x


def _to_unsigned(x: int) -> int:
    return x & MASK64


class BasePointer:
    """128-bit row pointer (reference Key)."""

    __slots__ = ("lo", "hi")

    def __init__(self, lo: int, hi: int):
        self.lo = _to_unsigned(lo)
        self.hi = _to_unsigned(hi)

    def __eq__(self, other: object) -> bool:
        return (
            isinstance(other, BasePointer)
            and self.lo == other.lo
            and self.hi == other.hi
        )

    def __lt__(self, other: "BasePointer") -> bool:
        return (self.hi, self.lo) < (other.hi, other.lo)

    def __le__(self, other: "BasePointer") -> bool:
        return (self.hi, self.lo) <= (other.hi, other.lo)

    def __gt__(self, other: "BasePointer") -> bool:
        return (self.hi, self.lo) > (other.hi, other.lo)

    def __ge__(self, other: "BasePointer") -> bool:
        return (self.hi, self.lo) >= (other.hi, other.lo)

    def __hash__(self) -> int:
        return hash((self.lo, self.hi))

    def __repr__(self) -> str:
        return f"^{(self.hi << 64 | self.lo):032X}"

    @property
    def shard(self) -> int:
        return self.lo & SHARD_MASK

    def as_signed_pair(self) -> tuple[int, int]:
        return _to_signed(self.lo), _to_signed(self.hi)

    @staticmethod
    def from_signed_pair(lo: int, hi: int) -> "BasePointer":
        return Pointer(_to_unsigned(lo), _to_unsigned(hi))


class Pointer(BasePointer):
    """Public pointer type; generic subscript (Pointer[int]) is accepted."""

    def __class_getitem__(cls, item: Any) -> Any:
        return cls


def ref_scalar(*args: Any, optional: bool = False) -> Pointer:
    """pw.Table.pointer_from equivalent for scalar python values."""
    if optional and any(a is None for a in args):
        return None  # type: ignore[return-value]
    lo, hi = hash_values(args)
    return Pointer(lo, hi)


@dataclass
class DataRow:
    """Captured update-stream row (reference internals/api.py:40)."""

    key: BasePointer
    values: list[Any]
    time: int = 0
    diff: int = 1
    shard: int | None = None

    def __iter__(self):
        yield from (self.key, self.values, self.time, self.diff)


def squash_updates(rows: list[DataRow], *, terminate_on_error: bool = True) -> dict:
    """Fold an update stream into final state (reference internals/api.py:197-226)."""
    state: dict[BasePointer, list[Any]] = {}
    rows = sorted(rows, key=lambda r: (r.time, r.diff))
    for row in rows:
        if row.diff == 1:
            if row.key in state and terminate_on_error:
                raise KeyError(f"duplicate insert for key {row.key}")
            state[row.key] = row.values
        elif row.diff == -1:
            if row.key not in state and terminate_on_error:
                raise KeyError(f"delete of missing key {row.key}")
            state.pop(row.key, None)
        else:
            raise ValueError(f"unexpected diff {row.diff}")
    return state


class PyObjectWrapper:
    """Opaque wrapper for arbitrary python objects stored in columns."""

    __slots__ = ("value",)

    def __init__(self, value: Any):
        self.value = value

    def __eq__(self, other: object) -> bool:
        return isinstance(other, PyObjectWrapper) and self.value == other.value

    def __hash__(self) -> int:
        return hash(("PyObjectWrapper", id(type(self.value))))

    def __repr__(self) -> str:
        return f"PyObjectWrapper({self.value!r})"


def wrap_py_object(value: Any) -> PyObjectWrapper:
    return PyObjectWrapper(value)


class _ErrorValue:
    """Singleton Value::Error equivalent — propagates through expressions."""

    _instance = None

    def __new__(cls):
        if cls._instance is None:
            cls._instance = super().__new__(cls)
        return cls._instance

    def __repr__(self) -> str:
        return "Error"

    def __bool__(self) -> bool:
        raise ValueError("cannot convert Error value to bool")


ERROR = _ErrorValue()


class _PendingValue:
    _instance = None

    def __new__(cls):
        if cls._instance is None:
            cls._instance = super().__new__(cls)
        return cls._instance

    def __repr__(self) -> str:
        return "Pending"


PENDING = _PendingValue()


def unsafe_make_pointer(value) -> "BasePointer":
    """Pointer directly from a trusted integer (reference api
    unsafe_make_pointer) — no hashing; collision safety is the caller's
    responsibility."""
    v = int(value)
    return Pointer(v & MASK64, 0)
