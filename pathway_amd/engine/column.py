"""Columnar value storage for the engine.

Columns are SoA device buffers (torch tensors) wherever the dtype allows —
int/float/bool/datetime/pointer columns live in HBM3E; strings are
dictionary-encoded (device int64 codes + host pool with precomputed 128-bit
value hashes, mirrored on device) so the hot path never touches host
strings; arbitrary python objects fall back to host object arrays.

Reference semantics: Value enum of value.rs:208-232; the device layouts are
MI355X-native (columnar, hash-precomputed) rather than the reference's
32-byte tagged enum.
"""

from __future__ import annotations

from typing import Any, Sequence

import numpy as np
import torch

from pathway_amd.internals import dtype as dt
from pathway_amd.internals.api import (
    ERROR,
    BasePointer,
    Pointer,
    hash128,
    serialize_value,
)
from pathway_amd.engine import hashing

_MASK64 = (1 << 64) - 1


def _signed(x: int) -> int:
    x &= _MASK64
    return x - (1 << 64) if x >= (1 << 63) else x


class StringPool:
    """Process-wide dictionary encoding of strings (code -> str).

    Codes are worker-local; cross-worker exchange always ships the string
    bytes (see parallel/exchange.py), so pools never need to agree.
    Precomputes the canonical 128-bit value hash per pool entry; group-key
    hashing of a string column is then a device-side gather.
    """

    def __init__(self) -> None:
        self._strs: list[str] = []
        self._index: dict[str, int] = {}
        self._hash_lo: list[int] = []
        self._hash_hi: list[int] = []
        self._device_hashes: dict[Any, tuple[torch.Tensor, torch.Tensor, int]] = {}
        # True when every worker is known to hold an identical pool (codes
        # agree across ranks) — lets the exchange ship codes instead of bytes
        self.synchronized = False

    def __len__(self) -> int:
        return len(self._strs)

    def code(self, s: str) -> int:
        c = self._index.get(s)
        if c is None:
            c = len(self._strs)
            self._index[s] = c
            self._strs.append(s)
            lo, hi = hash128(serialize_value(s))
            self._hash_lo.append(_signed(lo))
            self._hash_hi.append(_signed(hi))
        return c

    def codes(self, values: Sequence[str]) -> np.ndarray:
        idx = self._index
        out = np.empty(len(values), dtype=np.int64)
        for i, v in enumerate(values):
            if v is None:
                out[i] = -1
            else:
                c = idx.get(v)
                if c is None:
                    c = self.code(v)
                out[i] = c
        return out

    def value(self, code: int) -> str | None:
        return None if code < 0 else self._strs[code]

    def extend_from(self, strings: Sequence[str]) -> np.ndarray:
        return self.codes(strings)

    def hash_tensors(self, device) -> tuple[torch.Tensor, torch.Tensor]:
        """(pool_size,) int64 lo/hi hash tensors on `device`, cached."""
        key = str(device)
        cached = self._device_hashes.get(key)
        n = len(self._strs)
        if cached is not None and cached[2] == n:
            return cached[0], cached[1]
        lo = torch.tensor(self._hash_lo, dtype=torch.int64, device=device)
        hi = torch.tensor(self._hash_hi, dtype=torch.int64, device=device)
        self._device_hashes[key] = (lo, hi, n)
        return lo, hi


GLOBAL_STRING_POOL = StringPool()


def obj_array(values) -> np.ndarray:
    """Object ndarray built element-wise — np.array() would flatten nested
    sequences of equal length into a 2-D array."""
    arr = np.empty(len(values), dtype=object)
    for i, v in enumerate(values):
        arr[i] = v
    return arr


class Column:
    dtype: dt.DType

    def __len__(self) -> int:
        raise NotImplementedError

    def take(self, idx: torch.Tensor) -> "Column":
        raise NotImplementedError

    def to_pylist(self) -> list[Any]:
        raise NotImplementedError

    def value_hash(self) -> tuple[torch.Tensor, torch.Tensor]:
        """Per-row canonical 128-bit value hash, as device int64 tensors."""
        raise NotImplementedError

    def to_device(self, device) -> "Column":
        raise NotImplementedError

    def slice(self, start: int, stop: int) -> "Column":
        n = len(self)
        idx = torch.arange(start, min(stop, n), dtype=torch.int64, device=self._device())
        return self.take(idx)

    def _device(self):
        return torch.device("cpu")


_KIND_BY_DTYPE = {
    dt.INT: "int",
    dt.FLOAT: "float",
    dt.BOOL: "bool",
    dt.DATE_TIME_NAIVE: "datetime_naive",
    dt.DATE_TIME_UTC: "datetime_utc",
    dt.DURATION: "duration",
}

_TORCH_DTYPE = {
    dt.INT: torch.int64,
    dt.FLOAT: torch.float64,
    dt.BOOL: torch.bool,
    dt.DATE_TIME_NAIVE: torch.int64,
    dt.DATE_TIME_UTC: torch.int64,
    dt.DURATION: torch.int64,
}


class TensorColumn(Column):
    """Numeric/bool/datetime column; optional validity mask for Optional dtypes."""

    def __init__(
        self,
        tensor: torch.Tensor,
        dtype: dt.DType,
        mask: torch.Tensor | None = None,
    ):
        self.tensor = tensor
        self.dtype = dtype
        self.mask = mask  # bool tensor, True = valid; None = all valid

    def __len__(self) -> int:
        return int(self.tensor.shape[0])

    def _device(self):
        return self.tensor.device

    def take(self, idx: torch.Tensor) -> "TensorColumn":
        return TensorColumn(
            self.tensor.index_select(0, idx),
            self.dtype,
            self.mask.index_select(0, idx) if self.mask is not None else None,
        )

    def to_pylist(self) -> list[Any]:
        vals = self.tensor.cpu().tolist()
        base = dt.unoptionalize(self.dtype)
        if base in (dt.DATE_TIME_NAIVE, dt.DATE_TIME_UTC, dt.DURATION):
            from pathway_amd.internals import datetime_types as dtt

            conv = {
                dt.DATE_TIME_NAIVE: dtt.DateTimeNaive.from_ns,
                dt.DATE_TIME_UTC: dtt.DateTimeUtc.from_ns,
                dt.DURATION: dtt.Duration.from_ns,
            }[base]
            vals = [conv(v) for v in vals]
        if self.mask is not None:
            m = self.mask.cpu().tolist()
            vals = [v if ok else None for v, ok in zip(vals, m)]
        return vals

    def value_hash(self) -> tuple[torch.Tensor, torch.Tensor]:
        base = dt.unoptionalize(self.dtype)
        kind = _KIND_BY_DTYPE.get(base, "int")
        t = self.tensor
        if t.dim() > 1:
            raise NotImplementedError("hashing of array columns by rows")
        lo, hi = hashing.column_value_hash(t, kind)
        if self.mask is not None:
            nlo, nhi = hashing.none_value_hash(len(self), t.device)
            lo = torch.where(self.mask, lo, nlo)
            hi = torch.where(self.mask, hi, nhi)
        return lo, hi

    def to_device(self, device) -> "TensorColumn":
        return TensorColumn(
            self.tensor.to(device),
            self.dtype,
            self.mask.to(device) if self.mask is not None else None,
        )

    @staticmethod
    def concat(cols: list["TensorColumn"]) -> "TensorColumn":
        dtype = cols[0].dtype
        t = torch.cat([c.tensor for c in cols])
        if any(c.mask is not None for c in cols):
            masks = [
                c.mask
                if c.mask is not None
                else torch.ones(len(c), dtype=torch.bool, device=c.tensor.device)
                for c in cols
            ]
            m = torch.cat(masks)
        else:
            m = None
        return TensorColumn(t, dtype, m)


class PointerColumn(Column):
    """128-bit pointer column stored as (n, 2) int64."""

    def __init__(self, pairs: torch.Tensor, dtype: dt.DType = dt.POINTER):
        assert pairs.dim() == 2 and pairs.shape[1] == 2
        self.pairs = pairs
        self.dtype = dtype

    def __len__(self) -> int:
        return int(self.pairs.shape[0])

    def _device(self):
        return self.pairs.device

    def take(self, idx: torch.Tensor) -> "PointerColumn":
        return PointerColumn(self.pairs.index_select(0, idx), self.dtype)

    def to_pylist(self) -> list[Any]:
        out = []
        for lo, hi in self.pairs.cpu().tolist():
            out.append(BasePointer.from_signed_pair(lo, hi))
        return out

    def value_hash(self) -> tuple[torch.Tensor, torch.Tensor]:
        return hashing.pointer_value_hash(self.pairs)

    def to_device(self, device) -> "PointerColumn":
        return PointerColumn(self.pairs.to(device), self.dtype)

    @staticmethod
    def concat(cols: list["PointerColumn"]) -> "PointerColumn":
        return PointerColumn(torch.cat([c.pairs for c in cols]), cols[0].dtype)


class StringColumn(Column):
    """Dictionary-encoded string column; codes live on device."""

    def __init__(
        self,
        codes: torch.Tensor,
        pool: StringPool | None = None,
        dtype: dt.DType = dt.STR,
    ):
        self.codes = codes
        self.pool = pool or GLOBAL_STRING_POOL
        self.dtype = dtype

    def __len__(self) -> int:
        return int(self.codes.shape[0])

    def _device(self):
        return self.codes.device

    def take(self, idx: torch.Tensor) -> "StringColumn":
        return StringColumn(self.codes.index_select(0, idx), self.pool, self.dtype)

    def to_pylist(self) -> list[Any]:
        pool = self.pool
        return [pool.value(c) for c in self.codes.cpu().tolist()]

    def value_hash(self) -> tuple[torch.Tensor, torch.Tensor]:
        lo_t, hi_t = self.pool.hash_tensors(self.codes.device)
        codes = self.codes
        nlo, nhi = hashing.none_value_hash_scalar(codes.device)
        if codes.is_cuda:
            from pathway_amd import ops

            return ops.pool_hash_gpu(codes, lo_t, hi_t, int(nlo), int(nhi))
        valid = codes >= 0
        safe = codes.clamp_min(0)
        lo = lo_t.index_select(0, safe)
        hi = hi_t.index_select(0, safe)
        # unconditional None-blend: avoids a device→host .all() sync per call
        lo = torch.where(valid, lo, nlo)
        hi = torch.where(valid, hi, nhi)
        return lo, hi

    def to_device(self, device) -> "StringColumn":
        return StringColumn(self.codes.to(device), self.pool, self.dtype)

    @staticmethod
    def from_strings(values: Sequence[str | None], device="cpu") -> "StringColumn":
        codes = GLOBAL_STRING_POOL.codes(list(values))
        return StringColumn(torch.from_numpy(codes).to(device))

    @staticmethod
    def concat(cols: list["StringColumn"]) -> "StringColumn":
        return StringColumn(
            torch.cat([c.codes for c in cols]), cols[0].pool, cols[0].dtype
        )


class ObjectColumn(Column):
    """Host object array column (tuples, Json, arbitrary python values)."""

    def __init__(self, values: np.ndarray, dtype: dt.DType = dt.ANY):
        if not isinstance(values, np.ndarray):
            arr = np.empty(len(values), dtype=object)
            for i, v in enumerate(values):
                arr[i] = v
            values = arr
        self.values = values
        self.dtype = dtype

    def __len__(self) -> int:
        return int(self.values.shape[0])

    def take(self, idx: torch.Tensor) -> "ObjectColumn":
        return ObjectColumn(self.values[idx.cpu().numpy()], self.dtype)

    def to_pylist(self) -> list[Any]:
        return list(self.values)

    def value_hash(self) -> tuple[torch.Tensor, torch.Tensor]:
        n = len(self)
        lo = np.empty(n, dtype=np.int64)
        hi = np.empty(n, dtype=np.int64)
        for i, v in enumerate(self.values):
            l, h = hash128(serialize_value(v))
            lo[i] = _signed(l)
            hi[i] = _signed(h)
        return torch.from_numpy(lo), torch.from_numpy(hi)

    def to_device(self, device) -> "ObjectColumn":
        return self  # host column stays host-side

    @staticmethod
    def concat(cols: list["ObjectColumn"]) -> "ObjectColumn":
        return ObjectColumn(np.concatenate([c.values for c in cols]), cols[0].dtype)


def concat_columns(cols: list[Column]) -> Column:
    head = cols[0]
    if len(cols) == 1:
        return head
    same = all(type(c) is type(head) for c in cols)
    if same and isinstance(head, TensorColumn):
        if all(c.tensor.dtype == head.tensor.dtype for c in cols):  # type: ignore[attr-defined]
            return TensorColumn.concat(cols)  # type: ignore[arg-type]
    if same and isinstance(head, PointerColumn):
        return PointerColumn.concat(cols)  # type: ignore[arg-type]
    if same and isinstance(head, StringColumn) and all(
        c.pool is head.pool for c in cols  # type: ignore[attr-defined]
    ):
        return StringColumn.concat(cols)  # type: ignore[arg-type]
    return ObjectColumn.concat([as_object_column(c) for c in cols])


def as_object_column(col: Column) -> ObjectColumn:
    if isinstance(col, ObjectColumn):
        return col
    arr = np.empty(len(col), dtype=object)
    for i, v in enumerate(col.to_pylist()):
        arr[i] = v
    return ObjectColumn(arr, col.dtype)


def column_from_pylist(
    values: Sequence[Any], dtype: dt.DType, device="cpu"
) -> Column:
    """Build the best-fitting column for python values of declared dtype."""
    base = dt.unoptionalize(dtype)
    n = len(values)
    has_none = any(v is None for v in values)
    has_error = any(v is ERROR for v in values)
    if has_error:
        arr = np.empty(n, dtype=object)
        for i, v in enumerate(values):
            arr[i] = v
        return ObjectColumn(arr, dtype)
    if base in (dt.INT, dt.BOOL, dt.FLOAT):
        td = _TORCH_DTYPE[base]
        if has_none:
            filled = [v if v is not None else 0 for v in values]
            t = torch.tensor(filled, dtype=td, device=device)
            mask = torch.tensor(
                [v is not None for v in values], dtype=torch.bool, device=device
            )
            return TensorColumn(t, dtype, mask)
        t = torch.tensor(list(values), dtype=td, device=device)
        return TensorColumn(t, dtype)
    if base in (dt.DATE_TIME_NAIVE, dt.DATE_TIME_UTC, dt.DURATION):
        from pathway_amd.internals import datetime_types as dtt

        ns = [dtt.to_ns(v) if v is not None else 0 for v in values]
        t = torch.tensor(ns, dtype=torch.int64, device=device)
        mask = (
            torch.tensor([v is not None for v in values], dtype=torch.bool, device=device)
            if has_none
            else None
        )
        return TensorColumn(t, dtype, mask)
    if base == dt.STR:
        return StringColumn.from_strings(list(values), device=device)
    if isinstance(base, dt.Pointer):
        if has_none:
            arr = np.empty(n, dtype=object)
            for i, v in enumerate(values):
                arr[i] = v
            return ObjectColumn(arr, dtype)
        pairs = torch.tensor(
            [list(v.as_signed_pair()) for v in values],
            dtype=torch.int64,
            device=device,
        ).reshape(n, 2)
        return PointerColumn(pairs, dtype)
    arr = np.empty(n, dtype=object)
    for i, v in enumerate(values):
        arr[i] = v
    return ObjectColumn(arr, dtype)


def infer_and_build_column(values: Sequence[Any], device="cpu") -> tuple[Column, dt.DType]:
    """Infer dtype from python values then build the column."""
    kinds = {dt.dtype_of_value(v) for v in values if v is not None}
    has_none = any(v is None for v in values)
    if len(kinds) == 0:
        dtype: dt.DType = dt.NONE
    elif len(kinds) == 1:
        dtype = next(iter(kinds))
    elif kinds == {dt.INT, dt.FLOAT}:
        dtype = dt.FLOAT
    else:
        dtype = dt.ANY
    if has_none and dtype not in (dt.NONE, dt.ANY):
        dtype = dt.Optional(dtype)
    if dt.unoptionalize(dtype) == dt.FLOAT:
        values = [float(v) if v is not None else None for v in values]
    return column_from_pylist(values, dtype, device), dtype
