"""Dtype lattice for the pathway_amd framework.

Mirrors the behavior of the reference dtype system
(/root/reference/python/pathway/internals/dtype.py, ~1101 LoC): a small
lattice of column types with Optional wrappers, numpy/python mapping, and
least-upper-bound rules used by expression type inference.

MI355X-native design note: every dtype carries its *device representation*
(`torch_dtype` or None for host object columns) so the engine can decide
which columns live in HBM3E as torch tensors and which stay host-side.
"""

from __future__ import annotations

import datetime
from typing import Any, Hashable

import numpy as np
import pandas as pd

from pathway_amd.internals import datetime_types as _dt_types


class DType:
    """Base class for all pathway_amd column dtypes."""

    _name: str = "DType"
    #: torch dtype name for device-resident columns; None => host object column
    torch_repr: str | None = None

    def __repr__(self) -> str:
        return self._name

    def __eq__(self, other: object) -> bool:
        return type(self) is type(other)

    def __hash__(self) -> int:
        return hash(type(self))

    @property
    def typehint(self) -> Any:
        return Any

    def is_optional(self) -> bool:
        return False

    def is_device_representable(self) -> bool:
        return self.torch_repr is not None


class _Simple(DType):
    def __init__(self, name: str, typehint: Any, torch_repr: str | None):
        self._name = name
        self._typehint = typehint
        self.torch_repr = torch_repr

    @property
    def typehint(self) -> Any:
        return self._typehint

    def __eq__(self, other: object) -> bool:
        return isinstance(other, _Simple) and other._name == self._name

    def __hash__(self) -> int:
        return hash(self._name)


INT = _Simple("INT", int, "int64")
FLOAT = _Simple("FLOAT", float, "float64")
BOOL = _Simple("BOOL", bool, "bool")
STR = _Simple("STR", str, None)  # host object column + device varlen arena
BYTES = _Simple("BYTES", bytes, None)
ANY = _Simple("ANY", Any, None)
NONE = _Simple("NONE", type(None), None)
DATE_TIME_NAIVE = _Simple("DATE_TIME_NAIVE", datetime.datetime, "int64")
DATE_TIME_UTC = _Simple("DATE_TIME_UTC", datetime.datetime, "int64")
DURATION = _Simple("DURATION", datetime.timedelta, "int64")
JSON = _Simple("JSON", Any, None)
PY_OBJECT_WRAPPER = _Simple("PY_OBJECT_WRAPPER", Any, None)
FUTURE = _Simple("FUTURE", Any, None)


class Pointer(DType):
    """128-bit row pointer (the reference's Key, value.rs:40-66)."""

    _name = "POINTER"
    torch_repr = "int64x2"

    def __init__(self, *args: Any):
        self.args = args

    def __eq__(self, other: object) -> bool:
        return isinstance(other, Pointer)

    def __hash__(self) -> int:
        return hash("POINTER")

    @property
    def typehint(self) -> Any:
        from pathway_amd.internals.api import BasePointer

        return BasePointer


POINTER = Pointer()


class Optional(DType):
    def __init__(self, wrapped: DType):
        while isinstance(wrapped, Optional):
            wrapped = wrapped.wrapped
        self.wrapped = wrapped
        self.torch_repr = wrapped.torch_repr

    def __repr__(self) -> str:
        return f"Optional({self.wrapped!r})"

    def __eq__(self, other: object) -> bool:
        return isinstance(other, Optional) and other.wrapped == self.wrapped

    def __hash__(self) -> int:
        return hash(("Optional", self.wrapped))

    @property
    def typehint(self) -> Any:
        import typing

        return typing.Optional[self.wrapped.typehint]

    def is_optional(self) -> bool:
        return True


class List(DType):
    torch_repr = None

    def __init__(self, wrapped: DType = ANY):
        self.wrapped = wrapped

    def __repr__(self) -> str:
        return f"List({self.wrapped!r})"

    def __eq__(self, other: object) -> bool:
        return isinstance(other, List) and other.wrapped == self.wrapped

    def __hash__(self) -> int:
        return hash(("List", self.wrapped))


class Tuple(DType):
    torch_repr = None

    def __init__(self, *args: DType):
        self.args = args

    def __repr__(self) -> str:
        return f"Tuple({', '.join(map(repr, self.args))})"

    def __eq__(self, other: object) -> bool:
        return isinstance(other, Tuple) and other.args == self.args

    def __hash__(self) -> int:
        return hash(("Tuple", self.args))


ANY_TUPLE = List(ANY)


class Array(DType):
    """N-dim numeric array column (reference Value::IntArray/FloatArray)."""

    def __init__(self, n_dim: int | None = None, wrapped: DType = FLOAT):
        self.n_dim = n_dim
        self.wrapped = wrapped
        self.torch_repr = wrapped.torch_repr

    def __repr__(self) -> str:
        return f"Array({self.n_dim}, {self.wrapped!r})"

    def __eq__(self, other: object) -> bool:
        return (
            isinstance(other, Array)
            and other.n_dim == self.n_dim
            and other.wrapped == self.wrapped
        )

    def __hash__(self) -> int:
        return hash(("Array", self.n_dim, self.wrapped))


class Callable(DType):
    torch_repr = None

    def __init__(self, arg_types: Any = ..., return_type: DType = ANY):
        self.arg_types = arg_types
        self.return_type = return_type

    def __repr__(self) -> str:
        return f"Callable(..., {self.return_type!r})"


_PY_TYPE_MAP: dict[Any, DType] = {}


def _init_py_type_map() -> None:
    import typing

    _PY_TYPE_MAP.update(
        {
            int: INT,
            float: FLOAT,
            bool: BOOL,
            str: STR,
            bytes: BYTES,
            type(None): NONE,
            Any: ANY,
            typing.Any: ANY,
            datetime.datetime: DATE_TIME_NAIVE,
            datetime.timedelta: DURATION,
            _dt_types.DateTimeNaive: DATE_TIME_NAIVE,
            _dt_types.DateTimeUtc: DATE_TIME_UTC,
            _dt_types.Duration: DURATION,
            pd.Timestamp: DATE_TIME_NAIVE,
            pd.Timedelta: DURATION,
            np.int64: INT,
            np.int32: INT,
            np.float64: FLOAT,
            np.float32: FLOAT,
            np.bool_: BOOL,
            np.ndarray: Array(),
            list: ANY_TUPLE,
            tuple: ANY_TUPLE,
            dict: JSON,
        }
    )


_init_py_type_map()


def wrap(input_type: Any) -> DType:
    """Convert a python typehint (or DType) into a DType."""
    import typing

    if isinstance(input_type, DType):
        return input_type
    if input_type is None:
        return NONE
    # Handle Optional[X] / Union[X, None]
    origin = typing.get_origin(input_type)
    if origin is typing.Union:
        args = typing.get_args(input_type)
        non_none = [a for a in args if a is not type(None)]
        if len(non_none) == 1 and len(args) == 2:
            return Optional(wrap(non_none[0]))
        return ANY
    if origin in (list,):
        args = typing.get_args(input_type)
        return List(wrap(args[0]) if args else ANY)
    if origin in (tuple,):
        args = typing.get_args(input_type)
        if args and args[-1] is Ellipsis:
            return List(wrap(args[0]))
        return Tuple(*[wrap(a) for a in args]) if args else ANY_TUPLE
    try:
        from pathway_amd.internals.api import BasePointer

        if isinstance(input_type, type) and issubclass(input_type, BasePointer):
            return POINTER
        if origin is not None and isinstance(origin, type) and issubclass(origin, BasePointer):
            return POINTER
    except ImportError:
        pass
    try:
        from pathway_amd.internals.json import Json

        if input_type is Json:
            return JSON
    except ImportError:
        pass
    if isinstance(input_type, Hashable) and input_type in _PY_TYPE_MAP:
        return _PY_TYPE_MAP[input_type]
    # MRO fallback: subclasses of mapped types (e.g. user subclasses of
    # pd.Timestamp) resolve to the mapped base instead of ANY.
    if isinstance(input_type, type):
        for base in input_type.__mro__[1:]:
            if base in _PY_TYPE_MAP:
                return _PY_TYPE_MAP[base]
    return ANY


def unoptionalize(dtype: DType) -> DType:
    return dtype.wrapped if isinstance(dtype, Optional) else dtype


def types_lca(a: DType, b: DType, raising: bool = False) -> DType:
    """Least common ancestor in the dtype lattice (reference dtype.py)."""
    if a == b:
        return a
    if a == NONE:
        return b if b.is_optional() or b in (ANY, NONE) else Optional(b)
    if b == NONE:
        return a if a.is_optional() or a in (ANY, NONE) else Optional(a)
    if isinstance(a, Optional) or isinstance(b, Optional):
        inner = types_lca(unoptionalize(a), unoptionalize(b), raising=raising)
        return Optional(inner) if inner != ANY else ANY
    if {a, b} == {INT, FLOAT}:
        return FLOAT
    if isinstance(a, Pointer) and isinstance(b, Pointer):
        return POINTER
    if raising:
        raise TypeError(f"cannot compute lca of {a!r} and {b!r}")
    return ANY


def dtype_of_value(value: Any) -> DType:
    from pathway_amd.internals.api import BasePointer
    from pathway_amd.internals.json import Json

    if value is None:
        return NONE
    if isinstance(value, BasePointer):
        return POINTER
    if isinstance(value, bool) or isinstance(value, np.bool_):
        return BOOL
    if isinstance(value, (int, np.integer)):
        return INT
    if isinstance(value, (float, np.floating)):
        return FLOAT
    if isinstance(value, str):
        return STR
    if isinstance(value, bytes):
        return BYTES
    if isinstance(value, Json):
        return JSON
    if isinstance(value, np.ndarray):
        return Array(value.ndim, INT if value.dtype.kind == "i" else FLOAT)
    if isinstance(value, (tuple, list)):
        return ANY_TUPLE
    if isinstance(value, datetime.timedelta):
        return DURATION
    if isinstance(value, datetime.datetime):
        return DATE_TIME_UTC if value.tzinfo is not None else DATE_TIME_NAIVE
    if isinstance(value, dict):
        return JSON
    return ANY


def to_numpy_dtype(dtype: DType) -> Any:
    d = unoptionalize(dtype)
    if d == INT:
        return np.int64
    if d == FLOAT:
        return np.float64
    if d == BOOL:
        return np.bool_
    return object
