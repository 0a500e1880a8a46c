"""expr.str namespace (reference internals/expressions/string.py, 931 LoC)."""

from __future__ import annotations

from typing import Any

from pathway_amd.internals import dtype as dt
from pathway_amd.internals.expression import ColumnExpression, MethodCallExpression


class StringNamespace:
    def __init__(self, expr: ColumnExpression):
        self._expr = expr

    def _m(self, name: str, *args: Any, return_type=None) -> MethodCallExpression:
        return MethodCallExpression(f"str.{name}", self._expr, *args, return_type=return_type)

    def lower(self):
        return self._m("lower", return_type=dt.STR)

    def upper(self):
        return self._m("upper", return_type=dt.STR)

    def reversed(self):
        return self._m("reversed", return_type=dt.STR)

    def len(self):
        return self._m("len", return_type=dt.INT)

    def strip(self, chars: Any = None):
        return self._m("strip", *( [chars] if chars is not None else [] ), return_type=dt.STR)

    def lstrip(self, chars: Any = None):
        return self._m("lstrip", *([chars] if chars is not None else []), return_type=dt.STR)

    def rstrip(self, chars: Any = None):
        return self._m("rstrip", *([chars] if chars is not None else []), return_type=dt.STR)

    def startswith(self, prefix: Any):
        return self._m("startswith", prefix, return_type=dt.BOOL)

    def endswith(self, suffix: Any):
        return self._m("endswith", suffix, return_type=dt.BOOL)

    def count(self, sub: Any):
        return self._m("count", sub, return_type=dt.INT)

    def find(self, sub: Any, start: Any = None, end: Any = None):
        args = [sub] + [a for a in (start, end) if a is not None]
        return self._m("find", *args, return_type=dt.INT)

    def rfind(self, sub: Any, start: Any = None, end: Any = None):
        args = [sub] + [a for a in (start, end) if a is not None]
        return self._m("rfind", *args, return_type=dt.INT)

    def replace(self, old: Any, new: Any, count: Any = None):
        args = [old, new] + ([count] if count is not None else [])
        return self._m("replace", *args, return_type=dt.STR)

    def split(self, sep: Any = None, maxsplit: Any = None):
        args = [a for a in (sep,) if a is not None]
        return self._m("split", *args, return_type=dt.ANY_TUPLE)

    def slice(self, start: Any, end: Any):
        return self._m("slice", start, end, return_type=dt.STR)

    def title(self):
        return self._m("title", return_type=dt.STR)

    def swap_case(self):
        return self._m("swapcase", return_type=dt.STR)

    # reference name (string.py swapcase); swap_case kept as an alias
    swapcase = swap_case

    def ljust(self, width: Any, fillchar: Any = None):
        args = [width] + ([fillchar] if fillchar is not None else [])
        return self._m("ljust", *args, return_type=dt.STR)

    def rjust(self, width: Any, fillchar: Any = None):
        args = [width] + ([fillchar] if fillchar is not None else [])
        return self._m("rjust", *args, return_type=dt.STR)

    def removeprefix(self, prefix: Any):
        return self._m("removeprefix", prefix, return_type=dt.STR)

    def removesuffix(self, suffix: Any):
        return self._m("removesuffix", suffix, return_type=dt.STR)

    def parse_int(self, optional: bool = False):
        return self._m("parse_int", return_type=dt.Optional(dt.INT) if optional else dt.INT)

    def parse_float(self, optional: bool = False):
        return self._m("parse_float", return_type=dt.Optional(dt.FLOAT) if optional else dt.FLOAT)

    def parse_bool(self, optional: bool = False):
        return self._m("parse_bool", return_type=dt.Optional(dt.BOOL) if optional else dt.BOOL)

    def to_bytes(self, encoding: str = "utf-8"):
        return self._m("to_bytes", encoding, return_type=dt.BYTES)


class BinaryNamespace:
    def __init__(self, expr: ColumnExpression):
        self._expr = expr

    def decode(self, encoding: str = "utf-8"):
        return MethodCallExpression("bin.decode", self._expr, encoding, return_type=dt.STR)

    def base64_encode(self):
        return MethodCallExpression("bin.base64_encode", self._expr, return_type=dt.STR)

    def base64_decode(self):
        return MethodCallExpression("bin.base64_decode", self._expr, return_type=dt.BYTES)
