"""Json value type (reference python/pathway/internals/json.py behavior)."""

from __future__ import annotations

import json as _json
from typing import Any


class Json:
    """Immutable JSON value wrapper, indexable with [] like the reference."""

    __slots__ = ("_value",)

    NULL: "Json"

    def __init__(self, value: Any = None):
        if isinstance(value, Json):
            value = value._value
        self._value = value

    @property
    def value(self) -> Any:
        return self._value

    def dumps(self) -> str:
        return _json.dumps(self._value, sort_keys=True, separators=(",", ":"))

    @staticmethod
    def parse(s: str | bytes) -> "Json":
        return Json(_json.loads(s))

    def __getitem__(self, item: Any) -> "Json":
        return Json(self._value[item])

    def __iter__(self):
        for v in self._value:
            yield Json(v)

    def __len__(self) -> int:
        return len(self._value)

    def __eq__(self, other: object) -> bool:
        if isinstance(other, Json):
            return self._value == other._value
        return self._value == other

    def __hash__(self) -> int:
        return hash(self.dumps())

    def __repr__(self) -> str:
        return f"pw.Json({self._value!r})"

    def __str__(self) -> str:
        return self.dumps()

    # conversion helpers mirroring the reference .as_* API
    def as_int(self) -> int:
        if isinstance(self._value, bool) or not isinstance(self._value, int):
            raise ValueError(f"Json {self!r} is not an int")
        return self._value

    def as_float(self) -> float:
        if isinstance(self._value, bool) or not isinstance(self._value, (int, float)):
            raise ValueError(f"Json {self!r} is not a float")
        return float(self._value)

    def as_str(self) -> str:
        if not isinstance(self._value, str):
            raise ValueError(f"Json {self!r} is not a str")
        return self._value

    def as_bool(self) -> bool:
        if not isinstance(self._value, bool):
            raise ValueError(f"Json {self!r} is not a bool")
        return self._value

    def as_list(self) -> list:
        if not isinstance(self._value, list):
            raise ValueError(f"Json {self!r} is not a list")
        return self._value

    def as_dict(self) -> dict:
        if not isinstance(self._value, dict):
            raise ValueError(f"Json {self!r} is not a dict")
        return self._value


Json.NULL = Json(None)
