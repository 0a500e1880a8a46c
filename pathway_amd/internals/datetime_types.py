"""DateTimeNaive / DateTimeUtc / Duration (reference src/engine/time.rs).

Nanosecond-precision types backed by pandas Timestamp/Timedelta (the
reference's Python API also surfaces pandas-compatible types); on device
they are stored as int64 nanoseconds.
"""

from __future__ import annotations

from typing import Any

import pandas as pd


class DateTimeNaive(pd.Timestamp):
    def __new__(cls, *args: Any, **kwargs: Any):
        obj = pd.Timestamp.__new__(cls, *args, **kwargs)
        if obj.tzinfo is not None:
            raise ValueError("DateTimeNaive cannot have a timezone")
        return obj

    @staticmethod
    def from_ns(ns: int) -> "DateTimeNaive":
        return DateTimeNaive(pd.Timestamp(ns, unit="ns"))


class DateTimeUtc(pd.Timestamp):
    def __new__(cls, *args: Any, **kwargs: Any):
        obj = pd.Timestamp.__new__(cls, *args, **kwargs)
        if obj.tzinfo is None:
            raise ValueError("DateTimeUtc must have a timezone")
        return obj

    @staticmethod
    def from_ns(ns: int) -> "DateTimeUtc":
        return DateTimeUtc(pd.Timestamp(ns, unit="ns", tz="UTC"))


class Duration(pd.Timedelta):
    def __new__(cls, *args: Any, **kwargs: Any):
        return pd.Timedelta.__new__(cls, *args, **kwargs)

    @staticmethod
    def from_ns(ns: int) -> "Duration":
        return Duration(pd.Timedelta(ns, unit="ns"))


def to_ns(value: Any) -> int:
    """Canonical int64-ns representation of a datetime-like value."""
    if isinstance(value, pd.Timestamp):
        return int(value.value)
    if isinstance(value, pd.Timedelta):
        return int(value.value)
    import datetime

    if isinstance(value, datetime.timedelta):
        # Exact conversion through pandas — float seconds would round
        # ns-precision values.
        return int(pd.Timedelta(value).value)
    if isinstance(value, datetime.datetime):
        return int(pd.Timestamp(value).value)
    raise TypeError(f"not a datetime-like value: {value!r}")
