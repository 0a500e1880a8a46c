"""Temporal type helpers (reference stdlib/temporal/utils.py behavior)."""
from __future__ import annotations

import datetime
from typing import Any

from pathway_amd.internals import dtype as dt

TimeEventType = Any
IntervalType = Any


def get_default_origin(time_event_type):
    """Zero point of a time-event type (reference utils.py:16)."""
    d = dt.wrap(time_event_type) if not isinstance(time_event_type, dt.DType) else time_event_type
    d = dt.unoptionalize(d)
    if d == dt.INT:
        return 0
    if d == dt.FLOAT:
        return 0.0
    if d in (dt.DATE_TIME_NAIVE, dt.DATE_TIME_UTC):
        import pandas as pd

        return pd.Timestamp(0)
    raise TypeError(f"no default origin for {time_event_type!r}")


def zero_length_interval(interval_type):
    """Zero of an interval/duration type (reference utils.py:27)."""
    if interval_type in (int,):
        return 0
    if interval_type in (float,):
        return 0.0
    if interval_type is datetime.timedelta:
        return datetime.timedelta(0)
    import pandas as pd

    if interval_type is pd.Timedelta:
        return pd.Timedelta(0)
    raise TypeError(f"no zero interval for {interval_type!r}")


def check_joint_types(parameters: dict) -> None:
    """Validate that time/interval parameter dtypes are mutually
    compatible (reference utils.py:46) — int/float with numeric intervals,
    datetimes with durations."""
    groups = {
        "numeric": {dt.INT, dt.FLOAT, dt.DURATION},
        "datetime": {dt.DATE_TIME_NAIVE, dt.DATE_TIME_UTC, dt.DURATION},
    }
    seen = []
    for name, (value_dtype, _expected) in parameters.items():
        d = dt.unoptionalize(
            value_dtype if isinstance(value_dtype, dt.DType) else dt.wrap(value_dtype)
        )
        seen.append((name, d))
    for family in groups.values():
        if all(d in family or d == dt.ANY for _, d in seen):
            return
    raise TypeError(
        "incompatible temporal parameter types: "
        + ", ".join(f"{n}={d!r}" for n, d in seen)
    )
