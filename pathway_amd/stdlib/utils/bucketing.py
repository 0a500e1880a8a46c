"""Time-bucketing helpers (reference stdlib/utils/bucketing.py)."""
from __future__ import annotations

import datetime


def truncate_to_minutes(time: datetime.datetime) -> datetime.datetime:
    """Drop seconds and sub-seconds from a datetime."""
    return time.replace(second=0, microsecond=0)


def truncate_to_hours(time: datetime.datetime) -> datetime.datetime:
    """Drop minutes and below from a datetime."""
    return time.replace(minute=0, second=0, microsecond=0)
