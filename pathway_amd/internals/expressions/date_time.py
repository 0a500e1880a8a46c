"""expr.dt namespace (reference internals/expressions/date_time.py, 1666 LoC).

Datetime columns are int64 nanoseconds on device; field extraction runs as
host-vectorized pandas ops (cold path) or tensor arithmetic (hot path).
"""

from __future__ import annotations

from typing import Any

from pathway_amd.internals import dtype as dt
from pathway_amd.internals.expression import ColumnExpression, MethodCallExpression
from pathway_amd.engine.expression_eval import register_method, _host_method, _tensor_method

import torch


class DateTimeNamespace:
    def __init__(self, expr: ColumnExpression):
        self._expr = expr

    def _m(self, name: str, *args: Any, return_type=None) -> MethodCallExpression:
        return MethodCallExpression(f"dt.{name}", self._expr, *args, return_type=return_type)

    def year(self):
        return self._m("year", return_type=dt.INT)

    def month(self):
        return self._m("month", return_type=dt.INT)

    def day(self):
        return self._m("day", return_type=dt.INT)

    def hour(self):
        return self._m("hour", return_type=dt.INT)

    def minute(self):
        return self._m("minute", return_type=dt.INT)

    def second(self):
        return self._m("second", return_type=dt.INT)

    def millisecond(self):
        return self._m("millisecond", return_type=dt.INT)

    def microsecond(self):
        return self._m("microsecond", return_type=dt.INT)

    def nanosecond(self):
        return self._m("nanosecond", return_type=dt.INT)

    def timestamp(self, unit: str = "ns"):
        return self._m("timestamp", unit, return_type=dt.FLOAT if unit != "ns" else dt.INT)

    def strftime(self, fmt: Any):
        return self._m("strftime", fmt, return_type=dt.STR)

    def strptime(self, fmt: Any, contains_timezone: bool = False):
        return self._m(
            "strptime",
            fmt,
            return_type=dt.DATE_TIME_UTC if contains_timezone else dt.DATE_TIME_NAIVE,
        )

    def to_utc(self, from_timezone: Any):
        return self._m("to_utc", from_timezone, return_type=dt.DATE_TIME_UTC)

    def to_naive_in_timezone(self, timezone: Any):
        return self._m("to_naive_in_timezone", timezone, return_type=dt.DATE_TIME_NAIVE)

    def round(self, duration: Any):
        return self._m("round", duration, return_type=dt.ANY)

    def floor(self, duration: Any):
        return self._m("floor", duration, return_type=dt.ANY)

    def days(self):
        return self._m("days", return_type=dt.INT)

    def hours(self):
        return self._m("hours", return_type=dt.INT)

    def minutes(self):
        return self._m("minutes", return_type=dt.INT)

    def seconds(self):
        return self._m("seconds", return_type=dt.INT)

    def milliseconds(self):
        return self._m("milliseconds", return_type=dt.INT)

    def microseconds(self):
        return self._m("microseconds", return_type=dt.INT)

    def nanoseconds(self):
        return self._m("nanoseconds", return_type=dt.INT)

    def weekday(self):
        return self._m("weekday", return_type=dt.INT)

    def from_timestamp(self, unit: str):
        return self._m("from_timestamp", unit, return_type=dt.DATE_TIME_NAIVE)

    def utc_from_timestamp(self, unit: str):
        return self._m(
            "utc_from_timestamp", unit, return_type=dt.DATE_TIME_UTC
        )

    def to_duration(self, unit: Any = "ns"):
        return self._m("to_duration", unit, return_type=dt.DURATION)

    def weeks(self):
        return self._m("weeks", return_type=dt.INT)

    def add_duration_in_timezone(self, duration: Any, timezone: Any):
        """Wall-clock addition in a time zone (DST-aware; reference
        date_time.py:855)."""
        return self._m(
            "add_duration_in_timezone", duration, timezone,
            return_type=dt.DATE_TIME_NAIVE,
        )

    def subtract_duration_in_timezone(self, duration: Any, timezone: Any):
        return self._m(
            "subtract_duration_in_timezone", duration, timezone,
            return_type=dt.DATE_TIME_NAIVE,
        )

    def subtract_date_time_in_timezone(self, date_time: Any, timezone: Any):
        """a - b computed through the zone's wall clock (DST-aware;
        reference date_time.py:943)."""
        return self._m(
            "subtract_date_time_in_timezone", date_time, timezone,
            return_type=dt.DURATION,
        )

    def utc_now(self):
        raise NotImplementedError("dt.utc_now is a table stream op; see pw.io")


def _pd_field(field: str):
    import pandas as pd

    def f(v):
        return getattr(pd.Timestamp(v), field)

    return f


register_method("dt.year", _host_method(_pd_field("year"), dt.INT))
register_method("dt.month", _host_method(_pd_field("month"), dt.INT))
register_method("dt.day", _host_method(_pd_field("day"), dt.INT))
register_method("dt.hour", _host_method(_pd_field("hour"), dt.INT))
register_method("dt.minute", _host_method(_pd_field("minute"), dt.INT))
register_method("dt.second", _host_method(_pd_field("second"), dt.INT))
register_method(
    "dt.millisecond", _host_method(lambda v: _pd_field("microsecond")(v) // 1000, dt.INT)
)
register_method("dt.microsecond", _host_method(_pd_field("microsecond"), dt.INT))
register_method("dt.weekday", _host_method(lambda v: v.weekday(), dt.INT))


def _timestamp_impl(cols, ctx):
    import pandas as pd
    from pathway_amd.engine.column import TensorColumn, column_from_pylist

    c = cols[0]
    unit = cols[1].to_pylist()[0] if len(cols) > 1 and len(cols[1]) else "ns"
    div = {"ns": 1, "us": 1e3, "ms": 1e6, "s": 1e9}[unit]
    if isinstance(c, TensorColumn):
        t = c.tensor
        if unit == "ns":
            return TensorColumn(t, dt.INT, c.mask)
        return TensorColumn(t.to(torch.float64) / div, dt.FLOAT, c.mask)
    vals = [None if v is None else pd.Timestamp(v).value / div for v in c.to_pylist()]
    return column_from_pylist(vals, dt.FLOAT, ctx.device)


register_method("dt.timestamp", _timestamp_impl)
register_method(
    "dt.strftime",
    _host_method(lambda v, fmt: __import__("pandas").Timestamp(v).strftime(fmt), dt.STR),
)


def _strptime(v, fmt):
    import pandas as pd
    from pathway_amd.internals.datetime_types import DateTimeNaive, DateTimeUtc

    ts = pd.to_datetime(v, format=fmt)
    if ts.tzinfo is not None:
        return DateTimeUtc(ts.tz_convert("UTC"))
    return DateTimeNaive(ts)


register_method("dt.strptime", _host_method(_strptime, dt.ANY))


def _dur_field(divisor: int):
    def f(v):
        import pandas as pd

        return int(pd.Timedelta(v).value // divisor)

    return f


register_method("dt.nanoseconds", _host_method(_dur_field(1), dt.INT))
register_method("dt.microseconds", _host_method(_dur_field(10**3), dt.INT))
register_method("dt.milliseconds", _host_method(_dur_field(10**6), dt.INT))
register_method("dt.seconds", _host_method(_dur_field(10**9), dt.INT))
register_method("dt.minutes", _host_method(_dur_field(60 * 10**9), dt.INT))
register_method("dt.hours", _host_method(_dur_field(3600 * 10**9), dt.INT))
register_method("dt.days", _host_method(_dur_field(86400 * 10**9), dt.INT))


def _from_timestamp(v, unit):
    import pandas as pd

    from pathway_amd.internals.datetime_types import DateTimeNaive

    return DateTimeNaive(pd.Timestamp(v, unit=unit))


register_method("dt.from_timestamp", _host_method(_from_timestamp, dt.DATE_TIME_NAIVE))


def _round_dur(v, dur):
    import pandas as pd

    return type(v)(pd.Timestamp(v).round(pd.Timedelta(dur)))


def _floor_dur(v, dur):
    import pandas as pd

    return type(v)(pd.Timestamp(v).floor(pd.Timedelta(dur)))


register_method("dt.round", _host_method(_round_dur, dt.ANY))
register_method("dt.floor", _host_method(_floor_dur, dt.ANY))


def _to_utc(v, tz):
    import pandas as pd

    from pathway_amd.internals.datetime_types import DateTimeUtc

    return DateTimeUtc(pd.Timestamp(v).tz_localize(tz).tz_convert("UTC"))


def _to_naive(v, tz):
    import pandas as pd

    from pathway_amd.internals.datetime_types import DateTimeNaive

    return DateTimeNaive(pd.Timestamp(v).tz_convert(tz).tz_localize(None))


register_method("dt.to_utc", _host_method(_to_utc, dt.DATE_TIME_UTC))
register_method("dt.to_naive_in_timezone", _host_method(_to_naive, dt.DATE_TIME_NAIVE))

# str/bin extras that need imports
register_method(
    "str.to_bytes", _host_method(lambda s, enc: s.encode(enc), dt.BYTES)
)
register_method("bin.decode", _host_method(lambda b, enc: b.decode(enc), dt.STR))
register_method(
    "bin.base64_encode",
    _host_method(lambda b: __import__("base64").b64encode(b).decode(), dt.STR),
)
register_method(
    "bin.base64_decode",
    _host_method(lambda s: __import__("base64").b64decode(s), dt.BYTES),
)


_UNIT_NS = {
    "W": 7 * 24 * 3600 * 10**9,
    "D": 24 * 3600 * 10**9, "day": 24 * 3600 * 10**9, "days": 24 * 3600 * 10**9,
    "h": 3600 * 10**9, "hr": 3600 * 10**9, "hour": 3600 * 10**9, "hours": 3600 * 10**9,
    "m": 60 * 10**9, "min": 60 * 10**9, "minute": 60 * 10**9, "minutes": 60 * 10**9,
    "s": 10**9, "sec": 10**9, "second": 10**9, "seconds": 10**9,
    "ms": 10**6, "millisecond": 10**6, "milliseconds": 10**6, "millis": 10**6, "milli": 10**6,
    "us": 10**3,
    "ns": 1, "nano": 1, "nanos": 1, "nanosecond": 1, "nanoseconds": 1,
}


def _to_duration(v, unit):
    from pathway_amd.internals.datetime_types import Duration

    return Duration(int(v) * _UNIT_NS[unit], unit="ns")


def _utc_from_timestamp(v, unit):
    import pandas as pd

    from pathway_amd.internals.datetime_types import DateTimeUtc

    mul = {"s": 10**9, "ms": 10**6, "us": 10**3, "ns": 1}[unit]
    return DateTimeUtc(pd.Timestamp(int(v * mul), unit="ns", tz="UTC"))


def _weeks(v):
    import pandas as pd

    return int(pd.Timedelta(v).value // (7 * 24 * 3600 * 10**9))


def _add_dur_tz(v, dur, tz, sign=1):
    import pandas as pd

    from pathway_amd.internals.datetime_types import DateTimeNaive

    aware = pd.Timestamp(v).tz_localize(tz, ambiguous=True)
    res = aware + sign * pd.Timedelta(dur)
    return DateTimeNaive(res.tz_convert(tz).tz_localize(None))


def _sub_dt_tz(a, b, tz):
    import pandas as pd

    from pathway_amd.internals.datetime_types import Duration

    aa = pd.Timestamp(a).tz_localize(tz, ambiguous=True)
    bb = pd.Timestamp(b).tz_localize(tz, ambiguous=True)
    return Duration((aa - bb).value, unit="ns")


register_method("dt.to_duration", _host_method(_to_duration, dt.DURATION))
register_method(
    "dt.utc_from_timestamp", _host_method(_utc_from_timestamp, dt.DATE_TIME_UTC)
)
register_method("dt.weeks", _host_method(_weeks, dt.INT))
register_method(
    "dt.add_duration_in_timezone",
    _host_method(_add_dur_tz, dt.DATE_TIME_NAIVE),
)
register_method(
    "dt.subtract_duration_in_timezone",
    _host_method(lambda v, d, tz: _add_dur_tz(v, d, tz, sign=-1), dt.DATE_TIME_NAIVE),
)
register_method(
    "dt.subtract_date_time_in_timezone",
    _host_method(_sub_dt_tz, dt.DURATION),
)
