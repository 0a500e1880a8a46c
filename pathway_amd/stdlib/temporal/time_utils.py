"""Wall-clock utilities: utc_now stream, inactivity detection, update
timestamps (reference stdlib/temporal/time_utils.py:42-230 behavior).
"""

from __future__ import annotations

import datetime
from functools import cache

import pathway_amd as pw


class TimestampSchema(pw.Schema):
    timestamp_utc: pw.DateTimeUtc


class TimestampSubject(pw.io.python.ConnectorSubject):
    """Emits the current UTC time every `refresh_rate` (forever, or
    `max_ticks` times when bounded for tests)."""

    def __init__(
        self,
        refresh_rate: datetime.timedelta,
        initial_delay: datetime.timedelta = datetime.timedelta(0),
        max_ticks: int | None = None,
    ):
        super().__init__()
        self._refresh_rate = refresh_rate
        self._initial_delay = initial_delay
        self._max_ticks = max_ticks

    def run(self) -> None:
        import time as _time

        _time.sleep(self._initial_delay.total_seconds())
        ticks = 0
        while self._max_ticks is None or ticks < self._max_ticks:
            now_utc = datetime.datetime.now(tz=datetime.timezone.utc)
            self.next(timestamp_utc=pw.DateTimeUtc(now_utc))
            self.commit()
            ticks += 1
            _time.sleep(self._refresh_rate.total_seconds())


@cache
def utc_now(
    refresh_rate: datetime.timedelta = datetime.timedelta(seconds=60),
    initial_delay: datetime.timedelta = datetime.timedelta(seconds=0),
    max_ticks: int | None = None,
):
    """Continuously updating stream of the current UTC time (one row per
    tick; reference time_utils.py:42)."""
    return pw.io.python.read(
        TimestampSubject(
            refresh_rate=refresh_rate,
            initial_delay=initial_delay,
            max_ticks=max_ticks,
        ),
        schema=TimestampSchema,
    )


def _now_utc() -> pw.DateTimeUtc:
    return pw.DateTimeUtc(datetime.datetime.now(tz=datetime.timezone.utc))


def inactivity_detection(
    self: pw.Table,
    allowed_inactivity_period,
    refresh_rate=None,
    instance=None,
    _max_ticks: int | None = None,
):
    """Detect periods with no additions to an (append-only) table
    (reference time_utils.py:72-185): emits one row per inactivity period
    with the UTC timestamp of the last activity before the gap and the
    first activity that ended it (None while ongoing)."""
    if refresh_rate is None:
        refresh_rate = datetime.timedelta(seconds=1)

    utc_now_table = utc_now(
        refresh_rate=refresh_rate, max_ticks=_max_ticks
    ).reduce(timestamp_utc=pw.reducers.latest(pw.this.timestamp_utc))

    @pw.udf(deterministic=True)
    def stamp(for_test_only: pw.Pointer) -> pw.DateTimeUtc:
        return _now_utc()

    latest_activities = (
        self.select(instance=instance, timestamp_utc=stamp(pw.this.id))
        .groupby(pw.this.instance)
        .reduce(
            pw.this.instance,
            timestamp_utc=pw.reducers.latest(pw.this.timestamp_utc),
        )
    )

    start_timestamp_utc = _now_utc()
    latest_inactivities = (
        latest_activities.join_right(utc_now_table)
        .select(
            pw.left.instance,
            pw.left.timestamp_utc,
            now_utc=pw.right.timestamp_utc,
        )
        .with_columns(
            timestamp_utc=pw.coalesce(pw.this.timestamp_utc, start_timestamp_utc)
        )
        .filter(pw.this.timestamp_utc + allowed_inactivity_period < pw.this.now_utc)
        .select(pw.this.instance, inactivity_timestamp_utc=pw.this.timestamp_utc)
    )

    inactivities = (
        latest_inactivities._remove_retractions()
        .groupby(pw.this.instance, pw.this.inactivity_timestamp_utc)
        .reduce(pw.this.instance, pw.this.inactivity_timestamp_utc)
    )

    latest_resumed = (
        inactivities.groupby(pw.this.instance)
        .reduce(
            pw.this.instance,
            inactivity_timestamp_utc=pw.reducers.latest(
                pw.this.inactivity_timestamp_utc
            ),
        )
        .join_inner(latest_activities, pw.left.instance == pw.right.instance)
        .select(
            pw.left.instance,
            pw.left.inactivity_timestamp_utc,
            latest_activity_timestamp_utc=pw.right.timestamp_utc,
        )
        .filter(
            pw.this.inactivity_timestamp_utc < pw.this.latest_activity_timestamp_utc
        )
    )

    resumed = (
        latest_resumed._remove_retractions()
        .groupby(pw.this.instance, pw.this.inactivity_timestamp_utc)
        .reduce(
            pw.this.instance,
            pw.this.inactivity_timestamp_utc,
            resumed_activity_timestamp_utc=pw.reducers.earliest(
                pw.this.latest_activity_timestamp_utc
            ),
        )
    )

    out = inactivities.join_left(
        resumed,
        pw.left.instance == pw.right.instance,
        pw.left.inactivity_timestamp_utc == pw.right.inactivity_timestamp_utc,
    ).select(
        pw.left.instance,
        pw.left.inactivity_timestamp_utc,
        pw.right.resumed_activity_timestamp_utc,
    )
    if instance is None:
        out = out.without(pw.this.instance)
    return out


def add_update_timestamp_utc(
    self: pw.Table,
    refresh_rate=None,
    update_timestamp_column_name: str = "updated_timestamp_utc",
    _max_ticks: int | None = None,
):
    """Add a column holding the UTC time each row was last updated
    (reference time_utils.py:191-230)."""
    if refresh_rate is None:
        refresh_rate = datetime.timedelta(seconds=1)
    utc_single = utc_now(refresh_rate=refresh_rate, max_ticks=_max_ticks).reduce(
        timestamp_utc=pw.reducers.latest(pw.this.timestamp_utc)
    )

    @pw.udf(deterministic=True)
    def stamp_now(ts) -> pw.DateTimeUtc:
        return ts if ts is not None else _now_utc()

    joined = self.asof_now_join_left(utc_single).select(
        *[pw.left[n] for n in self.column_names()],
        **{update_timestamp_column_name: stamp_now(pw.right.timestamp_utc)},
    )
    return joined
