"""Debezium CDC envelope parser (reference data_format/debezium.rs).

A Debezium message (JSON, or Avro via the registry) carries
``{"before": ..., "after": ..., "op": "c|u|d|r", "source": {...}, "ts_ms": N}``,
optionally wrapped in a Kafka-Connect ``{"schema": ..., "payload": ...}``
envelope.  The parser turns each message into engine update events:

  op=c/r  ->  +after
  op=d    ->  -before
  op=u    ->  -before, +after

matching the reference's DebeziumMessageParser (insert/delete pairs with
the key taken from the message key or the configured primary-key columns).
"""

from __future__ import annotations

import json
from dataclasses import dataclass
from typing import Any


@dataclass
class ChangeEvent:
    values: dict[str, Any]
    diff: int  # +1 insert, -1 delete
    key: tuple | None = None
    ts_ms: int | None = None


def unwrap_connect_envelope(obj: Any) -> Any:
    """Strip the Kafka-Connect {"schema":..., "payload":...} wrapper."""
    if isinstance(obj, dict) and set(obj.keys()) <= {"schema", "payload"} and "payload" in obj:
        return obj["payload"]
    return obj


def parse_message(
    value: bytes | str | dict | None,
    key: bytes | str | dict | None = None,
    *,
    primary_key: list[str] | None = None,
) -> list[ChangeEvent]:
    """Parse one Debezium message into change events.

    A None value (tombstone) yields no events — the preceding op=d already
    carried the delete, matching the reference's tombstone handling.
    """
    if value is None or value == b"" or value == "":
        return []
    obj = value if isinstance(value, dict) else json.loads(value)
    obj = unwrap_connect_envelope(obj)
    if not isinstance(obj, dict) or "op" not in obj:
        raise ValueError("not a Debezium envelope (missing 'op')")

    kobj = None
    if key not in (None, b"", ""):
        kobj = key if isinstance(key, dict) else json.loads(key)
        kobj = unwrap_connect_envelope(kobj)

    def key_of(row: dict[str, Any]) -> tuple | None:
        if kobj is not None:
            return tuple(kobj.values()) if isinstance(kobj, dict) else (kobj,)
        if primary_key:
            return tuple(row[c] for c in primary_key)
        return None

    op = obj["op"]
    ts = obj.get("ts_ms")
    before = obj.get("before")
    after = obj.get("after")
    out: list[ChangeEvent] = []
    if op in ("c", "r"):
        if after is None:
            raise ValueError(f"op={op} with null 'after'")
        out.append(ChangeEvent(after, +1, key_of(after), ts))
    elif op == "d":
        if before is None:
            raise ValueError("op=d with null 'before'")
        out.append(ChangeEvent(before, -1, key_of(before), ts))
    elif op == "u":
        if before is not None:
            out.append(ChangeEvent(before, -1, key_of(before), ts))
        if after is None:
            raise ValueError("op=u with null 'after'")
        out.append(ChangeEvent(after, +1, key_of(after), ts))
    else:
        raise ValueError(f"unknown Debezium op {op!r}")
    return out


def format_message(
    before: dict[str, Any] | None,
    after: dict[str, Any] | None,
    *,
    source: dict[str, Any] | None = None,
    ts_ms: int | None = None,
) -> bytes:
    """Build a Debezium-envelope JSON message (for sinks/tests)."""
    if after is not None and before is None:
        op = "c"
    elif after is None and before is not None:
        op = "d"
    elif after is not None and before is not None:
        op = "u"
    else:
        raise ValueError("before and after cannot both be null")
    return json.dumps(
        {
            "before": before,
            "after": after,
            "op": op,
            "source": source or {},
            "ts_ms": ts_ms,
        }
    ).encode("utf-8")
