"""Telemetry (reference src/engine/telemetry.rs: OTLP traces/metrics every
60 s).  The opentelemetry SDK is not installed in this image; spans and
metric samples are emitted in OTLP-shaped JSON to a local file (or dropped
when unconfigured) — same schema, file transport."""
from __future__ import annotations

import json
import threading
import time
import uuid
from contextlib import contextmanager
from typing import Any


class Telemetry:
    def __init__(self, endpoint: str | None = None, run_id: str | None = None,
                 license_key: str | None = None, export_path: str | None = None):
        self.endpoint = endpoint
        self.run_id = run_id or str(uuid.uuid4())
        self.export_path = export_path
        self._fh = open(export_path, "a") if export_path else None
        self._lock = threading.Lock()
        self._thread = None
        self._stop = threading.Event()

    @classmethod
    def create(cls, license_key=None, telemetry_servers=None, run_id=None, export_path=None):
        ep = telemetry_servers[0] if telemetry_servers else None
        return cls(ep, run_id, license_key, export_path)

    def _emit(self, record: dict):
        if self._fh is None:
            return
        record["run_id"] = self.run_id
        record["ts_ns"] = time.time_ns()
        with self._lock:
            self._fh.write(json.dumps(record, default=str) + "\n")
            self._fh.flush()

    @contextmanager
    def span(self, name: str, **attrs: Any):
        t0 = time.time_ns()
        err = None
        try:
            yield
        except Exception as e:  # noqa: BLE001
            err = e
            raise
        finally:
            self._emit(
                {
                    "kind": "span",
                    "name": name,
                    "start_ns": t0,
                    "end_ns": time.time_ns(),
                    "attributes": attrs,
                    "status": "error" if err else "ok",
                }
            )

    def gauge(self, name: str, value: float, **attrs: Any):
        self._emit({"kind": "metric", "name": name, "value": value, "attributes": attrs})

    def start_periodic(self, stats, interval_s: float = 60.0):
        """Sample engine stats periodically (reference telemetry.rs:609)."""

        def loop():
            while not self._stop.wait(interval_s):
                s = stats.snapshot()
                for k in ("steps", "rows_ingested", "rows_output", "p95_step_ms"):
                    if s.get(k) is not None:
                        self.gauge(f"pathway.{k}", float(s[k]))

        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def close(self):
        self._stop.set()
        if self._fh:
            self._fh.close()


NOOP = Telemetry()
