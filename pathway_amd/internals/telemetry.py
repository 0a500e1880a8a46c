"""Telemetry (reference src/engine/telemetry.rs: OTLP traces/metrics
exported every 60 s).

Two transports, no SDK dependency:
  * OTLP/HTTP with JSON encoding (a spec transport) — spans batch to
    <endpoint>/v1/traces and metric samples to /v1/metrics in the
    standard resourceSpans/resourceMetrics shape, on the 60 s sampler
    thread (telemetry.rs:609 analog);
  * a local JSONL file (export_path) for offline runs.
"""
from __future__ import annotations

import json
import threading
import time
import uuid
from contextlib import contextmanager
from typing import Any


class Telemetry:
    def __init__(self, endpoint: str | None = None, run_id: str | None = None,
                 license_key: str | None = None, export_path: str | None = None,
                 service_name: str = "pathway_amd"):
        self.endpoint = endpoint.rstrip("/") if endpoint else None
        self.run_id = run_id or str(uuid.uuid4())
        self.export_path = export_path
        self.service_name = service_name
        self._fh = open(export_path, "a") if export_path else None
        self._lock = threading.Lock()
        self._thread = None
        self._stop = threading.Event()
        self._pending_spans: list[dict] = []
        self._pending_metrics: list[dict] = []
        self._trace_id = uuid.uuid4().hex

    @classmethod
    def create(cls, license_key=None, telemetry_servers=None, run_id=None, export_path=None):
        ep = telemetry_servers[0] if telemetry_servers else None
        return cls(ep, run_id, license_key, export_path)

    def _emit(self, record: dict):
        record["run_id"] = self.run_id
        record["ts_ns"] = time.time_ns()
        with self._lock:
            if self._fh is not None:
                self._fh.write(json.dumps(record, default=str) + "\n")
                self._fh.flush()
            if self.endpoint:
                if record.get("kind") == "span":
                    self._pending_spans.append(record)
                else:
                    self._pending_metrics.append(record)

    # -- OTLP/HTTP JSON encoding (opentelemetry-proto JSON mapping) --

    def _otlp_resource(self) -> dict:
        return {"attributes": [
            {"key": "service.name",
             "value": {"stringValue": self.service_name}},
            {"key": "pathway.run_id", "value": {"stringValue": self.run_id}},
        ]}

    @staticmethod
    def _otlp_attrs(attrs: dict) -> list:
        out = []
        for k, v in (attrs or {}).items():
            if isinstance(v, bool):
                val = {"boolValue": v}
            elif isinstance(v, int):
                val = {"intValue": str(v)}
            elif isinstance(v, float):
                val = {"doubleValue": v}
            else:
                val = {"stringValue": str(v)}
            out.append({"key": str(k), "value": val})
        return out

    def flush_otlp(self) -> None:
        """POST pending spans/metrics as OTLP/HTTP JSON."""
        if not self.endpoint:
            return
        with self._lock:
            spans, self._pending_spans = self._pending_spans, []
            metrics, self._pending_metrics = self._pending_metrics, []
        from pathway_amd.io import _rest

        if spans:
            body = {"resourceSpans": [{
                "resource": self._otlp_resource(),
                "scopeSpans": [{
                    "scope": {"name": "pathway_amd"},
                    "spans": [{
                        "traceId": self._trace_id,
                        "spanId": uuid.uuid4().hex[:16],
                        "name": s["name"],
                        "kind": 1,
                        "startTimeUnixNano": str(s["start_ns"]),
                        "endTimeUnixNano": str(s["end_ns"]),
                        "attributes": self._otlp_attrs(s.get("attributes")),
                        "status": {"code": 2 if s.get("status") == "error" else 1},
                    } for s in spans],
                }],
            }]}
            try:
                _rest.request("POST", f"{self.endpoint}/v1/traces", body=body,
                              retries=0)
            except Exception:
                pass  # telemetry must never break the engine
        if metrics:
            body = {"resourceMetrics": [{
                "resource": self._otlp_resource(),
                "scopeMetrics": [{
                    "scope": {"name": "pathway_amd"},
                    "metrics": [{
                        "name": m["name"],
                        "gauge": {"dataPoints": [{
                            "timeUnixNano": str(m["ts_ns"]),
                            "asDouble": float(m["value"]),
                            "attributes": self._otlp_attrs(m.get("attributes")),
                        }]},
                    } for m in metrics],
                }],
            }]}
            try:
                _rest.request("POST", f"{self.endpoint}/v1/metrics", body=body,
                              retries=0)
            except Exception:
                pass

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
                self.flush_otlp()

        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def close(self):
        self._stop.set()
        self.flush_otlp()
        if self._fh:
            self._fh.close()


NOOP = Telemetry()
