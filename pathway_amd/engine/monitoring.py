"""Progress monitoring + HTTP metrics
(reference src/engine/dataflow/monitoring.rs ProberStats,
src/engine/http_server.rs /metrics OpenMetrics + JSON status,
python internals/monitoring.py rich dashboard).
"""

from __future__ import annotations

import json
import threading
import time
from collections import defaultdict
from dataclasses import dataclass, field
from typing import Any


@dataclass
class OperatorStats:
    rows_in: int = 0
    rows_out: int = 0
    steps: int = 0
    total_time_s: float = 0.0


@dataclass
class RunStats:
    """Engine-wide counters, updated per step by the Runtime."""

    started_at: float = field(default_factory=time.time)
    current_time: int | None = None
    steps: int = 0
    rows_ingested: int = 0
    rows_output: int = 0
    step_latencies_ms: list = field(default_factory=list)
    operators: dict = field(default_factory=lambda: defaultdict(OperatorStats))
    exchanges: int = 0
    exchange_rows: int = 0
    #: max-destination load over mean (1.0 = perfectly balanced); EWMA
    exchange_skew: float = 1.0

    def record_step(self, t: int, latency_s: float, ingested: int, output: int):
        self.current_time = t
        self.steps += 1
        self.rows_ingested += ingested
        self.rows_output += output
        self.step_latencies_ms.append(latency_s * 1000)
        if len(self.step_latencies_ms) > 1000:
            self.step_latencies_ms = self.step_latencies_ms[-1000:]

    def observe_exchange(self, rows: int, skew_ratio: float) -> None:
        """Shard skew telemetry: max/mean destination load of
        one all-to-all partition, EWMA-smoothed."""
        self.exchanges += 1
        self.exchange_rows += rows
        self.exchange_skew = 0.9 * self.exchange_skew + 0.1 * skew_ratio

    def latency_quantile(self, q: float) -> float | None:
        if not self.step_latencies_ms:
            return None
        xs = sorted(self.step_latencies_ms)
        return xs[min(int(len(xs) * q), len(xs) - 1)]

    def snapshot(self) -> dict:
        return {
            "uptime_s": time.time() - self.started_at,
            "current_time": self.current_time,
            "steps": self.steps,
            "rows_ingested": self.rows_ingested,
            "rows_output": self.rows_output,
            "p50_step_ms": self.latency_quantile(0.5),
            "p95_step_ms": self.latency_quantile(0.95),
            "exchanges": self.exchanges,
            "exchange_rows": self.exchange_rows,
            "exchange_skew": round(self.exchange_skew, 4),
            "operators": {
                name: vars(st) for name, st in self.operators.items()
            },
        }

    def openmetrics(self) -> str:
        s = self.snapshot()
        lines = [
            "# TYPE pathway_steps counter",
            f"pathway_steps_total {s['steps']}",
            "# TYPE pathway_rows_ingested counter",
            f"pathway_rows_ingested_total {s['rows_ingested']}",
            "# TYPE pathway_rows_output counter",
            f"pathway_rows_output_total {s['rows_output']}",
            "# TYPE pathway_exchange_skew gauge",
            f"pathway_exchange_skew {s['exchange_skew']}",
            "# TYPE pathway_step_latency_ms gauge",
            f"pathway_step_latency_ms{{quantile=\"0.5\"}} {s['p50_step_ms'] or 0}",
            f"pathway_step_latency_ms{{quantile=\"0.95\"}} {s['p95_step_ms'] or 0}",
            "# EOF",
        ]
        return "\n".join(lines) + "\n"


GLOBAL_STATS = RunStats()


def start_http_server(stats: RunStats, port: int | None = None) -> Any:
    """/metrics (OpenMetrics) + /status (JSON) — reference http_server.rs:
    port 20000 + process id."""
    import os
    from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

    if port is None:
        port = 20000 + int(os.environ.get("PATHWAY_PROCESS_ID", "0"))

    class Handler(BaseHTTPRequestHandler):
        def do_GET(self):
            if self.path.startswith("/metrics"):
                body = stats.openmetrics().encode()
                ctype = "application/openmetrics-text; version=1.0.0"
            else:
                body = json.dumps(stats.snapshot(), default=str).encode()
                ctype = "application/json"
            self.send_response(200)
            self.send_header("Content-Type", ctype)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, *a):
            pass

    httpd = ThreadingHTTPServer(("0.0.0.0", port), Handler)
    th = threading.Thread(target=httpd.serve_forever, daemon=True)
    th.start()
    return httpd


class ConsoleMonitor:
    """Lightweight terminal dashboard (reference internals/monitoring.py
    rich table) — prints a periodic one-line progress summary."""

    def __init__(self, stats: RunStats, interval_s: float = 5.0):
        self.stats = stats
        self.interval_s = interval_s
        self._last = 0.0

    def maybe_report(self):
        now = time.time()
        if now - self._last >= self.interval_s:
            self._last = now
            s = self.stats.snapshot()
            print(
                f"[pathway_amd] t={s['current_time']} steps={s['steps']} "
                f"in={s['rows_ingested']} out={s['rows_output']} "
                f"p95={s['p95_step_ms'] and round(s['p95_step_ms'], 2)}ms",
                flush=True,
            )


EXIT_CODE_UPSCALE = 77
EXIT_CODE_DOWNSCALE = 78


class WorkloadTracker:
    """Elastic-scaling advice (reference src/engine/workload_tracker.rs +
    dataflow.rs:7455-7499): sliding-window busy-fraction → ScaleUp/Down.

    Enabled by PATHWAY_ELASTIC=1 under the `pathway_amd spawn` supervisor
    (which restarts with ±workers on the exit codes); requires persistence
    so the restart can recover state."""

    def __init__(self, window: int = 50, high: float = 0.85, low: float = 0.15):
        self.window = window
        self.high = high
        self.low = low
        self.samples: list[float] = []

    def add_point(self, busy_fraction: float) -> str | None:
        self.samples.append(busy_fraction)
        if len(self.samples) < self.window:
            return None
        self.samples = self.samples[-self.window :]
        avg = sum(self.samples) / len(self.samples)
        if avg > self.high:
            return "up"
        if avg < self.low:
            return "down"
        return None


class DetailedMetricsRecorder:
    """Per-operator time series into SQLite (reference detailed-metrics
    dir + web dashboard: set_monitoring_config(detailed_metrics_dir=...)
    -> metrics.db read by web_dashboard/dashboard.py)."""

    def __init__(self, directory: str, run_id: str | None = None):
        import os
        import sqlite3
        import uuid

        os.makedirs(directory, exist_ok=True)
        self.path = os.path.join(directory, "metrics.db")
        self.run_id = run_id or uuid.uuid4().hex[:12]
        self.conn = sqlite3.connect(self.path)
        self.conn.execute(
            "CREATE TABLE IF NOT EXISTS operator_metrics ("
            " run_id TEXT, ts REAL, engine_time INTEGER, operator TEXT,"
            " steps INTEGER, rows_in INTEGER, rows_out INTEGER,"
            " total_time_s REAL)"
        )
        self.conn.execute(
            "CREATE TABLE IF NOT EXISTS run_metrics ("
            " run_id TEXT, ts REAL, engine_time INTEGER, steps INTEGER,"
            " rows_ingested INTEGER, rows_output INTEGER, p50_ms REAL,"
            " p95_ms REAL)"
        )
        self.conn.commit()

    def record(self, stats: "RunStats") -> None:
        import time as _t

        now = _t.time()
        s = stats.snapshot()
        self.conn.execute(
            "INSERT INTO run_metrics VALUES (?,?,?,?,?,?,?,?)",
            (self.run_id, now, s["current_time"] or 0, s["steps"],
             s["rows_ingested"], s["rows_output"],
             s["p50_step_ms"] or 0.0, s["p95_step_ms"] or 0.0),
        )
        for name, st in stats.operators.items():
            self.conn.execute(
                "INSERT INTO operator_metrics VALUES (?,?,?,?,?,?,?,?)",
                (self.run_id, now, s["current_time"] or 0, name, st.steps,
                 st.rows_in, st.rows_out, st.total_time_s),
            )
        self.conn.commit()

    def close(self) -> None:
        self.conn.close()
