"""Web dashboard (reference python/pathway/web_dashboard/dashboard.py):
FastAPI app over the engine RunStats (and optional detailed-metrics dir)."""
from __future__ import annotations

from typing import Any


def create_app(stats=None, detailed_metrics_dir: str | None = None):
    from fastapi import FastAPI
    from fastapi.responses import HTMLResponse, JSONResponse

    if stats is None:
        from pathway_amd.engine.monitoring import GLOBAL_STATS as stats  # noqa: N813

    app = FastAPI(title="pathway_amd dashboard")

    @app.get("/api/stats")
    def api_stats():
        return JSONResponse(stats.snapshot())

    @app.get("/api/operators")
    def api_operators():
        s = stats.snapshot()
        return JSONResponse(s.get("operators", {}))

    @app.get("/api/history")
    def api_history(limit: int = 500):
        """Time series from the detailed-metrics SQLite (reference
        detailed_metrics_dir -> metrics.db)."""
        import os
        import sqlite3

        if not detailed_metrics_dir:
            return JSONResponse({"error": "no detailed_metrics_dir"}, status_code=404)
        db = os.path.join(detailed_metrics_dir, "metrics.db")
        if not os.path.exists(db):
            return JSONResponse({"runs": [], "points": []})
        conn = sqlite3.connect(db)
        try:
            points = conn.execute(
                "SELECT ts, engine_time, steps, rows_ingested, rows_output,"
                " p50_ms, p95_ms FROM run_metrics ORDER BY ts DESC LIMIT ?",
                (limit,),
            ).fetchall()
            ops = conn.execute(
                "SELECT operator, MAX(steps), MAX(rows_in), MAX(rows_out),"
                " MAX(total_time_s) FROM operator_metrics GROUP BY operator"
            ).fetchall()
        finally:
            conn.close()
        return JSONResponse({
            "points": [
                {"ts": p[0], "engine_time": p[1], "steps": p[2],
                 "rows_ingested": p[3], "rows_output": p[4],
                 "p50_ms": p[5], "p95_ms": p[6]}
                for p in reversed(points)
            ],
            "operators": [
                {"operator": o[0], "steps": o[1], "rows_in": o[2],
                 "rows_out": o[3], "total_time_s": o[4]}
                for o in ops
            ],
        })

    @app.get("/metrics")
    def metrics():
        from fastapi.responses import PlainTextResponse

        return PlainTextResponse(stats.openmetrics())

    @app.get("/")
    def index():
        from pathway_amd.stdlib.viz import render_svg

        s = stats.snapshot()
        rows = "".join(
            f"<tr><td>{k}</td><td>{v}</td></tr>"
            for k, v in s.items()
            if k != "operators"
        )
        op_rows = "".join(
            f"<tr><td>{name}</td><td>{st['steps']}</td>"
            f"<td>{st['rows_in']}</td><td>{st['rows_out']}</td>"
            f"<td>{st['total_time_s']:.4f}</td></tr>"
            for name, st in (s.get("operators") or {}).items()
        )
        spark = render_svg(list(stats.step_latencies_ms)[-200:])
        return HTMLResponse(
            "<html><head><title>pathway_amd</title></head><body>"
            "<h2>pathway_amd engine</h2><table border=1>"
            f"{rows}</table>"
            f"<h3>step latency (ms, last 200)</h3>{spark}"
            "<h3>operators</h3><table border=1>"
            "<tr><th>operator</th><th>steps</th><th>rows in</th>"
            "<th>rows out</th><th>total s</th></tr>"
            f"{op_rows}</table></body></html>"
        )

    return app


def run_dashboard(host: str = "127.0.0.1", port: int = 8501, stats=None):
    import uvicorn

    uvicorn.run(create_app(stats), host=host, port=port, log_level="warning")
