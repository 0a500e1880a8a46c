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

    @app.get("/metrics")
    def metrics():
        from fastapi.responses import PlainTextResponse

        return PlainTextResponse(stats.openmetrics())

    @app.get("/")
    def index():
        s = stats.snapshot()
        rows = "".join(
            f"<tr><td>{k}</td><td>{v}</td></tr>"
            for k, v in s.items()
            if k != "operators"
        )
        return HTMLResponse(
            "<html><head><title>pathway_amd</title></head><body>"
            "<h2>pathway_amd engine</h2><table border=1>"
            f"{rows}</table></body></html>"
        )

    return app


def run_dashboard(host: str = "127.0.0.1", port: int = 8501, stats=None):
    import uvicorn

    uvicorn.run(create_app(stats), host=host, port=port, log_level="warning")
