"""Live table visualization (reference stdlib/viz: panel/bokeh plots).

panel/bokeh are not part of this image, so the MI355X build implements
the live-plot path natively: `plot(table)` attaches a subscriber that
maintains the table's current state and serves a self-contained HTML
page (inline JS polling a JSON endpoint) with an SVG line chart — the
same "live updating plot of a streaming table" capability, zero
dependencies.  `render_svg` is the pure renderer (also used for static
export); `table_viz` returns the current HTML snapshot.
"""

from __future__ import annotations

import json
import threading
from typing import Any


def render_svg(series: list[float], *, width: int = 640, height: int = 240,
               stroke: str = "#2a6fdb") -> str:
    """Static SVG line chart of a numeric series."""
    if not series:
        return (f'<svg xmlns="http://www.w3.org/2000/svg" width="{width}" '
                f'height="{height}"></svg>')
    lo, hi = min(series), max(series)
    span = (hi - lo) or 1.0
    n = len(series)
    pts = []
    for i, v in enumerate(series):
        x = i * (width - 20) / max(n - 1, 1) + 10
        y = height - 10 - (v - lo) * (height - 20) / span
        pts.append(f"{x:.1f},{y:.1f}")
    return (
        f'<svg xmlns="http://www.w3.org/2000/svg" width="{width}" '
        f'height="{height}"><polyline fill="none" stroke="{stroke}" '
        f'stroke-width="2" points="{" ".join(pts)}"/></svg>'
    )


_PAGE = """<!doctype html>
<html><head><title>pathway_amd live plot</title></head>
<body>
<h3>%(title)s</h3>
<div id="chart"></div>
<table id="rows" border="1" cellpadding="4"></table>
<script>
async function tick() {
  try {
    const r = await fetch('/data');
    const d = await r.json();
    document.getElementById('chart').innerHTML = d.svg;
    const tbl = document.getElementById('rows');
    tbl.innerHTML = '';
    if (d.rows.length) {
      const head = tbl.insertRow();
      for (const c of d.columns) head.insertCell().innerHTML = '<b>' + c + '</b>';
      for (const row of d.rows.slice(-50)) {
        const tr = tbl.insertRow();
        for (const v of row) tr.insertCell().textContent = v;
      }
    }
  } catch (e) {}
  setTimeout(tick, %(refresh_ms)d);
}
tick();
</script></body></html>
"""


class LivePlot:
    """State collector + HTTP server behind `plot(table)`."""

    def __init__(self, columns: list[str], value_column: str | None,
                 title: str, refresh_ms: int):
        self.columns = columns
        self.value_column = value_column or (columns[0] if columns else None)
        self.title = title
        self.refresh_ms = refresh_ms
        self.state: dict = {}
        self.lock = threading.Lock()
        self.server = None

    def on_change(self, key, row, time, is_addition):
        with self.lock:
            if is_addition:
                self.state[repr(key)] = [row.get(c) for c in self.columns]
            else:
                self.state.pop(repr(key), None)

    def _series(self) -> list[float]:
        if self.value_column is None:
            return []
        idx = self.columns.index(self.value_column)
        out = []
        for vals in self.state.values():
            v = vals[idx]
            if isinstance(v, (int, float)) and not isinstance(v, bool):
                out.append(float(v))
        return out

    def html(self) -> str:
        return _PAGE % {"title": self.title, "refresh_ms": self.refresh_ms}

    def data(self) -> dict:
        with self.lock:
            rows = [[str(v) for v in vals] for vals in self.state.values()]
        return {"columns": self.columns, "rows": rows,
                "svg": render_svg(sorted(self._series()))}

    def serve(self, host: str = "127.0.0.1", port: int = 0):
        from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

        lp = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def do_GET(self):
                if self.path.startswith("/data"):
                    body = json.dumps(lp.data()).encode()
                    ctype = "application/json"
                else:
                    body = lp.html().encode()
                    ctype = "text/html"
                self.send_response(200)
                self.send_header("Content-Type", ctype)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

        self.server = ThreadingHTTPServer((host, port), Handler)
        th = threading.Thread(target=self.server.serve_forever, daemon=True)
        th.start()
        return self.server


def plot(table, value_column: str | None = None, *, title: str = "pathway table",
         refresh_ms: int = 500, serve: bool = True, host: str = "127.0.0.1",
         port: int = 0) -> LivePlot:
    """Live plot of a streaming table (reference table.plot): subscribes
    to the update stream; serves an auto-refreshing HTML page."""
    import pathway_amd as pw

    columns = table.column_names()
    lp = LivePlot(columns, value_column, title, refresh_ms)
    pw.io.subscribe(table, lp.on_change)
    if serve:
        lp.serve(host, port)
    return lp


def table_viz(table, **kwargs) -> LivePlot:
    return plot(table, serve=False, **kwargs)
