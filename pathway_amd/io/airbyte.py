"""pw.io.airbyte — run Airbyte sources via the Airbyte protocol.

Reference: python/pathway/io/airbyte + vendored airbyte_serverless.  An
Airbyte source is any executable that, given ``read --config <file>
[--catalog <file>] [--state <file>]``, prints Airbyte-protocol JSON
lines on stdout: RECORD (data rows), STATE (incremental checkpoints),
LOG/TRACE.  This build launches the executable directly (the
``exec:``/``python:`` flavors of the reference's venv/docker runners —
docker is unavailable offline) and re-runs it per refresh interval,
passing back the last STATE for incremental reads.

Config YAML layout (reference airbyte_serverless):

    source:
      exec: python my_source.py         # or: docker_image: ... (unsupported)
      config: { ... }                   # passed as --config JSON file
      streams: [stream1, stream2]
"""

from __future__ import annotations

import json
import os
import subprocess
import tempfile
import time as _time
from typing import Any

import yaml


class AirbyteReader:
    def __init__(self, source, exec_cmd: list[str], config: dict,
                 streams: list[str] | None, *, mode: str = "streaming",
                 refresh_interval: float = 60.0,
                 env: dict | None = None,
                 max_runs: int | None = None):
        self.source = source
        self.exec_cmd = exec_cmd
        self.config = config
        self.streams = streams
        self.mode = mode
        self.refresh_interval = refresh_interval
        self.env = env
        self.max_runs = max_runs
        self.state: Any = None
        #: key tuple -> emitted (key, row) for dedup/incremental reads
        self.seen: set = set()

    def _catalog(self) -> dict:
        streams = self.streams or []
        return {
            "streams": [
                {"stream": {"name": s, "json_schema": {},
                            "supported_sync_modes": ["full_refresh", "incremental"]},
                 "sync_mode": "incremental", "destination_sync_mode": "append"}
                for s in streams
            ]
        }

    def _run_once(self, tmpdir: str) -> None:
        cfg = os.path.join(tmpdir, "config.json")
        with open(cfg, "w") as f:
            json.dump(self.config, f)
        cat = os.path.join(tmpdir, "catalog.json")
        with open(cat, "w") as f:
            json.dump(self._catalog(), f)
        cmd = self.exec_cmd + ["read", "--config", cfg, "--catalog", cat]
        if self.state is not None:
            st = os.path.join(tmpdir, "state.json")
            with open(st, "w") as f:
                json.dump(self.state, f)
            cmd += ["--state", st]
        proc = subprocess.Popen(
            cmd, stdout=subprocess.PIPE, stderr=subprocess.DEVNULL,
            text=True, env={**os.environ, **(self.env or {})},
        )
        assert proc.stdout is not None
        for line in proc.stdout:
            line = line.strip()
            if not line:
                continue
            try:
                msg = json.loads(line)
            except json.JSONDecodeError:
                continue
            mtype = msg.get("type")
            if mtype == "RECORD":
                rec = msg["record"]
                if self.streams and rec.get("stream") not in self.streams:
                    continue
                data = rec.get("data", {})
                fp = (rec.get("stream"), json.dumps(data, sort_keys=True))
                if fp in self.seen:
                    continue  # already ingested in a previous run
                self.seen.add(fp)
                self.source.emit([rec.get("stream"), _wrap_json(data)])
            elif mtype == "STATE":
                self.state = msg["state"]
        proc.wait()
        if proc.returncode not in (0, None):
            raise RuntimeError(
                f"airbyte source exited with code {proc.returncode}"
            )

    def run(self) -> None:
        try:
            runs = 0
            with tempfile.TemporaryDirectory() as tmpdir:
                while True:
                    self._run_once(tmpdir)
                    runs += 1
                    if self.mode == "static":
                        return
                    if self.max_runs is not None and runs >= self.max_runs:
                        return
                    _time.sleep(self.refresh_interval)
        except Exception as e:
            self.source.fail(e)
        finally:
            self.source.finish()


def _wrap_json(data):
    from pathway_amd.internals.json import Json

    return Json(data)


def read(
    config: str | dict,
    streams: list[str] | None = None,
    *,
    mode: str = "streaming",
    execution_type: str | None = None,
    env_vars: dict | None = None,
    refresh_interval_ms: int = 60000,
    name: str | None = None,
    _max_runs: int | None = None,
    **kwargs: Any,
):
    """Read from an Airbyte source.

    `config` is a YAML file path or dict with reference airbyte_serverless
    layout; the `exec:` key names the source command (docker images are
    not runnable in this offline build).  Output table: (stream, data)
    with data as Json — matching the reference's airbyte table shape.
    """
    from pathway_amd.engine.nodes import InputNode
    from pathway_amd.engine.streaming import StreamingSource, spawn_reader
    from pathway_amd.internals import dtype as dt
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.table import Table
    from pathway_amd.internals.universe import Universe

    if isinstance(config, str):
        with open(config) as f:
            config = yaml.safe_load(f)
    src_cfg = config.get("source", config)
    exec_spec = src_cfg.get("exec") or src_cfg.get("python")
    if not exec_spec:
        raise ValueError(
            "airbyte source needs an 'exec:' command (docker_image is not "
            "runnable in this offline environment)"
        )
    exec_cmd = exec_spec if isinstance(exec_spec, list) else exec_spec.split()
    src_streams = streams or src_cfg.get("streams")

    names = ["stream", "data"]
    dtypes = [dt.STR, dt.JSON]
    src = StreamingSource(names, dtypes, name=name)
    reader = AirbyteReader(
        src, exec_cmd, src_cfg.get("config", {}), src_streams,
        mode=mode, refresh_interval=refresh_interval_ms / 1000.0,
        env=env_vars, max_runs=_max_runs,
    )
    src.reader = reader
    spawn_reader(reader.run, src)
    node = InputNode(src, get_device())
    return Table(node, {n: d for n, d in zip(names, dtypes)}, Universe())
