"""Global run graph: registry of sinks + run()/reset machinery.

The analog of the reference's ParseGraph G (internals/parse_graph.py:103)
plus GraphRunner (graph_runner/__init__.py:40): tables carry their engine
node eagerly; running = topo walk + Runtime over the registered sinks.
"""

from __future__ import annotations

from typing import Any

from pathway_amd.internals.config import get_device


class RunGraph:
    def __init__(self) -> None:
        self.sinks: list[Any] = []  # engine sink nodes (Output/Subscribe/Capture)
        self.error_log_tables: list[Any] = []
        self.services: list[Any] = []  # rest_connector webservers
        self.comm = None  # parallel context (set by parallel.init)

    def add_sink(self, node: Any) -> None:
        self.sinks.append(node)

    def clear(self) -> None:
        self.sinks.clear()
        self.error_log_tables.clear()
        self.services.clear()

    def run_nodes(
        self,
        extra_sinks: list[Any] | None = None,
        monitoring=None,
        persistence_config=None,
        **kwargs,
    ):
        from pathway_amd.engine.runtime import Runtime

        sinks = list(self.sinks) + list(extra_sinks or [])
        if not sinks:
            return None
        pm = None
        if persistence_config is not None:
            from pathway_amd.persistence.engine import PersistenceManager
            from pathway_amd.internals.config import pathway_config

            mode = getattr(persistence_config, "persistence_mode", None)
            if mode is not None and "udf_caching" in str(mode).lower():
                # UdfCaching: point the default DiskCache at the backend
                # (reference PersistenceMode::UdfCaching — deterministic
                # replay of non-deterministic UDFs)
                import os as _os

                backend = getattr(persistence_config, "backend", None)
                path = getattr(backend, "path", None)
                if path:
                    _os.environ["PATHWAY_PERSISTENT_STORAGE"] = str(path)
            pm = PersistenceManager(persistence_config, worker=pathway_config.process_id)
        rt = Runtime(sinks, device=get_device(), comm=self.comm, persistence=pm)
        http_server = None
        if kwargs.get("with_http_server"):
            from pathway_amd.engine.monitoring import start_http_server

            http_server = start_http_server(rt.stats)
        if monitoring is not None and str(monitoring) not in ("none", "MonitoringLevel.NONE"):
            from pathway_amd.engine.monitoring import ConsoleMonitor

            rt.monitor = ConsoleMonitor(rt.stats)
        reset_all(rt.nodes)
        rt.run()
        if self.services:
            # serving mode: keep the engine live for rest_connector traffic
            from pathway_amd.io.http import serve_forever

            servers = serve_forever(self.services, rt)
            if not kwargs.get("_serve_in_background"):
                import time as _time

                try:
                    while True:
                        _time.sleep(0.2)
                        rt.run()  # drain any streaming sources
                except KeyboardInterrupt:
                    pass
                finally:
                    for s_ in servers:
                        s_.shutdown()
        if pm is not None:
            pm.close()
        if http_server is not None:
            http_server.shutdown()
        return rt


G = RunGraph()


def reset_all(nodes) -> None:
    for n in nodes:
        reset = getattr(n, "reset", None)
        if reset is not None:
            reset()


def run(
    *,
    debug: bool = False,
    monitoring_level: Any = None,
    with_http_server: bool = False,
    default_logging: bool = True,
    persistence_config: Any = None,
    runtime_typechecking: bool | None = None,
    license_key: str | None = None,
    terminate_on_error: bool | None = None,
    **kwargs: Any,
):
    """pw.run(): execute every registered output (reference internals/run.py:13)."""
    return G.run_nodes(
        monitoring=monitoring_level,
        persistence_config=persistence_config,
        with_http_server=with_http_server,
        **kwargs,
    )


def run_all(**kwargs: Any):
    return run(**kwargs)
