"""pw.io.logstash — Logstash HTTP-input sink (reference io/logstash).

Posts one JSON document per row to a Logstash ``http`` input plugin
endpoint; retries per the reference's wrapper semantics.
"""

from __future__ import annotations

import json
from typing import Any

from pathway_amd.io import _rest


def write(
    table,
    endpoint: str,
    n_retries: int = 0,
    retry_policy: Any = None,
    connect_timeout_ms: int | None = None,
    request_timeout_ms: int | None = None,
    *,
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    names = table.column_names()
    timeout = (request_timeout_ms or 30000) / 1000.0

    def writer(batch):
        for _key, values, time, diff in batch.rows():
            rec = dict(zip(names, values))
            rec["time"] = time
            rec["diff"] = diff
            _rest.request(
                "POST", endpoint, body=rec,
                timeout=timeout, retries=n_retries,
            )

    node = OutputNode(table._node, writer, get_device())
    G.add_sink(node)
    return node
