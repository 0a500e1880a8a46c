"""pw.io.slack — Slack alert sink over chat.postMessage.

Reference: python/pathway/io/slack (send_alerts).  Posts one
chat.postMessage per row with a bearer token.
"""

from __future__ import annotations

from typing import Any

from pathway_amd.io import _rest

DEFAULT_BASE = "https://slack.com/api"


def send_alerts(
    alerts,
    slack_channel_id: str,
    slack_token: str,
    *,
    base_url: str = DEFAULT_BASE,
    name: str | None = None,
    **kwargs: Any,
):
    from pathway_amd.engine.runtime import OutputNode
    from pathway_amd.internals.config import get_device
    from pathway_amd.internals.rungraph import G

    names = alerts.column_names()

    def writer(batch):
        for _key, values, time, diff in batch.rows():
            if diff <= 0:
                continue
            text = str(values[0]) if len(names) == 1 else str(
                dict(zip(names, values))
            )
            out = _rest.request(
                "POST", f"{base_url}/chat.postMessage",
                body={"channel": slack_channel_id, "text": text},
                headers={"Authorization": f"Bearer {slack_token}"},
            )
            if out and out.get("ok") is False:
                raise RuntimeError(f"slack error: {out.get('error')}")

    node = OutputNode(alerts._node, writer, get_device())
    G.add_sink(node)
    return node


write = send_alerts
