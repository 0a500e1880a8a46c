"""pw.io.slack (reference io/slack) — API-parity surface.

Requires the slack_sdk client library (offline image: raises at call time).
"""
from __future__ import annotations

from typing import Any

from pathway_amd.io._utils import require_client


def send_alerts(table, *args: Any, name: str | None = None, **kwargs: Any):
    require_client("slack", "slack_sdk")
    raise NotImplementedError("pw.io.slack.send_alerts: client library loaded but offline transport is unavailable in this environment")


def write(table, *args: Any, name: str | None = None, **kwargs: Any):
    require_client("slack", "slack_sdk")
    raise NotImplementedError("pw.io.slack.write: client library loaded but offline transport is unavailable in this environment")
