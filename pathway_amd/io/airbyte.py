"""pw.io.airbyte (reference io/airbyte + vendored airbyte_serverless)."""
from __future__ import annotations

from typing import Any


def read(config_file_path: str, streams: list[str], *, mode: str = "streaming",
         refresh_interval_ms: int = 60000, name: str | None = None, **kwargs: Any):
    raise NotImplementedError(
        "pw.io.airbyte needs docker or an airbyte source binary (offline image)"
    )
