"""SharePoint connector (reference xpacks/connectors/sharepoint surface)."""
from __future__ import annotations

from typing import Any


def read(
    url: str,
    *,
    tenant: str | None = None,
    client_id: str | None = None,
    cert_path: str | None = None,
    thumbprint: str | None = None,
    root_path: str | None = None,
    mode: str = "streaming",
    with_metadata: bool = False,
    refresh_interval: int = 30,
    **kwargs: Any,
):
    raise RuntimeError(
        "pw.xpacks.connectors.sharepoint needs Office365 API access, "
        "unavailable in this offline environment"
    )
