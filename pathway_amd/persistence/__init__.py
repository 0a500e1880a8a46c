"""pw.persistence (reference python/pathway/persistence/__init__.py:13-230).

Config/Backend surface; checkpoint/recovery engine wiring lands with the
persistence phase.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any


class Backend:
    @classmethod
    def filesystem(cls, path: str) -> "Backend":
        b = cls()
        b.kind = "filesystem"
        b.path = path
        return b

    @classmethod
    def s3(cls, root_path: str, bucket_settings: Any = None) -> "Backend":
        b = cls()
        b.kind = "s3"
        b.path = root_path
        b.settings = bucket_settings
        return b

    @classmethod
    def azure(cls, root_path: str, account: Any = None, *,
              account_url: str | None = None, container: str | None = None,
              sas_token: str | None = None, **kw) -> "Backend":
        b = cls()
        b.kind = "azure"
        b.path = root_path
        b.account_url = account_url or (
            getattr(account, "account_url", None) if account else None
        ) or ""
        b.container = container or (
            getattr(account, "container", None) if account else None
        ) or ""
        b.sas_token = sas_token
        return b

    @classmethod
    def mock(cls, events: Any = None) -> "Backend":
        b = cls()
        b.kind = "mock"
        b.path = None
        return b


@dataclass
class Config:
    backend: Backend | None = None
    snapshot_interval_ms: int = 0
    persistence_mode: Any = None
    snapshot_access: Any = None
    continue_after_replay: bool = True

    @classmethod
    def simple_config(cls, backend: Backend, **kwargs) -> "Config":
        return cls(backend=backend, **kwargs)


def get_persistence_engine_config(config):
    """Engine-facing view of a persistence Config (reference
    persistence/__init__.py helper)."""
    return config
