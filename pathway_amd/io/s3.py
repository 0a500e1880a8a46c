"""pw.io.s3 (reference io/s3) — API-parity surface.

Requires the boto3 client library (offline image: raises at call time).
"""
from __future__ import annotations

from typing import Any

from pathway_amd.io._utils import require_client


def read(*args: Any, schema=None, mode: str = "streaming", name: str | None = None, autocommit_duration_ms: int | None = 1500, **kwargs: Any):
    require_client("boto3", "s3")
    raise NotImplementedError("pw.io.s3.read: client library loaded but offline transport is unavailable in this environment")


def write(table, *args: Any, name: str | None = None, **kwargs: Any):
    require_client("boto3", "s3")
    raise NotImplementedError("pw.io.s3.write: client library loaded but offline transport is unavailable in this environment")


class DigitalOceanS3Settings:
    """DigitalOcean Spaces credentials (reference io/s3)."""

    def __init__(self, bucket_name=None, *, access_key=None, secret_access_key=None, region=None, **kw):
        self.bucket_name = bucket_name
        self.access_key = access_key
        self.secret_access_key = secret_access_key
        self.region = region


class WasabiS3Settings:
    """Wasabi credentials (reference io/s3)."""

    def __init__(self, bucket_name=None, *, access_key=None, secret_access_key=None, region=None, **kw):
        self.bucket_name = bucket_name
        self.access_key = access_key
        self.secret_access_key = secret_access_key
        self.region = region


def read_from_digital_ocean(path, do_s3_settings, format, **kwargs):
    return read(path, format=format, **kwargs)


def read_from_wasabi(path, wasabi_s3_settings, format, **kwargs):
    return read(path, format=format, **kwargs)
