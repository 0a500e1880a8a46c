"""Tiny JSON-over-HTTP helper shared by the REST-API connectors.

urllib-based (no external client libraries): one call = one request with
JSON (or raw) body, JSON reply, basic auth/token headers, retries.
"""

from __future__ import annotations

import json
import time
import urllib.error
import urllib.request
from typing import Any


class RestError(RuntimeError):
    def __init__(self, status: int, body: str):
        super().__init__(f"HTTP {status}: {body[:300]}")
        self.status = status


def request(
    method: str,
    url: str,
    *,
    body: Any = None,
    raw_body: bytes | None = None,
    headers: dict[str, str] | None = None,
    timeout: float = 30.0,
    retries: int = 2,
    content_type: str = "application/json",
) -> Any:
    data = raw_body
    if data is None and body is not None:
        data = json.dumps(body, default=str).encode()
    hdrs = {"Content-Type": content_type, **(headers or {})}
    last: Exception | None = None
    for attempt in range(retries + 1):
        req = urllib.request.Request(url, data=data, method=method, headers=hdrs)
        try:
            with urllib.request.urlopen(req, timeout=timeout) as resp:
                payload = resp.read()
                if not payload:
                    return None
                ctype = resp.headers.get("Content-Type", "")
                if "json" in ctype:
                    return json.loads(payload)
                return payload
        except urllib.error.HTTPError as e:
            raise RestError(e.code, e.read().decode("utf-8", "replace")) from e
        except (urllib.error.URLError, TimeoutError) as e:
            last = e
            if attempt < retries:
                time.sleep(0.2 * (attempt + 1))
    raise RestError(0, f"connection failed: {last}")
