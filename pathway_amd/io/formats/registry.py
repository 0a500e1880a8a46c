"""Confluent schema-registry REST client (reference avro.rs schema registry).

Speaks the standard registry HTTP API:

  GET  /schemas/ids/{id}              -> {"schema": "<json string>"}
  POST /subjects/{subject}/versions   -> {"id": N}
  GET  /subjects/{subject}/versions/latest -> {"id", "schema", ...}

Schemas are cached by id and by subject.  Works against any registry
implementation, including the in-process fake used by the tests
(tests/fakes/fake_registry.py).
"""

from __future__ import annotations

import json
import urllib.request
from typing import Any

CONTENT_TYPE = "application/vnd.schemaregistry.v1+json"


class SchemaRegistryClient:
    def __init__(self, url: str, *, timeout: float = 10.0,
                 headers: dict[str, str] | None = None):
        self.url = url.rstrip("/")
        self.timeout = timeout
        self.headers = headers or {}
        self._by_id: dict[int, Any] = {}
        self._by_subject: dict[str, tuple[int, Any]] = {}

    def _request(self, method: str, path: str, body: dict | None = None) -> Any:
        req = urllib.request.Request(
            self.url + path,
            data=json.dumps(body).encode() if body is not None else None,
            method=method,
            headers={"Content-Type": CONTENT_TYPE, **self.headers},
        )
        with urllib.request.urlopen(req, timeout=self.timeout) as resp:
            return json.loads(resp.read())

    def get_schema(self, schema_id: int) -> Any:
        """Parsed Avro schema for a registry id (cached)."""
        cached = self._by_id.get(schema_id)
        if cached is not None:
            return cached
        out = self._request("GET", f"/schemas/ids/{schema_id}")
        schema = json.loads(out["schema"])
        self._by_id[schema_id] = schema
        return schema

    def register(self, subject: str, schema: Any) -> int:
        """Register a schema under a subject; returns the schema id."""
        out = self._request(
            "POST", f"/subjects/{subject}/versions",
            {"schema": json.dumps(schema)},
        )
        sid = out["id"]
        self._by_id[sid] = schema
        self._by_subject[subject] = (sid, schema)
        return sid

    def latest(self, subject: str) -> tuple[int, Any]:
        """(id, schema) of the subject's latest version (cached)."""
        cached = self._by_subject.get(subject)
        if cached is not None:
            return cached
        out = self._request("GET", f"/subjects/{subject}/versions/latest")
        sid = out["id"]
        schema = json.loads(out["schema"])
        self._by_id[sid] = schema
        self._by_subject[subject] = (sid, schema)
        return sid, schema
