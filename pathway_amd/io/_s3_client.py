"""Pure-python S3 REST client (path-style, SigV4).

Replaces the reference's rust-s3/aws-sdk dependency
(src/connectors/data_storage + src/persistence/backends/s3.rs) with a
from-scratch implementation of the S3 HTTP API: PutObject, GetObject,
DeleteObject, HeadObject, ListObjectsV2, CopyObject.  Speaks AWS
Signature Version 4, so it works against real S3/MinIO endpoints; the
in-process fake (tests/fakes/fake_s3.py) accepts any signature and
exercises the same request/XML paths.
"""

from __future__ import annotations

import datetime
import hashlib
import hmac
import urllib.error
import urllib.parse
import urllib.request
import xml.etree.ElementTree as ET
from dataclasses import dataclass
from typing import Any


class S3Error(RuntimeError):
    def __init__(self, status: int, body: str):
        super().__init__(f"S3 error {status}: {body[:300]}")
        self.status = status


@dataclass
class S3Object:
    key: str
    size: int
    etag: str
    last_modified: str


class S3Client:
    def __init__(
        self,
        endpoint: str,
        *,
        access_key: str = "",
        secret_key: str = "",
        region: str = "us-east-1",
        timeout: float = 30.0,
    ):
        self.endpoint = endpoint.rstrip("/")
        self.access_key = access_key
        self.secret_key = secret_key
        self.region = region
        self.timeout = timeout

    # -- SigV4 --

    def _sign(self, method: str, path: str, query: dict[str, str],
              headers: dict[str, str], payload: bytes) -> dict[str, str]:
        now = datetime.datetime.now(datetime.timezone.utc)
        amz_date = now.strftime("%Y%m%dT%H%M%SZ")
        datestamp = now.strftime("%Y%m%d")
        host = urllib.parse.urlparse(self.endpoint).netloc
        payload_hash = hashlib.sha256(payload).hexdigest()
        headers = {
            **headers,
            "host": host,
            "x-amz-date": amz_date,
            "x-amz-content-sha256": payload_hash,
        }
        if not self.access_key:
            return headers  # anonymous (fake endpoints)
        canonical_query = "&".join(
            f"{urllib.parse.quote(k, safe='')}={urllib.parse.quote(v, safe='')}"
            for k, v in sorted(query.items())
        )
        signed_names = sorted(headers.keys())
        canonical_headers = "".join(
            f"{k}:{headers[k].strip()}\n" for k in signed_names
        )
        signed_headers = ";".join(signed_names)
        canonical_request = "\n".join([
            method,
            urllib.parse.quote(path),
            canonical_query,
            canonical_headers,
            signed_headers,
            payload_hash,
        ])
        scope = f"{datestamp}/{self.region}/s3/aws4_request"
        string_to_sign = "\n".join([
            "AWS4-HMAC-SHA256",
            amz_date,
            scope,
            hashlib.sha256(canonical_request.encode()).hexdigest(),
        ])

        def _hmac(key: bytes, msg: str) -> bytes:
            return hmac.new(key, msg.encode(), hashlib.sha256).digest()

        k = _hmac(f"AWS4{self.secret_key}".encode(), datestamp)
        k = _hmac(k, self.region)
        k = _hmac(k, "s3")
        k = _hmac(k, "aws4_request")
        signature = hmac.new(k, string_to_sign.encode(), hashlib.sha256).hexdigest()
        headers["Authorization"] = (
            f"AWS4-HMAC-SHA256 Credential={self.access_key}/{scope}, "
            f"SignedHeaders={signed_headers}, Signature={signature}"
        )
        return headers

    def _request(self, method: str, bucket: str, key: str = "",
                 query: dict[str, str] | None = None,
                 body: bytes = b"",
                 extra_headers: dict[str, str] | None = None) -> tuple[int, bytes, dict]:
        query = query or {}
        path = f"/{bucket}" + (f"/{key}" if key else "")
        headers = self._sign(method, path, query, extra_headers or {}, body)
        qs = urllib.parse.urlencode(query)
        url = self.endpoint + urllib.parse.quote(path) + (f"?{qs}" if qs else "")
        req = urllib.request.Request(url, data=body if method in ("PUT", "POST") else None,
                                     method=method, headers=headers)
        try:
            with urllib.request.urlopen(req, timeout=self.timeout) as resp:
                return resp.status, resp.read(), dict(resp.headers)
        except urllib.error.HTTPError as e:
            if e.code == 404:
                return 404, e.read(), dict(e.headers)
            raise S3Error(e.code, e.read().decode("utf-8", "replace")) from e

    # -- operations --

    def put_object(self, bucket: str, key: str, data: bytes) -> None:
        status, body, _ = self._request("PUT", bucket, key, body=data)
        if status >= 300:
            raise S3Error(status, body.decode("utf-8", "replace"))

    def get_object(self, bucket: str, key: str) -> bytes | None:
        status, body, _ = self._request("GET", bucket, key)
        if status == 404:
            return None
        return body

    def head_object(self, bucket: str, key: str) -> dict | None:
        status, _, headers = self._request("HEAD", bucket, key)
        if status == 404:
            return None
        return headers

    def delete_object(self, bucket: str, key: str) -> None:
        self._request("DELETE", bucket, key)

    def copy_object(self, bucket: str, src_key: str, dst_key: str) -> None:
        status, body, _ = self._request(
            "PUT", bucket, dst_key,
            extra_headers={"x-amz-copy-source": f"/{bucket}/{src_key}"},
        )
        if status >= 300:
            raise S3Error(status, body.decode("utf-8", "replace"))

    def list_objects(self, bucket: str, prefix: str = "") -> list[S3Object]:
        out: list[S3Object] = []
        token: str | None = None
        while True:
            query = {"list-type": "2", "prefix": prefix}
            if token:
                query["continuation-token"] = token
            status, body, _ = self._request("GET", bucket, query=query)
            if status == 404:
                return out
            root = ET.fromstring(body)
            ns = ""
            if root.tag.startswith("{"):
                ns = root.tag.split("}")[0] + "}"
            for c in root.findall(f"{ns}Contents"):
                out.append(
                    S3Object(
                        key=c.findtext(f"{ns}Key"),
                        size=int(c.findtext(f"{ns}Size") or 0),
                        etag=(c.findtext(f"{ns}ETag") or "").strip('"'),
                        last_modified=c.findtext(f"{ns}LastModified") or "",
                    )
                )
            token = root.findtext(f"{ns}NextContinuationToken")
            if not token:
                return out


def client_from_settings(settings: Any) -> tuple[S3Client, str]:
    """Build (client, bucket) from an AwsS3Settings-style object/dict."""
    if isinstance(settings, dict):
        d = settings
    else:
        d = {
            "bucket_name": getattr(settings, "bucket_name", None),
            "endpoint": getattr(settings, "endpoint", None),
            "access_key": getattr(settings, "access_key", ""),
            "secret_access_key": getattr(settings, "secret_access_key", ""),
            "region": getattr(settings, "region", "us-east-1"),
        }
    endpoint = d.get("endpoint") or f"https://s3.{d.get('region', 'us-east-1')}.amazonaws.com"
    client = S3Client(
        endpoint,
        access_key=d.get("access_key") or "",
        secret_key=d.get("secret_access_key") or d.get("secret_key") or "",
        region=d.get("region") or "us-east-1",
    )
    return client, d.get("bucket_name") or d.get("bucket") or ""
