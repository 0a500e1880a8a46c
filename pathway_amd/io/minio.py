"""pw.io.minio — MinIO connector (reference io/minio).

MinIO speaks the S3 API; this wraps pw.io.s3 with MinIO-style settings.
"""

from __future__ import annotations

from typing import Any

from pathway_amd.io import s3 as _s3


class MinIOSettings:
    def __init__(
        self,
        endpoint: str | None = None,
        bucket_name: str | None = None,
        access_key: str | None = None,
        secret_access_key: str | None = None,
        *,
        with_path_style: bool = True,
        region: str | None = None,
        **kw: Any,
    ):
        self.endpoint = endpoint
        self.bucket_name = bucket_name
        self.access_key = access_key
        self.secret_access_key = secret_access_key
        self.with_path_style = with_path_style
        self.region = region

    def create_aws_settings(self) -> _s3.AwsS3Settings:
        endpoint = self.endpoint
        if endpoint and "://" not in endpoint:
            endpoint = f"https://{endpoint}"
        return _s3.AwsS3Settings(
            bucket_name=self.bucket_name,
            access_key=self.access_key,
            secret_access_key=self.secret_access_key,
            region=self.region,
            endpoint=endpoint,
        )


def read(path: str, minio_settings: MinIOSettings, *, format: str = "plaintext",
         **kwargs: Any):
    return _s3.read(
        path, aws_s3_settings=minio_settings.create_aws_settings(),
        format=format, **kwargs,
    )


def write(table, path: str, minio_settings: MinIOSettings, *, format: str = "json",
          **kwargs: Any):
    return _s3.write(
        table, path, aws_s3_settings=minio_settings.create_aws_settings(),
        format=format, **kwargs,
    )
