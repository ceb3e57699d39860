"""AWS S3 artefact store (optional backend).

Wire-level parity with the reference's boto3 usage
(``stage_1_train_model.py:59-71``: ``list_objects`` by prefix,
``get_object``/``upload_file``).  boto3 is not bundled with this image, so
the import is deferred; constructing an :class:`S3Store` without boto3
raises a clear error, mirroring the reference's fail-hard behaviour on
missing AWS credentials (``stage_1:123-125``).
"""
from __future__ import annotations

from bodywork_mlops_demo_amd.store.base import ArtefactStore


class S3Store(ArtefactStore):
    def __init__(self, bucket: str):
        try:
            import boto3  # type: ignore
        except ImportError as e:  # pragma: no cover - env without boto3
            raise RuntimeError(
                "S3Store requires boto3 (pip install boto3) and AWS credentials"
            ) from e
        self.bucket = bucket
        self._s3 = boto3.client("s3")

    @property
    def uri(self) -> str:
        return f"s3://{self.bucket}"

    def list_keys(self, prefix: str) -> list[str]:
        keys: list[str] = []
        token = None
        while True:
            kw = {"Bucket": self.bucket, "Prefix": prefix}
            if token:
                kw["ContinuationToken"] = token
            resp = self._s3.list_objects_v2(**kw)
            keys += [o["Key"] for o in resp.get("Contents", [])]
            if not resp.get("IsTruncated"):
                return sorted(keys)
            token = resp["NextContinuationToken"]

    def get_bytes(self, key: str) -> bytes:
        return self._s3.get_object(Bucket=self.bucket, Key=key)["Body"].read()

    def put_bytes(self, key: str, data: bytes) -> None:
        self._s3.put_object(Bucket=self.bucket, Key=key, Body=data)

    def exists(self, key: str) -> bool:
        try:
            self._s3.head_object(Bucket=self.bucket, Key=key)
            return True
        except Exception:
            return False

    def delete(self, key: str) -> None:
        self._s3.delete_object(Bucket=self.bucket, Key=key)
