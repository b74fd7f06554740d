"""Object store abstraction — the reference's local-or-S3 layer.

The reference routes every checkpoint/bulk artifact through boto3-S3 or the
local FS behind one function pair (``list_objects`` local-or-S3,
photon/server/s3_utils.py:114-155; upload/download at :275,480,551). On an
MI355X node the bulk PARAMETER path is RCCL (fed/runtime.py) — the store is
only for checkpoints and run artifacts, so the interface is small:

    store.list(prefix)      -> sorted relative keys
    store.upload(src, key)  / store.download(key, dst)
    store.open_read(key)    / store.write_bytes(key, data)
    store.exists(key)       / store.delete(prefix)

``LocalStore`` is the default; ``S3Store`` keeps the comm_stack.s3 config
surface alive and activates when boto3 + endpoint credentials exist (this
container has no network, so it stays a configured-but-inactive backend,
exactly like the reference run with ``comm_stack.shm=true``).
"""

from __future__ import annotations

import shutil
from pathlib import Path


class LocalStore:
    def __init__(self, root: str | Path):
        self.root = Path(root)

    def _p(self, key: str) -> Path:
        return self.root / key

    def list(self, prefix: str = "") -> list[str]:
        base = self._p(prefix)
        if not base.exists():
            return []
        return sorted(
            str(p.relative_to(self.root))
            for p in base.rglob("*")
            if p.is_file()
        )

    def upload(self, src: str | Path, key: str) -> None:
        dst = self._p(key)
        dst.parent.mkdir(parents=True, exist_ok=True)
        shutil.copy2(src, dst)

    def download(self, key: str, dst: str | Path) -> None:
        Path(dst).parent.mkdir(parents=True, exist_ok=True)
        shutil.copy2(self._p(key), dst)

    def write_bytes(self, key: str, data: bytes) -> None:
        dst = self._p(key)
        dst.parent.mkdir(parents=True, exist_ok=True)
        dst.write_bytes(data)

    def read_bytes(self, key: str) -> bytes:
        return self._p(key).read_bytes()

    def exists(self, key: str) -> bool:
        return self._p(key).exists()

    def delete(self, prefix: str) -> None:
        p = self._p(prefix)
        if p.is_dir():
            shutil.rmtree(p)
        elif p.exists():
            p.unlink()


class S3Store:
    """S3 backend keeping the reference's S3CommConfig surface
    (photon/conf/base_schema.py:265-281). Requires boto3 + credentials."""

    def __init__(self, bucket: str, endpoint_url: str | None = None, prefix: str = ""):
        import boto3  # optional dependency; no network in CI

        self.bucket = bucket
        self.prefix = prefix.rstrip("/")
        self.client = boto3.client("s3", endpoint_url=endpoint_url)

    def _k(self, key: str) -> str:
        return f"{self.prefix}/{key}" if self.prefix else key

    def list(self, prefix: str = "") -> list[str]:
        keys = []
        paginator = self.client.get_paginator("list_objects_v2")
        for page in paginator.paginate(Bucket=self.bucket, Prefix=self._k(prefix)):
            for obj in page.get("Contents", []):
                k = obj["Key"]
                if self.prefix:
                    k = k[len(self.prefix) + 1 :]
                keys.append(k)
        return sorted(keys)

    def upload(self, src, key: str) -> None:
        self.client.upload_file(str(src), self.bucket, self._k(key))

    def download(self, key: str, dst) -> None:
        Path(dst).parent.mkdir(parents=True, exist_ok=True)
        self.client.download_file(self.bucket, self._k(key), str(dst))

    def write_bytes(self, key: str, data: bytes) -> None:
        self.client.put_object(Bucket=self.bucket, Key=self._k(key), Body=data)

    def read_bytes(self, key: str) -> bytes:
        return self.client.get_object(Bucket=self.bucket, Key=self._k(key))["Body"].read()

    def exists(self, key: str) -> bool:
        try:
            self.client.head_object(Bucket=self.bucket, Key=self._k(key))
            return True
        except Exception:
            return False

    def delete(self, prefix: str) -> None:
        for k in self.list(prefix):
            self.client.delete_object(Bucket=self.bucket, Key=self._k(k))


def get_store(cfg: dict):
    """Build the store from the photon config: comm_stack.s3 + s3_comm config
    select S3; anything else is the local saving_path."""
    comm = (cfg.get("comm_stack") or {}) if isinstance(cfg, dict) else {}
    if comm.get("s3"):
        s3 = cfg.get("s3_comm") or {}
        return S3Store(
            bucket=str(s3.get("bucket_name", "photon")),
            endpoint_url=s3.get("endpoint_url"),
            prefix=str(s3.get("prefix", "")),
        )
    photon = cfg.get("photon") or {}
    return LocalStore(photon.get("saving_path") or "checkpoints")
