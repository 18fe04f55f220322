"""Payload blob stores.

Role parity with the reference's Store interface + S3/File backends
(reference: pkg/storage/store.go:26-42, s3_store.go, file_store.go), with
the MI355X-native tier order: device HBM (torch tensors, 288 GB/GPU) →
pinned host memory → local disk → S3-compatible remote (durable/off-node
tier, SigV4 over stdlib HTTPS — S3Store below).
"""
from __future__ import annotations

import json
import os
import threading
import time
import typing as _t


class StoreError(RuntimeError):
    pass


class BlobNotFound(StoreError):
    pass


class Store:
    """Abstract blob store: bytes in, bytes out, namespaced by key."""

    name = "abstract"

    def write(self, key: str, data: bytes) -> None:
        raise NotImplementedError

    def read(self, key: str) -> bytes:
        raise NotImplementedError

    def delete(self, key: str) -> None:
        raise NotImplementedError

    def list(self, prefix: str = "") -> _t.List[str]:
        raise NotImplementedError

    def exists(self, key: str) -> bool:
        try:
            self.read(key)
            return True
        except BlobNotFound:
            return False


class MemStore(Store):
    """In-memory store (test double; reference: pkg/storage/store_mock.go)."""

    name = "mem"

    def __init__(self):
        self._data: _t.Dict[str, _t.Tuple[bytes, float]] = {}
        self._lock = threading.Lock()

    def write(self, key: str, data: bytes) -> None:
        with self._lock:
            self._data[key] = (bytes(data), time.time())

    def read(self, key: str) -> bytes:
        with self._lock:
            item = self._data.get(key)
        if item is None:
            raise BlobNotFound(key)
        return item[0]

    def delete(self, key: str) -> None:
        with self._lock:
            self._data.pop(key, None)

    def list(self, prefix: str = "") -> _t.List[str]:
        with self._lock:
            return sorted(k for k in self._data if k.startswith(prefix))

    def mtime(self, key: str) -> _t.Optional[float]:
        with self._lock:
            item = self._data.get(key)
        return item[1] if item else None


class FileStore(Store):
    """Local-disk store (reference: pkg/storage/file_store.go)."""

    name = "file"

    def __init__(self, root: str):
        self.root = os.path.abspath(root)
        os.makedirs(self.root, exist_ok=True)

    def _path(self, key: str) -> str:
        safe = key.lstrip("/")
        path = os.path.abspath(os.path.join(self.root, safe))
        if not path.startswith(self.root + os.sep) and path != self.root:
            raise StoreError(f"key {key!r} escapes the store root")
        return path

    def write(self, key: str, data: bytes) -> None:
        path = self._path(key)
        os.makedirs(os.path.dirname(path), exist_ok=True)
        tmp = path + ".tmp"
        with open(tmp, "wb") as fh:
            fh.write(data)
        os.replace(tmp, path)

    def read(self, key: str) -> bytes:
        try:
            with open(self._path(key), "rb") as fh:
                return fh.read()
        except FileNotFoundError:
            raise BlobNotFound(key) from None

    def delete(self, key: str) -> None:
        try:
            os.remove(self._path(key))
        except FileNotFoundError:
            pass

    def list(self, prefix: str = "") -> _t.List[str]:
        out = []
        for dirpath, _dirs, files in os.walk(self.root):
            for f in files:
                if f.endswith(".tmp"):
                    continue
                rel = os.path.relpath(os.path.join(dirpath, f), self.root)
                key = rel.replace(os.sep, "/")
                if key.startswith(prefix):
                    out.append(key)
        return sorted(out)

    def mtime(self, key: str) -> _t.Optional[float]:
        try:
            return os.path.getmtime(self._path(key))
        except OSError:
            return None


class TensorStore:
    """Device-resident payload table: step-edge tensors stay in HBM.

    This replaces the reference's S3 offload for the hot path (SURVEY.md
    §2.6): a `$storageRef` of kind "tensor" is an indirection to a tensor
    kept on-device; `spill()` demotes cold tensors to pinned host memory,
    and `drop_to(file_store)` demotes further to disk via safetensors-free
    raw serialization (torch.save).
    """

    name = "tensor"

    def __init__(self, device: _t.Optional[str] = None, capacity_bytes: _t.Optional[int] = None):
        self._table: _t.Dict[str, _t.Any] = {}
        self._meta: _t.Dict[str, dict] = {}
        self._lock = threading.Lock()
        self.device = device
        self.capacity_bytes = capacity_bytes
        self._bytes = 0

    @staticmethod
    def _nbytes(tensor) -> int:
        return tensor.numel() * tensor.element_size()

    def put(self, key: str, tensor) -> dict:
        with self._lock:
            old = self._table.pop(key, None)
            if old is not None:
                self._bytes -= self._nbytes(old)
            self._table[key] = tensor
            size = self._nbytes(tensor)
            self._bytes += size
            meta = {
                "key": key,
                "kind": "tensor",
                "dtype": str(tensor.dtype).replace("torch.", ""),
                "shape": list(tensor.shape),
                "device": str(tensor.device),
                "size": size,
            }
            self._meta[key] = meta
            return dict(meta)

    def get(self, key: str):
        with self._lock:
            t = self._table.get(key)
        if t is None and key.startswith("native/"):
            # payload held by the GIL-free native engram lane
            # (csrc/hip/native_engrams.cpp registry)
            try:
                from bobrapet_amd import _hipops

                t = _hipops.native_tensor_get(key)
            except ImportError:
                t = None
        if t is None:
            raise BlobNotFound(key)
        return t

    def delete(self, key: str) -> None:
        with self._lock:
            t = self._table.pop(key, None)
            self._meta.pop(key, None)
            if t is not None:
                self._bytes -= self._nbytes(t)

    def list(self, prefix: str = "") -> _t.List[str]:
        with self._lock:
            return sorted(k for k in self._table if k.startswith(prefix))

    @property
    def used_bytes(self) -> int:
        return self._bytes

    def spill(self, key: str) -> None:
        """Demote one tensor to pinned host memory (kept addressable)."""
        import torch

        with self._lock:
            t = self._table.get(key)
            if t is None or not t.is_cuda:
                return
            host = torch.empty_like(t, device="cpu", pin_memory=True)
            host.copy_(t, non_blocking=False)
            self._table[key] = host
            self._meta[key]["device"] = "cpu-pinned"

    def promote(self, key: str, device) -> None:
        """Bring a spilled tensor back to the device."""
        with self._lock:
            t = self._table.get(key)
            if t is None or t.is_cuda:
                return
            self._table[key] = t.to(device, non_blocking=False)
            self._meta[key]["device"] = str(device)


class S3Store(Store):
    """S3-compatible remote/durable blob tier (reference:
    pkg/storage/s3_store.go:45-184 — AWS SDK v2, path-style addressing,
    SSE/KMS headers, bounded retries).

    MI355X-native rebuild: plain HTTPS + SigV4 signed with the stdlib
    (hmac/hashlib/urllib) — no SDK dependency, works against any
    S3-compatible endpoint (MinIO, Ceph RGW, AWS).  The hot path never
    touches this tier (tensors stay in HBM; the File tier handles
    single-node spill) — this is the durable/off-node tier the reference
    ships for checkpoints and cross-cluster payloads.
    """

    name = "s3"

    def __init__(
        self,
        bucket: str,
        endpoint: str,
        region: str = "us-east-1",
        access_key: str = "",
        secret_key: str = "",
        prefix: str = "",
        sse: _t.Optional[str] = None,  # e.g. "AES256" or "aws:kms"
        sse_kms_key_id: _t.Optional[str] = None,
        retries: int = 3,
        timeout: float = 10.0,
    ):
        self.bucket = bucket
        self.endpoint = endpoint.rstrip("/")
        self.region = region
        self.access_key = access_key
        self.secret_key = secret_key
        self.prefix = prefix.strip("/")
        self.sse = sse
        self.sse_kms_key_id = sse_kms_key_id
        self.retries = max(1, retries)
        self.timeout = timeout

    # ---- SigV4 --------------------------------------------------------

    def _sign(self, method: str, path: str, query: str, headers: dict, payload_hash: str):
        import datetime
        import hashlib
        import hmac

        t = datetime.datetime.utcnow()
        amz_date = t.strftime("%Y%m%dT%H%M%SZ")
        datestamp = t.strftime("%Y%m%d")
        headers = dict(headers)
        headers["x-amz-date"] = amz_date
        headers["x-amz-content-sha256"] = payload_hash
        signed = sorted(k.lower() for k in headers)
        canonical_headers = "".join(f"{k}:{headers[_orig(k, headers)].strip()}\n" for k in signed)
        canonical = "\n".join(
            [method, path, query, canonical_headers, ";".join(signed), payload_hash]
        )
        scope = f"{datestamp}/{self.region}/s3/aws4_request"
        string_to_sign = "\n".join(
            [
                "AWS4-HMAC-SHA256",
                amz_date,
                scope,
                hashlib.sha256(canonical.encode()).hexdigest(),
            ]
        )

        def hm(key, msg):
            return hmac.new(key, msg.encode(), hashlib.sha256).digest()

        k = hm(("AWS4" + self.secret_key).encode(), datestamp)
        k = hm(k, self.region)
        k = hm(k, "s3")
        k = hm(k, "aws4_request")
        sig = hmac.new(k, string_to_sign.encode(), hashlib.sha256).hexdigest()
        headers["Authorization"] = (
            f"AWS4-HMAC-SHA256 Credential={self.access_key}/{scope}, "
            f"SignedHeaders={';'.join(signed)}, Signature={sig}"
        )
        return headers

    def _request(self, method: str, key: str = "", query: str = "", data: bytes = b"", sse: bool = False):
        import hashlib
        import urllib.error
        import urllib.request

        full_key = f"{self.prefix}/{key}" if self.prefix and key else (key or "")
        path = f"/{self.bucket}/{full_key}" if full_key else f"/{self.bucket}"
        url = f"{self.endpoint}{path}" + (f"?{query}" if query else "")
        host = self.endpoint.split("://", 1)[1]
        payload_hash = hashlib.sha256(data).hexdigest()
        headers = {"host": host}
        if sse and self.sse:
            headers["x-amz-server-side-encryption"] = self.sse
            if self.sse == "aws:kms" and self.sse_kms_key_id:
                headers["x-amz-server-side-encryption-aws-kms-key-id"] = self.sse_kms_key_id
        headers = self._sign(method, path, query, headers, payload_hash)
        headers.pop("host", None)  # urllib sets it
        last_exc: _t.Optional[Exception] = None
        for attempt in range(self.retries):
            req = urllib.request.Request(url, data=data if method in ("PUT", "POST") else None,
                                         headers=headers, method=method)
            try:
                with urllib.request.urlopen(req, timeout=self.timeout) as resp:
                    return resp.status, dict(resp.headers), resp.read()
            except urllib.error.HTTPError as exc:
                if exc.code == 404:
                    raise BlobNotFound(key) from None
                if exc.code < 500 or attempt == self.retries - 1:
                    raise StoreError(f"s3 {method} {key}: HTTP {exc.code}") from exc
                last_exc = exc
            except urllib.error.URLError as exc:
                if attempt == self.retries - 1:
                    raise StoreError(f"s3 {method} {key}: {exc}") from exc
                last_exc = exc
            import time as _time

            _time.sleep(0.1 * (2 ** attempt))
        raise StoreError(f"s3 {method} {key}: {last_exc}")

    # ---- Store interface ---------------------------------------------

    def write(self, key: str, data: bytes) -> None:
        self._request("PUT", key, data=data, sse=True)

    def read(self, key: str) -> bytes:
        _, _, body = self._request("GET", key)
        return body

    def delete(self, key: str) -> None:
        try:
            self._request("DELETE", key)
        except BlobNotFound:
            pass

    def list(self, prefix: str = "") -> _t.List[str]:
        import re
        import urllib.parse

        full = f"{self.prefix}/{prefix}" if self.prefix else prefix
        q = f"list-type=2&prefix={urllib.parse.quote(full, safe='')}"
        _, _, body = self._request("GET", "", query=q)
        keys = re.findall(rb"<Key>([^<]+)</Key>", body)
        out = []
        for k in keys:
            s = k.decode()
            if self.prefix and s.startswith(self.prefix + "/"):
                s = s[len(self.prefix) + 1 :]
            out.append(s)
        return sorted(out)

    def mtime(self, key: str) -> _t.Optional[float]:
        import email.utils

        try:
            _, headers, _ = self._request("HEAD", key)
        except (BlobNotFound, StoreError):
            return None
        lm = headers.get("Last-Modified")
        if not lm:
            return None
        try:
            return email.utils.parsedate_to_datetime(lm).timestamp()
        except (TypeError, ValueError):
            return None


def _orig(lower: str, headers: dict) -> str:
    for k in headers:
        if k.lower() == lower:
            return k
    return lower


def store_from_env(env: _t.Optional[_t.Mapping[str, str]] = None) -> Store:
    """Build the blob tier from the BUBU_STORAGE_* env contract
    (reference: pkg/storage/manager.go NewManager 256-308 — provider
    s3|file|none selected purely from env)."""
    import os as _os

    e = env if env is not None else _os.environ
    provider = (e.get("BUBU_STORAGE_PROVIDER") or "mem").lower()
    if provider == "s3":
        return S3Store(
            bucket=e.get("BUBU_STORAGE_S3_BUCKET", "bobrapet"),
            endpoint=e.get("BUBU_STORAGE_S3_ENDPOINT", "https://s3.amazonaws.com"),
            region=e.get("BUBU_STORAGE_S3_REGION", "us-east-1"),
            access_key=e.get("BUBU_STORAGE_S3_ACCESS_KEY", ""),
            secret_key=e.get("BUBU_STORAGE_S3_SECRET_KEY", ""),
            prefix=e.get("BUBU_STORAGE_PATH", ""),
            sse=e.get("BUBU_STORAGE_S3_SSE") or None,
            sse_kms_key_id=e.get("BUBU_STORAGE_S3_KMS_KEY_ID") or None,
        )
    if provider == "file":
        return FileStore(e.get("BUBU_STORAGE_PATH", "/tmp/bobrapet-store"))
    return MemStore()
