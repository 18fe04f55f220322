"""Payload blob stores.

Role parity with the reference's Store interface + S3/File backends
(reference: pkg/storage/store.go:26-42, s3_store.go, file_store.go), with
the MI355X-native tier order: device HBM (torch tensors, 288 GB/GPU) →
pinned host memory → local disk.  There is no S3 in-box; the FileStore
covers the durable tier.
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
