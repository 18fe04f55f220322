"""Payload offload engine: inline vs `$storageRef` indirection.

Role parity with the reference's StorageManager
(reference: pkg/storage/manager.go:177-966 — Hydrate/Dehydrate/
DehydrateInputs, the `$storageRef` walk, sub-path extraction DSL
path.go:23-139, ref-path validation 518-547, retention sweeper
retention.go:27-70), redesigned for the MI355X tiers: JSON payloads offload
to a byte store (mem/file), torch tensors stay resident in the HBM
TensorStore and travel by reference.
"""
from __future__ import annotations

import json
import re
import threading
import time
import typing as _t
import uuid

from .stores import BlobNotFound, FileStore, MemStore, Store, StoreError, TensorStore

STORAGE_REF_KEY = "$storageRef"
ENV_REF_KEY = "$envRef"
FILE_REF_KEY = "$fileRef"
DEFAULT_MAX_INLINE = 8 << 10  # bytes of JSON before a value offloads
MAX_WALK_DEPTH = 64

_KEY_RE = re.compile(r"^[A-Za-z0-9_\-./]+$")


def _is_tensor(value) -> bool:
    t = type(value)
    return t.__module__ == "torch" and t.__name__ in ("Tensor", "Parameter")


class RefError(StoreError):
    pass


class StorageManager:
    """Walks JSON-like payloads, inlining small values and offloading large
    ones behind `$storageRef` markers; hydrate reverses.  Torch tensors
    always go by reference (they are not JSON)."""

    def __init__(
        self,
        store: _t.Optional[Store] = None,
        tensors: _t.Optional[TensorStore] = None,
        max_inline_size: int = DEFAULT_MAX_INLINE,
        input_prefix: str = "inputs",
        output_prefix: str = "outputs",
    ):
        self.store = store or MemStore()
        self.tensors = tensors or TensorStore()
        self.max_inline_size = max_inline_size
        self.input_prefix = input_prefix
        self.output_prefix = output_prefix
        self._lock = threading.Lock()

    # ------------------------------------------------------------------
    # dehydrate: value → inline-or-ref
    # ------------------------------------------------------------------

    def dehydrate(self, value, prefix: _t.Optional[str] = None, *, _depth: int = 0):
        """Replace oversized / tensor leaves with `$storageRef` markers.

        Walks dicts/lists; a leaf (or subtree) whose JSON encoding exceeds
        max_inline_size is written to the blob store (reference:
        manager.go:465-517)."""
        prefix = prefix or self.output_prefix
        if _depth > MAX_WALK_DEPTH:
            raise RefError("payload nesting exceeds the walk depth cap")
        if _is_tensor(value):
            return self.offload_tensor(value, prefix)
        if isinstance(value, dict):
            if STORAGE_REF_KEY in value:
                return value  # already a ref
            return {
                k: self.dehydrate(v, prefix, _depth=_depth + 1) for k, v in value.items()
            }
        if isinstance(value, list):
            return [self.dehydrate(v, prefix, _depth=_depth + 1) for v in value]
        if isinstance(value, (str, bytes)):
            size = len(value)
            if size > self.max_inline_size:
                return self._offload_json(value, prefix)
        return value

    def dehydrate_document(self, value, prefix: _t.Optional[str] = None):
        """Dehydrate, then re-offload the WHOLE document if it is still over
        the inline cap after the walk (reference: manager.go:375-429)."""
        out = self.dehydrate(value, prefix)
        if self._json_size(out) > self.max_inline_size:
            return self._offload_json(out, prefix or self.output_prefix)
        return out

    def offload_tensor(self, tensor, prefix: _t.Optional[str] = None) -> dict:
        key = f"{prefix or self.output_prefix}/t-{uuid.uuid4().hex}"
        meta = self.tensors.put(key, tensor)
        return {STORAGE_REF_KEY: meta}

    def _offload_json(self, value, prefix: str) -> dict:
        key = f"{prefix}/b-{uuid.uuid4().hex}"
        if isinstance(value, bytes):
            data = value
            kind = "bytes"
        else:
            data = json.dumps(value, separators=(",", ":"), default=_json_default).encode()
            kind = "json"
        self.store.write(key, data)
        return {
            STORAGE_REF_KEY: {"key": key, "kind": kind, "size": len(data), "ts": time.time()}
        }

    @staticmethod
    def _json_size(value) -> int:
        try:
            return len(json.dumps(value, separators=(",", ":"), default=_json_default))
        except (TypeError, ValueError):
            return 0

    # ------------------------------------------------------------------
    # hydrate: ref → value
    # ------------------------------------------------------------------

    def hydrate(self, value, *, _depth: int = 0):
        """Resolve `$storageRef` markers back to values (reference:
        manager.go:312-372).  Tensor refs resolve to the live tensor."""
        if _depth > MAX_WALK_DEPTH:
            raise RefError("payload nesting exceeds the walk depth cap")
        if isinstance(value, dict):
            if STORAGE_REF_KEY in value:
                return self.resolve_ref(value)
            if ENV_REF_KEY in value:
                # environment indirection (role of the reference's
                # $configMapRef — kube_refs.go:100-167)
                import os

                ref = value[ENV_REF_KEY]
                name = ref.get("name") if isinstance(ref, dict) else str(ref)
                return os.environ.get(name, ref.get("default") if isinstance(ref, dict) else None)
            if FILE_REF_KEY in value:
                ref = value[FILE_REF_KEY]
                path = ref.get("path") if isinstance(ref, dict) else str(ref)
                self.validate_ref_key(path)
                with open(path, "r", encoding="utf-8") as fh:
                    text = fh.read()
                if isinstance(ref, dict) and ref.get("json"):
                    return json.loads(text)
                return text
            return {k: self.hydrate(v, _depth=_depth + 1) for k, v in value.items()}
        if isinstance(value, list):
            return [self.hydrate(v, _depth=_depth + 1) for v in value]
        return value

    def resolve_ref(self, ref_marker: dict):
        ref = ref_marker.get(STORAGE_REF_KEY)
        if not isinstance(ref, dict):
            raise RefError(f"malformed $storageRef: {ref_marker!r}")
        key = ref.get("key", "")
        self.validate_ref_key(key)
        kind = ref.get("kind", "json")
        if kind == "tensor":
            value = self.tensors.get(key)
        elif kind == "bytes":
            value = self.store.read(key)
        else:
            value = json.loads(self.store.read(key).decode())
            # the stored blob may itself contain refs (whole-document offload)
            value = self.hydrate(value)
        path = ref.get("path")
        if path:
            value = extract_path(value, path)
        return value

    @staticmethod
    def validate_ref_key(key: str) -> None:
        """Reject traversal / absolute ref keys (reference: manager.go:518-547)."""
        if not key or not _KEY_RE.match(key) or key.startswith("/") or ".." in key.split("/"):
            raise RefError(f"invalid storage ref key {key!r}")

    def contains_refs(self, value, *, _depth: int = 0) -> bool:
        """True if the payload carries ANY hydratable ref marker.

        Covers all three ref types ($storageRef, $envRef, $fileRef): the
        untrusted-input guard (engine._prepare_inputs) uses this to reject
        spoofed refs — an $envRef/$fileRef injected by a run submitter
        would otherwise exfiltrate process env/files into step outputs
        (reference analogue: storyrun_webhook.go:389-423 ref-spoofing
        rejection)."""
        if _depth > MAX_WALK_DEPTH:
            return False
        if isinstance(value, dict):
            if STORAGE_REF_KEY in value or ENV_REF_KEY in value or FILE_REF_KEY in value:
                return True
            return any(self.contains_refs(v, _depth=_depth + 1) for v in value.values())
        if isinstance(value, list):
            return any(self.contains_refs(v, _depth=_depth + 1) for v in value)
        return False

    # ------------------------------------------------------------------
    # retention sweeper (reference: pkg/storage/retention.go:27-70)
    # ------------------------------------------------------------------

    def sweep(
        self,
        older_than_seconds: float,
        max_scan: int = 2000,
        max_delete: int = 200,
        now: _t.Optional[float] = None,
    ) -> int:
        """Delete blobs older than the retention window; bounded per sweep."""
        now = now if now is not None else time.time()
        deleted = 0
        scanned = 0
        mtime = getattr(self.store, "mtime", None)
        if mtime is None:
            return 0
        for key in self.store.list():
            if scanned >= max_scan or deleted >= max_delete:
                break
            scanned += 1
            ts = mtime(key)
            if ts is not None and (now - ts) > older_than_seconds:
                self.store.delete(key)
                deleted += 1
        return deleted


def _json_default(value):
    if _is_tensor(value):
        return f"<tensor {tuple(value.shape)}>"
    return str(value)


# ---------------------------------------------------------------------------
# path DSL: dot / index / [*] wildcard extraction (reference: path.go:23-139)
# ---------------------------------------------------------------------------

_PATH_TOKEN = re.compile(r"\.?([A-Za-z0-9_\-]+)|\[(\d+|\*)\]|\['([^']+)'\]|\[\"([^\"]+)\"\]")


class PathError(RefError):
    pass


def parse_path(path: str) -> _t.List[_t.Union[str, int, None]]:
    """Parse "a.b[0].c" / "items[*].id" into tokens; None = wildcard."""
    tokens: _t.List[_t.Union[str, int, None]] = []
    pos = 0
    while pos < len(path):
        m = _PATH_TOKEN.match(path, pos)
        if not m:
            raise PathError(f"malformed path {path!r} at offset {pos}")
        if m.group(1) is not None:
            tokens.append(m.group(1))
        elif m.group(2) is not None:
            tokens.append(None if m.group(2) == "*" else int(m.group(2)))
        else:
            tokens.append(m.group(3) or m.group(4))
        pos = m.end()
    return tokens


def extract_path(value, path: str):
    """Extract a sub-value with the dot/index/[*] DSL
    (reference: hydrateFromStorageRefPath manager.go:668)."""
    return _extract(value, parse_path(path))


def _extract(value, tokens: _t.List[_t.Union[str, int, None]]):
    if not tokens:
        return value
    head, rest = tokens[0], tokens[1:]
    if head is None:  # wildcard
        if not isinstance(value, list):
            raise PathError(f"[*] applied to {type(value).__name__}")
        return [_extract(v, rest) for v in value]
    if isinstance(head, int):
        if not isinstance(value, list):
            raise PathError(f"index [{head}] applied to {type(value).__name__}")
        try:
            return _extract(value[head], rest)
        except IndexError:
            raise PathError(f"index [{head}] out of range") from None
    if isinstance(value, dict):
        if head not in value:
            raise PathError(f"key {head!r} not found")
        return _extract(value[head], rest)
    raise PathError(f"key {head!r} applied to {type(value).__name__}")
