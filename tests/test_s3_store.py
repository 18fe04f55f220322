"""S3Store against an in-process S3-compatible HTTP server (reference:
pkg/storage/s3_store.go — path-style, SSE headers, bounded retries).  The
fake speaks just enough of the S3 REST surface (PUT/GET/HEAD/DELETE +
ListObjectsV2) and records the SigV4/SSE headers for assertions."""
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pytest

from bobrapet_amd.storage.manager import StorageManager
from bobrapet_amd.storage.stores import BlobNotFound, S3Store, store_from_env


class _FakeS3(BaseHTTPRequestHandler):
    blobs = {}
    seen_headers = []
    fail_next = 0

    def log_message(self, *a):
        pass

    def _key(self):
        return self.path.split("?")[0].lstrip("/")

    def do_PUT(self):
        if _FakeS3.fail_next > 0:
            _FakeS3.fail_next -= 1
            self.send_response(500)
            self.end_headers()
            return
        _FakeS3.seen_headers.append(dict(self.headers))
        body = self.rfile.read(int(self.headers.get("Content-Length", 0)))
        _FakeS3.blobs[self._key()] = body
        self.send_response(200)
        self.end_headers()

    def do_GET(self):
        if "list-type=2" in self.path:
            import urllib.parse
            q = urllib.parse.parse_qs(self.path.split("?", 1)[1])
            prefix = q.get("prefix", [""])[0]
            bucket = self._key().split("/")[0]
            keys = [
                k.split("/", 1)[1]
                for k in sorted(_FakeS3.blobs)
                if k.startswith(bucket + "/") and k.split("/", 1)[1].startswith(prefix)
            ]
            xml = "<ListBucketResult>" + "".join(
                f"<Contents><Key>{k}</Key></Contents>" for k in keys
            ) + "</ListBucketResult>"
            self.send_response(200)
            self.end_headers()
            self.wfile.write(xml.encode())
            return
        body = _FakeS3.blobs.get(self._key())
        if body is None:
            self.send_response(404)
            self.end_headers()
            return
        self.send_response(200)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def do_HEAD(self):
        if self._key() in _FakeS3.blobs:
            self.send_response(200)
            self.send_header("Last-Modified", "Tue, 01 Jan 2030 00:00:00 GMT")
        else:
            self.send_response(404)
        self.end_headers()

    def do_DELETE(self):
        existed = _FakeS3.blobs.pop(self._key(), None) is not None
        self.send_response(204 if existed else 404)
        self.end_headers()


@pytest.fixture
def s3():
    _FakeS3.blobs = {}
    _FakeS3.seen_headers = []
    _FakeS3.fail_next = 0
    srv = ThreadingHTTPServer(("127.0.0.1", 0), _FakeS3)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    store = S3Store(
        bucket="payloads",
        endpoint=f"http://127.0.0.1:{srv.server_port}",
        region="us-east-1",
        access_key="AK",
        secret_key="SK",
        sse="AES256",
    )
    yield store
    srv.shutdown()


def test_roundtrip_list_delete(s3):
    s3.write("outputs/a", b"hello")
    s3.write("outputs/b", b"world")
    s3.write("inputs/c", b"x")
    assert s3.read("outputs/a") == b"hello"
    assert s3.list("outputs/") == ["outputs/a", "outputs/b"]
    s3.delete("outputs/a")
    with pytest.raises(BlobNotFound):
        s3.read("outputs/a")
    assert s3.mtime("outputs/b") is not None


def test_sigv4_and_sse_headers_present(s3):
    s3.write("k", b"v")
    h = {k.lower(): v for k, v in _FakeS3.seen_headers[-1].items()}
    auth = h.get("authorization", "")
    assert auth.startswith("AWS4-HMAC-SHA256 Credential=AK/")
    assert "SignedHeaders=" in auth and "Signature=" in auth
    assert h.get("x-amz-content-sha256")
    assert h.get("x-amz-server-side-encryption") == "AES256"


def test_retries_on_5xx(s3):
    _FakeS3.fail_next = 2
    s3.write("retry-me", b"ok")   # 2 x 500 then success
    assert s3.read("retry-me") == b"ok"


def test_storage_manager_over_s3(s3):
    mgr = StorageManager(store=s3, max_inline_size=16)
    doc = {"big": "y" * 200, "small": 1}
    out = mgr.dehydrate(doc)
    assert "$storageRef" in out["big"]
    assert mgr.hydrate(out) == doc


def test_store_from_env(s3):
    env = {
        "BUBU_STORAGE_PROVIDER": "s3",
        "BUBU_STORAGE_S3_BUCKET": "payloads",
        "BUBU_STORAGE_S3_ENDPOINT": s3.endpoint,
        "BUBU_STORAGE_S3_ACCESS_KEY": "AK",
        "BUBU_STORAGE_S3_SECRET_KEY": "SK",
        "BUBU_STORAGE_PATH": "runs",
    }
    st = store_from_env(env)
    st.write("x", b"1")
    assert st.read("x") == b"1"
    assert "payloads/runs/x" in _FakeS3.blobs
    assert store_from_env({"BUBU_STORAGE_PROVIDER": "mem"}).name == "mem"
