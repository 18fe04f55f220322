"""Storage/payload layer tests (reference parity: pkg/storage/manager.go —
hydrate/dehydrate walk, path DSL, ref validation, retention sweeper)."""
import pytest

from bobrapet_amd.storage import (
    BlobNotFound,
    FileStore,
    MemStore,
    PathError,
    RefError,
    StorageManager,
    extract_path,
    parse_path,
)


@pytest.fixture
def mgr():
    return StorageManager(store=MemStore(), max_inline_size=64)


class TestDehydrateHydrate:
    def test_small_values_stay_inline(self, mgr):
        v = {"a": 1, "b": "short"}
        assert mgr.dehydrate(v) == v

    def test_large_string_offloads(self, mgr):
        big = "x" * 200
        out = mgr.dehydrate({"big": big})
        assert "$storageRef" in out["big"]
        assert mgr.hydrate(out) == {"big": big}

    def test_whole_document_offload(self, mgr):
        doc = {f"k{i}": "v" * 10 for i in range(20)}
        out = mgr.dehydrate_document(doc)
        assert "$storageRef" in out
        assert mgr.hydrate(out) == doc

    def test_nested_and_lists(self, mgr):
        doc = {"outer": [{"inner": "y" * 100}, {"small": 1}]}
        out = mgr.dehydrate(doc)
        assert "$storageRef" in out["outer"][0]["inner"]
        assert out["outer"][1] == {"small": 1}
        assert mgr.hydrate(out) == doc

    def test_existing_ref_passthrough(self, mgr):
        ref = {"$storageRef": {"key": "outputs/x", "kind": "json"}}
        assert mgr.dehydrate({"r": ref})["r"] is ref

    def test_contains_refs(self, mgr):
        assert not mgr.contains_refs({"a": 1})
        out = mgr.dehydrate({"big": "x" * 100})
        assert mgr.contains_refs(out)

    def test_ref_key_validation(self, mgr):
        for bad in ("", "/abs/path", "a/../b", "sp ace"):
            with pytest.raises(RefError):
                mgr.validate_ref_key(bad)
        mgr.validate_ref_key("outputs/b-123abc")

    def test_ref_with_path_extraction(self, mgr):
        out = mgr.dehydrate_document({"items": [{"id": 1}, {"id": 2}], "pad": "z" * 100})
        ref = dict(out["$storageRef"])
        ref["path"] = "items[1].id"
        assert mgr.resolve_ref({"$storageRef": ref}) == 2

    def test_missing_blob(self, mgr):
        with pytest.raises(BlobNotFound):
            mgr.resolve_ref({"$storageRef": {"key": "outputs/nope", "kind": "json"}})


class TestPathDSL:
    def test_parse(self):
        assert parse_path("a.b[0].c") == ["a", "b", 0, "c"]
        assert parse_path("items[*].id") == ["items", None, "id"]
        assert parse_path("m['k-1'].v") == ["m", "k-1", "v"]

    def test_extract(self):
        data = {"items": [{"id": 1, "tags": ["a"]}, {"id": 2, "tags": ["b"]}]}
        assert extract_path(data, "items[0].id") == 1
        assert extract_path(data, "items[*].id") == [1, 2]
        assert extract_path(data, "items[1].tags[0]") == "b"

    def test_extract_errors(self):
        with pytest.raises(PathError):
            extract_path({"a": 1}, "b")
        with pytest.raises(PathError):
            extract_path({"a": [1]}, "a[5]")
        with pytest.raises(PathError):
            extract_path({"a": 1}, "a[*]")


class TestStores:
    def test_file_store_roundtrip(self, tmp_path):
        fs = FileStore(str(tmp_path))
        fs.write("outputs/a/b", b"hello")
        assert fs.read("outputs/a/b") == b"hello"
        assert fs.list("outputs/") == ["outputs/a/b"]
        fs.delete("outputs/a/b")
        with pytest.raises(BlobNotFound):
            fs.read("outputs/a/b")

    def test_file_store_traversal_guard(self, tmp_path):
        fs = FileStore(str(tmp_path))
        with pytest.raises(Exception):
            fs.write("../escape", b"x")

    def test_retention_sweep(self):
        store = MemStore()
        mgr = StorageManager(store=store, max_inline_size=8)
        mgr.dehydrate({"big": "x" * 100})
        assert len(store.list()) == 1
        # nothing old enough
        assert mgr.sweep(older_than_seconds=3600) == 0
        assert mgr.sweep(older_than_seconds=-1) == 1
        assert store.list() == []
