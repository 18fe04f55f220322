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

    def test_contains_refs_covers_env_and_file_refs(self, mgr):
        # spoofed env/file refs in untrusted inputs must be detectable
        # (exfiltration via injected {"$envRef": ...})
        assert mgr.contains_refs({"x": {"$envRef": {"name": "SECRET"}}})
        assert mgr.contains_refs([{"deep": {"$fileRef": {"path": "etc/x"}}}])
        assert not mgr.contains_refs({"x": {"envRef": "not-a-ref"}})

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


class TestHydrateDehydrateFuzz:
    """Property-based round-trip fuzzing of the $storageRef walker
    (reference: pkg/storage/manager_fuzz_test.go:28,67)."""

    @staticmethod
    def _json_values():
        from hypothesis import strategies as st

        scalars = st.one_of(
            st.none(),
            st.booleans(),
            st.integers(min_value=-(2**31), max_value=2**31),
            st.floats(allow_nan=False, allow_infinity=False, width=32),
            st.text(max_size=64),
        )
        return st.recursive(
            scalars,
            lambda children: st.one_of(
                st.lists(children, max_size=5),
                st.dictionaries(
                    st.text(min_size=1, max_size=12).filter(
                        lambda k: not k.startswith("$")
                    ),
                    children,
                    max_size=5,
                ),
            ),
            max_leaves=25,
        )

    def test_round_trip_any_json(self):
        from hypothesis import given, settings

        from bobrapet_amd.engine import EngineConfig, RunEngine

        eng = RunEngine(EngineConfig(cpu_workers=1, max_inline_size=48)).start()
        try:

            @settings(max_examples=120, deadline=None)
            @given(self._json_values())
            def check(doc):
                dehydrated = eng.storage.dehydrate_document(doc)
                back = eng.storage.hydrate(dehydrated)
                assert back == doc, (doc, dehydrated, back)

            check()
        finally:
            eng.stop()

    def test_spoofed_refs_never_resolve_foreign_data(self):
        """A crafted $storageRef in user data must not read another ref's
        payload after the untrusted-input scrub (spoof rejection)."""
        from bobrapet_amd.engine import EngineConfig, RunEngine

        eng = RunEngine(EngineConfig(cpu_workers=1, max_inline_size=16)).start()
        try:
            secret_ref = eng.storage.dehydrate_document({"secret": "x" * 64})
            assert isinstance(secret_ref, dict) and "$storageRef" in secret_ref
            eng.apply_yaml(
                """
kind: Story
metadata: {name: guard}
spec:
  steps:
    - {name: a, type: sleep, with: {duration: 0ms}}
"""
            )
            import pytest as _pytest

            with _pytest.raises(ValueError):
                eng.submit_run("default/guard", {"data": secret_ref})
        finally:
            eng.stop()
