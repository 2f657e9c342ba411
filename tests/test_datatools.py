"""Bulk ObjectStore datatool tests."""

import pytest

from metaflow_amd.datatools import ObjectStore
from metaflow_amd.exceptions import MFXException


def test_put_get_many(tmp_path):
    with ObjectStore(str(tmp_path / "store")) as s:
        objs = s.put_many([("x/a", b"aaa"), ("x/b", b"bb"), ("c", b"c")])
        assert [o.key for o in objs] == ["x/a", "x/b", "c"]
        assert s.get_many(["x/a", "c"]) == [b"aaa", b"c"]
        assert s.info("x/b").size == 2
        assert len(s.list_paths("x")) == 2


def test_put_files(tmp_path):
    src = tmp_path / "f.bin"
    src.write_bytes(b"payload" * 100)
    with ObjectStore(str(tmp_path / "store")) as s:
        [obj] = s.put_files([("models/f.bin", str(src))])
        assert obj.size == 700
        assert s.get("models/f.bin") == b"payload" * 100


def test_key_escape_rejected(tmp_path):
    with ObjectStore(str(tmp_path / "store")) as s:
        with pytest.raises(MFXException):
            s.put("../../etc/passwd", b"nope")


def test_fault_injection_retries(tmp_path, monkeypatch):
    """With 40% injected transient failures, the pooled retries still
    complete every operation (reference parity: s3op --inject-failure)."""
    import random

    random.seed(1234)
    monkeypatch.setenv("MFX_INJECT_IO_FAILURES", "40")
    with ObjectStore(str(tmp_path / "store"), retries=8) as s:
        s.put_many([("k%d" % i, b"v%d" % i) for i in range(50)])
        got = s.get_many(["k%d" % i for i in range(50)])
        assert got == [b"v%d" % i for i in range(50)]


def test_fault_injection_exhaustion(tmp_path, monkeypatch):
    import pytest

    monkeypatch.setenv("MFX_INJECT_IO_FAILURES", "100")
    with ObjectStore(str(tmp_path / "store"), retries=2) as s:
        with pytest.raises(MFXException):
            s.put("k", b"v")
