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
