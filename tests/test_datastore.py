"""Datastore unit tests: storage atomicity, CAS dedup, task lifecycle."""

import os

from metaflow_amd.datastore import FlowDataStore
from metaflow_amd.datastore.cas import (
    CODEC_GZIP,
    CODEC_RAW,
    ContentAddressedStore,
    pack,
    unpack,
)
from metaflow_amd.datastore.storage import LocalStorage


def make_fds(tmp_path, flow="TestFlow"):
    return FlowDataStore(flow, LocalStorage(str(tmp_path)))


def test_local_storage_roundtrip(tmp_path):
    s = LocalStorage(str(tmp_path))
    s.save_bytes(iter([("a/b/c", (b"hello", {"k": 1}))]))
    assert s.is_file(["a/b/c"]) == [True]
    assert s.is_file(["a/b/missing"]) == [False]
    [(path, data, meta)] = list(s.load_bytes(["a/b/c"]))
    assert data == b"hello"
    assert meta == {"k": 1}
    assert s.size_file("a/b/c") == 5


def test_local_storage_no_overwrite(tmp_path):
    s = LocalStorage(str(tmp_path))
    s.save_bytes(iter([("x", (b"one", None))]))
    s.save_bytes(iter([("x", (b"two", None))]), overwrite=False)
    [(_, data, _)] = list(s.load_bytes(["x"]))
    assert data == b"one"
    s.save_bytes(iter([("x", (b"two", None))]), overwrite=True)
    [(_, data, _)] = list(s.load_bytes(["x"]))
    assert data == b"two"


def test_pack_unpack():
    data = b"payload" * 100
    assert unpack(pack(data, CODEC_RAW)) == data
    assert unpack(pack(data, CODEC_GZIP)) == data
    assert len(pack(data, CODEC_GZIP)) < len(data)


def test_cas_dedup(tmp_path):
    s = LocalStorage(str(tmp_path))
    cas = ContentAddressedStore("data", s)
    blobs = [b"aaa", b"bbb", b"aaa"]
    results = cas.save_blobs(blobs)
    assert results[0][1] == results[2][1]  # same content -> same key
    loaded = dict(cas.load_blobs([k for _u, k in results]))
    assert loaded[results[0][1]] == b"aaa"
    assert loaded[results[1][1]] == b"bbb"
    # only two physical files
    files = []
    for dirpath, _dirs, names in os.walk(str(tmp_path)):
        files += [n for n in names if not n.endswith("_meta")]
    assert len(files) == 2


def test_cas_large_blob_raw(tmp_path):
    from metaflow_amd import config

    s = LocalStorage(str(tmp_path))
    cas = ContentAddressedStore("data", s)
    big = os.urandom(config.CAS_COMPRESS_MAX_SIZE + 1)
    [(_u, key)] = cas.save_blobs([big])
    [(k, data)] = list(cas.load_blobs([key]))
    assert data == big


def test_task_datastore_lifecycle(tmp_path):
    fds = make_fds(tmp_path)
    ds = fds.get_task_datastore("1", "start", "1", attempt=0, mode="w")
    ds.init_task()
    ds.save_artifacts([("x", 42), ("y", [1, 2, 3])])
    assert ds.latest_done_attempt() is None
    ds.done()
    assert ds.latest_done_attempt() == 0

    rd = fds.get_task_datastore("1", "start", "1")
    assert rd.attempt == 0
    assert "x" in rd
    assert rd["x"] == 42
    assert rd["y"] == [1, 2, 3]
    assert rd.artifact_sha("x") is not None


def test_task_datastore_attempts(tmp_path):
    fds = make_fds(tmp_path)
    a0 = fds.get_task_datastore("1", "s", "1", attempt=0, mode="w")
    a0.init_task()
    a0.save_artifacts([("v", "first")])
    a0.done()
    a1 = fds.get_task_datastore("1", "s", "1", attempt=1, mode="w")
    a1.init_task()
    a1.save_artifacts([("v", "second")])
    a1.done()
    rd = fds.get_task_datastore("1", "s", "1")
    assert rd.attempt == 1
    assert rd["v"] == "second"


def test_passdown(tmp_path):
    fds = make_fds(tmp_path)
    parent = fds.get_task_datastore("1", "a", "1", attempt=0, mode="w")
    parent.init_task()
    parent.save_artifacts([("big", list(range(100))), ("keep", "yes")])
    parent.done()

    child = fds.get_task_datastore("1", "b", "2", attempt=0, mode="w")
    child.init_task()
    rd_parent = fds.get_task_datastore("1", "a", "1")
    child.passdown(rd_parent)
    child.done()

    rd = fds.get_task_datastore("1", "b", "2")
    assert rd["keep"] == "yes"
    assert rd.artifact_sha("big") == rd_parent.artifact_sha("big")


def test_clone(tmp_path):
    fds = make_fds(tmp_path)
    orig = fds.get_task_datastore("1", "a", "1", attempt=0, mode="w")
    orig.init_task()
    orig.save_artifacts([("v", 99)])
    orig.save_metadata("transition", {"out_funcs": ["end"]})
    orig.save_metadata("foreach_stack", [])
    orig.done()

    new = fds.get_task_datastore("2", "a", "1", attempt=0, mode="w")
    new.clone(fds.get_task_datastore("1", "a", "1"))
    rd = fds.get_task_datastore("2", "a", "1")
    assert rd["v"] == 99
    ok = rd.load_metadata("attempt_ok")
    assert ok["ok"] and ok["cloned"]


def test_tensor_serializer_roundtrip():
    import torch

    from metaflow_amd.datastore import serializers

    for dtype in (torch.float32, torch.bfloat16, torch.int64):
        t = (torch.randn(3, 5) * 10).to(dtype)
        blob, enc = serializers.serialize(t)
        assert enc == serializers.ENC_TENSOR
        back = serializers.deserialize(blob, enc)
        assert back.dtype == dtype
        assert back.shape == t.shape
        assert torch.equal(back, t)


def test_flow_datastore_raw_data(tmp_path):
    fds = make_fds(tmp_path)
    [(uri, key)] = fds.save_data([b"code package"])
    [(k, data)] = fds.load_data([key])
    assert data == b"code package"


def test_checkpoint_save_load_roundtrip(tmp_path):
    """save_state_dict -> load_state_dict through the CAS raw fast path
    (cas.blob_file + single-copy readinto), mixed dtypes + non-tensor
    metadata; loaded tensors must be writable."""
    import torch

    from metaflow_amd.parallel.checkpoint import (
        load_state_dict,
        save_state_dict,
    )

    fds = make_fds(tmp_path)
    ds = fds.get_task_datastore("9", "train", "t1", attempt=0, mode="w")
    ds.init_task()
    torch.manual_seed(0)
    state = {
        "w": torch.randn(1000, 33, dtype=torch.bfloat16),
        "m": torch.randn(513, dtype=torch.float32),
        "step": 7,
    }
    index = save_state_dict(ds, state, name="u")
    assert set(index) == {"w", "m"}
    ds.done()

    rd = fds.get_task_datastore("9", "train", "t1")
    # the blobs must be locally resolvable as raw files (fast path)
    for info in index.values():
        loc = rd._ca_store.blob_file(info["sha"])
        assert loc is not None, "raw fast path not taken"
    out = load_state_dict(rd, name="u")
    assert out["step"] == 7
    for k in ("w", "m"):
        assert out[k].dtype == state[k].dtype
        assert torch.equal(out[k], state[k]), k
    out["w"] += 1  # writable (trained on after resume)


def test_serializer_registry_order_and_override():
    """Priority-ordered registry: lower priority wins; registering the
    same encoding again replaces the instance (extension override)."""
    from metaflow_amd.datastore import serializers as S

    order = S.get_ordered_serializers()
    prios = [s.priority for s in order]
    assert prios == sorted(prios)
    assert order[-1].encoding == S.ENC_PICKLE  # universal fallback last

    class Grabby(S.ArtifactSerializer):
        encoding = "test-grabby-v1"
        priority = 1  # runs before tensor + pickle

        def can_serialize(self, obj):
            return isinstance(obj, dict) and obj.get("grab") is True

        def serialize(self, obj):
            return b"GRABBED"

        def deserialize(self, data):
            assert data == b"GRABBED"
            return {"grab": True, "restored": True}

    try:
        blob, enc = S.serialize({"grab": True})
        assert enc == "test-grabby-v1" and blob == b"GRABBED"
        assert S.deserialize(blob, enc) == {"grab": True,
                                            "restored": True}
        # non-matching objects still fall through to pickle
        blob2, enc2 = S.serialize({"grab": False})
        assert enc2 == S.ENC_PICKLE
        assert S.deserialize(blob2, enc2) == {"grab": False}
    finally:
        S._REGISTRY.pop("test-grabby-v1", None)
        S._ORDERED = None


def test_serializer_failure_falls_through():
    """A serializer whose serialize() raises falls through to pickle
    instead of failing the artifact save."""
    from metaflow_amd.datastore import serializers as S

    class Broken(S.ArtifactSerializer):
        encoding = "test-broken-v1"
        priority = 1

        def can_serialize(self, obj):
            return isinstance(obj, (list, dict, int, str))

        def serialize(self, obj):
            raise RuntimeError("boom")

        def deserialize(self, data):
            raise RuntimeError("boom")

    try:
        blob, enc = S.serialize([1, 2, 3])
        assert enc == S.ENC_PICKLE
        assert S.deserialize(blob, enc) == [1, 2, 3]
    finally:
        S._REGISTRY.pop("test-broken-v1", None)
        S._ORDERED = None


def test_serializer_extension_flow(tmp_path, tmp_datastore):
    """An extension package contributes an artifact serializer via
    ARTIFACT_SERIALIZERS; a flow artifact round-trips through it with
    its encoding recorded (reference serializer.py:252 bootstrap)."""
    import os
    import subprocess
    import sys
    import textwrap

    ext = tmp_path / "metaflow_amd_extensions" / "serext"
    ext.mkdir(parents=True)
    (ext / "__init__.py").write_text(textwrap.dedent("""
        from metaflow_amd.datastore.serializers import ArtifactSerializer

        class Payload(object):
            def __init__(self, text):
                self.text = text

        class PayloadSerializer(ArtifactSerializer):
            encoding = "payload-v1"
            priority = 10

            def can_serialize(self, obj):
                return isinstance(obj, Payload)

            def serialize(self, obj):
                return obj.text.encode() + b"|EXT"

            def deserialize(self, data):
                assert data.endswith(b"|EXT")
                return Payload(data[:-4].decode())

        ARTIFACT_SERIALIZERS = [PayloadSerializer]
    """))
    flow = tmp_path / "ser_flow.py"
    flow.write_text(textwrap.dedent("""
        from metaflow_amd import FlowSpec, step
        from metaflow_amd_extensions.serext import Payload

        class SerFlow(FlowSpec):
            @step
            def start(self):
                self.art = Payload("hello")
                self.next(self.end)

            @step
            def end(self):
                assert self.art.text == "hello", self.art
                self.ok = True

        if __name__ == "__main__":
            SerFlow()
    """))
    from .test_runtime import REPO

    env = dict(os.environ)
    env["PYTHONPATH"] = os.pathsep.join(
        [str(tmp_path), REPO, env.get("PYTHONPATH", "")])
    env["MFX_NUM_GPUS"] = "0"
    proc = subprocess.run(
        [sys.executable, str(flow), "--quiet",
         "--datastore-root", tmp_datastore, "run"],
        env=env, capture_output=True, text=True, timeout=300)
    assert proc.returncode == 0, proc.stderr[-3000:]
    # the artifact's recorded encoding is the extension codec
    import json as _json

    found = []
    for root, _dirs, files in os.walk(tmp_datastore):
        for fn in files:
            if fn.endswith(".data"):
                info = _json.load(open(os.path.join(root, fn)))
                arts = info.get("artifacts", {})
                if "art" in arts:
                    found.append(arts["art"].get("encoding"))
    assert "payload-v1" in found, found


def test_native_parallel_load_matches_python(tmp_path):
    """The native parallel-pread load path returns byte-identical data
    to the Python path, for blobs above and below the 16 MiB chunk."""
    import numpy as np

    from metaflow_amd.datastore.cas import ContentAddressedStore
    from metaflow_amd.datastore.storage import LocalStorage
    from metaflow_amd.ops import cas_native

    try:
        engine = cas_native.engine()
    except ImportError:
        import pytest

        pytest.skip("_mfx_cas not built")
    store = ContentAddressedStore("data", LocalStorage(str(tmp_path)))
    rng = np.random.default_rng(7)
    blobs = [rng.bytes(100), rng.bytes(5 << 20), rng.bytes(40 << 20)]
    results = store.save_blobs(blobs, raw=True)
    for (want, (_uri, key)) in zip(blobs, results):
        loc = store.blob_file(key)
        assert loc is not None
        path, off = loc
        got_native = engine.load_blob_parallel(path, off)
        assert got_native == want
        # the public API (which routes through the native path) agrees
        [(k2, got_api)] = list(store.load_blobs([key]))
        assert k2 == key and bytes(got_api) == want


def test_cas_save_stream_matches_save_blobs(tmp_path):
    """Streaming save (pipelined hash+write) must produce the SAME keys
    as save_blobs for identical content (dedup across paths), for both
    the plain-sha (<16 MiB) and Merkle (>=16 MiB) regimes."""
    import numpy as np

    from metaflow_amd.datastore.cas import ContentAddressedStore
    from metaflow_amd.datastore.storage import LocalStorage

    store = ContentAddressedStore("data", LocalStorage(str(tmp_path)))
    rng = np.random.default_rng(3)
    for size in (3 << 20, 40 << 20):
        blob = rng.bytes(size)
        [(_u, key_ref)] = store.save_blobs([blob], raw=True)
        # chunks of 8 MiB-multiples (leaf contract), ragged tail
        chunks = [memoryview(blob)[o:o + (16 << 20)]
                  for o in range(0, size, 16 << 20)]
        _uri, key_stream = store.save_stream(iter(chunks), size)
        assert key_stream == key_ref, size
        [(_k, back)] = list(store.load_blobs([key_stream]))
        assert bytes(back) == blob
    # dedup: re-streaming identical content leaves one physical file
    blob = rng.bytes(20 << 20)
    _u1, k1 = store.save_stream(
        iter([memoryview(blob)]), len(blob))
    _u2, k2 = store.save_stream(
        iter([memoryview(blob)]), len(blob))
    assert k1 == k2
    import os as _os

    files = []
    for dirpath, _d, names in _os.walk(str(tmp_path)):
        files += [n for n in names if k1[:8] in n]
    assert len(files) == 1


def test_stream_saver_native_key_parity(tmp_path):
    """C++ StreamSaver produces the same content keys as parallel_key
    for both regimes, and the file round-trips through load_blobs."""
    import numpy as np
    import torch

    from metaflow_amd.datastore.cas import (
        CODEC_RAW,
        MAGIC,
        ContentAddressedStore,
        parallel_key,
    )
    from metaflow_amd.datastore.storage import LocalStorage

    try:
        from metaflow_amd.ops import _mfx_cas
    except ImportError:
        import pytest

        pytest.skip("_mfx_cas not built")
    store = ContentAddressedStore("data", LocalStorage(str(tmp_path)))
    header = MAGIC + bytes([1, CODEC_RAW, 0, 0])
    rng = np.random.default_rng(11)
    for size in (2 << 20, 40 << 20):
        blob = rng.bytes(size)
        t = torch.frombuffer(bytearray(blob), dtype=torch.uint8)
        import os as _os

        tmp = str(tmp_path / ("s%d.tmp" % size))
        sv = _mfx_cas.StreamSaver(tmp, header, 0)
        for o in range(0, size, 16 << 20):
            sv.feed(t[o:o + (16 << 20)])
        key = sv.finish()
        assert key == parallel_key(blob), size
        ap = store._storage._abs(store._key_path(key))
        _os.makedirs(_os.path.dirname(ap), exist_ok=True)
        _os.replace(tmp, ap)
        [(_k, back)] = list(store.load_blobs([key]))
        assert bytes(back) == blob


def test_torchless_native_load_engine(tmp_path):
    """A torch-less subprocess gets the STANDALONE _mfx_io pread engine
    for raw blob loads (the torch-linked engine is refused there —
    ops/cas_native.py's 10x-regression guard)."""
    import subprocess
    import sys
    import textwrap

    from .test_runtime import REPO

    code = textwrap.dedent("""
        import sys
        sys.path.insert(0, %r)
        from metaflow_amd.datastore import cas as C
        from metaflow_amd.datastore.cas import ContentAddressedStore
        from metaflow_amd.datastore.storage import LocalStorage
        import os
        store = ContentAddressedStore("f", LocalStorage(%r))
        data = os.urandom(6 << 20)
        [(_, key)] = store.save_blobs([data])
        [(_, got)] = list(store.load_blobs([key]))
        assert got == data
        eng = C._native_engine()
        assert "torch" not in sys.modules, "torch leaked into the task"
        print(type(eng).__name__)
    """) % (REPO, str(tmp_path))
    out = subprocess.run([sys.executable, "-c", code],
                         capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr[-2000:]
    # _IoEngine when the extension is built; None only if it is not
    assert out.stdout.strip() in ("_IoEngine", "None")
    assert out.stdout.strip() == "_IoEngine", \
        "_mfx_io extension missing — build_ext --inplace"


def test_numpy_serializer_roundtrip_and_readonly():
    """numpy-v1 codec: header+raw bytes round-trip for assorted
    dtypes/shapes; deserialized arrays are zero-copy READ-ONLY views
    over the CAS bytes; object dtype falls through to pickle."""
    import numpy as np

    from metaflow_amd.datastore import serializers

    cases = [np.arange(12, dtype=np.float32).reshape(3, 4),
             np.array(3.5),
             np.zeros((0, 5), dtype=np.int64),
             np.asfortranarray(np.arange(6).reshape(2, 3)),
             np.arange(6).reshape(2, 3)[:, ::2],      # non-contiguous
             np.array([True, False])]
    for a in cases:
        payload, enc = serializers.serialize(a)
        assert enc == "numpy-v1"
        b = serializers.deserialize(payload, enc)
        assert np.array_equal(b, a) and b.dtype == a.dtype
        assert not b.flags.writeable     # content-addressed = immutable
    payload, enc = serializers.serialize(np.array(["a", "b"],
                                                  dtype=object))
    assert enc == "pickle-v4"            # object dtype: pickle fallback


def test_numpy_artifact_through_flow(tmp_path, tmp_datastore):
    """A numpy artifact crosses steps via the numpy-v1 codec (the
    datastore records the encoding) and reads back equal."""
    import json
    import os
    import subprocess
    import sys
    import textwrap

    from .test_runtime import REPO, latest_run_id

    flow = tmp_path / "np_flow.py"
    flow.write_text(textwrap.dedent("""
        import numpy as np

        from metaflow_amd import FlowSpec, step

        class NpFlow(FlowSpec):
            @step
            def start(self):
                self.arr = np.arange(1024, dtype=np.float32)
                self.next(self.end)

            @step
            def end(self):
                assert self.arr.sum() == 1023 * 1024 / 2
                assert not self.arr.flags.writeable
                self.ok = True

        if __name__ == "__main__":
            NpFlow()
    """))
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    proc = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "run"],
        env=env, capture_output=True, text=True, timeout=300)
    assert proc.returncode == 0, proc.stderr[-3000:]
    run_id = latest_run_id(tmp_datastore, "NpFlow")
    data = json.load(open(os.path.join(
        tmp_datastore, "NpFlow", run_id, "start", "1", "0.data")))
    assert data["artifacts"]["arr"]["encoding"] == "numpy-v1"


def test_persist_provenance_skip(tmp_datastore):
    """persist() reuses the loaded artifact's index entry (no
    re-serialize/re-hash) ONLY when the attribute is the identical
    object AND immutable (read-only numpy / bytes); replaced or
    mutable values are re-serialized."""
    import numpy as np

    from metaflow_amd.datastore.flow_datastore import FlowDataStore
    from metaflow_amd.datastore.storage import LocalStorage
    from metaflow_amd.datastore.task_datastore import TaskDataStore

    fds = FlowDataStore("ProvFlow", LocalStorage(tmp_datastore))
    up = TaskDataStore(fds, "1", "start", "1", attempt=0, mode="w")
    up.init_task()
    arr = np.arange(1 << 16, dtype=np.uint8)
    up.save_artifacts([("big", arr), ("note", "hello")])
    up.done()

    down = TaskDataStore(fds, "1", "work", "2", attempt=0, mode="w")
    rd = TaskDataStore(fds, "1", "start", "1", mode="r")
    loaded = dict(rd.load_artifacts(["big", "note"]))
    assert not loaded["big"].flags.writeable

    class FakeFlow(object):
        pass

    flow = FakeFlow()
    flow.big = loaded["big"]
    flow.note = loaded["note"]
    flow.fresh = [1, 2, 3]
    # provenance is registered automatically at deserialization time
    flow._artifacts_to_persist = lambda: [
        ("big", flow.big), ("note", flow.note), ("fresh", flow.fresh)]

    calls = []
    orig = down.save_artifacts
    down.save_artifacts = lambda pairs: (
        calls.extend(n for n, _ in pairs), orig(pairs))
    down.persist(flow)
    # big + note skipped via provenance; only the new artifact saved
    assert calls == ["fresh"]
    assert down.artifact_info("big")["sha"] == \
        rd.artifact_info("big")["sha"]

    # a REPLACED object (same content, different identity) re-saves
    flow.big = loaded["big"].copy()          # writable copy
    calls.clear()
    down.persist(flow)
    assert "big" in calls


def test_provenance_skip_through_real_flow(tmp_path, tmp_datastore):
    """End-to-end: a read-only numpy artifact crosses a pass-through
    step keeping the SAME sha (no re-serialization drift), and a step
    that replaces it produces a new sha."""
    import json
    import os
    import subprocess
    import sys
    import textwrap

    from .test_runtime import REPO, latest_run_id

    flow = tmp_path / "prov_flow.py"
    flow.write_text(textwrap.dedent("""
        import numpy as np

        from metaflow_amd import FlowSpec, step

        class PvFlow(FlowSpec):
            @step
            def start(self):
                self.big = np.arange(1 << 16, dtype=np.uint8)
                self.next(self.mid)

            @step
            def mid(self):
                self.peek = int(self.big[:16].sum())   # loads big
                self.next(self.end)

            @step
            def end(self):
                self.big = self.big.copy() * 0 + 1     # replace
                self.total = int(self.big[:4].sum())

        if __name__ == "__main__":
            PvFlow()
    """))
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    proc = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "run"],
        env=env, capture_output=True, text=True, timeout=300)
    assert proc.returncode == 0, proc.stderr[-3000:]
    run_id = latest_run_id(tmp_datastore, "PvFlow")

    def sha(step):
        data = json.load(open(os.path.join(
            tmp_datastore, "PvFlow", run_id, step, {"start": "1",
            "mid": "2", "end": "3"}[step], "0.data")))
        return data["artifacts"]["big"]["sha"]

    assert sha("mid") == sha("start")       # provenance-skip kept sha
    assert sha("end") != sha("start")       # replacement re-saved


def test_provenance_skip_join_inputs(tmp_path, tmp_datastore):
    """The provenance fast path also covers JOIN inputs: a read-only
    array carried through `self.x = inputs[0].x` keeps its sha."""
    import json
    import os
    import subprocess
    import sys
    import textwrap

    from .test_runtime import REPO, latest_run_id

    flow = tmp_path / "provj_flow.py"
    flow.write_text(textwrap.dedent("""
        import numpy as np

        from metaflow_amd import FlowSpec, step

        class PvJFlow(FlowSpec):
            @step
            def start(self):
                self.big = np.arange(1 << 16, dtype=np.uint8)
                self.items = [0, 1]
                self.next(self.work, foreach="items")

            @step
            def work(self):
                self.part = int(self.input)
                self.next(self.join)

            @step
            def join(self, inputs):
                self.big = inputs[0].big       # join-input carry
                self.total = sum(i.part for i in inputs)
                self.next(self.end)

            @step
            def end(self):
                assert int(self.big[:4].sum()) == 6

        if __name__ == "__main__":
            PvJFlow()
    """))
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    proc = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "run"],
        env=env, capture_output=True, text=True, timeout=300)
    assert proc.returncode == 0, proc.stderr[-3000:]
    run_id = latest_run_id(tmp_datastore, "PvJFlow")

    def sha(step, tid):
        data = json.load(open(os.path.join(
            tmp_datastore, "PvJFlow", run_id, step, tid, "0.data")))
        return data["artifacts"]["big"]["sha"]

    assert sha("join", "4") == sha("start", "1")
