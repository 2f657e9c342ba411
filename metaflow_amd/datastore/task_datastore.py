"""Per-(run, step, task, attempt) artifact index + task lifecycle protocol.

Parity target: /root/reference/metaflow/datastore/task_datastore.py
(save_artifacts :379, load_artifacts :499, done() :796, passdown_partial
:865, persist :880, clone :850). The three marker files per attempt are the
transaction protocol the whole runtime leans on:

    <attempt>.attempt.json   attempt started
    <attempt>.data.json      artifact index (name -> sha/size/type/encoding)
    <attempt>.DONE.lock      commit marker (write-once, written LAST)
"""

import json
import sys
import time

from ..exceptions import DataArtifactMissingError, DataException
from . import serializers


# id(obj) -> (ref, index_info) for artifacts THIS process
# deserialized; persist() uses it to skip re-serializing loaded
# immutable artifacts wherever they came from — self.<attr> lazy loads
# AND join inputs (self.x = inputs[0].x). A task subprocess lives for
# one task, but the Client API runs in long-lived notebooks, so the
# registry must not leak: numpy arrays (the big case) are held by
# WEAKREF (the finalize callback removes the entry before the id can
# be reused), and non-weakrefable immutables (bytes/str/scalars) go
# into a small FIFO — evicting one only forfeits the optimization.
_PROVENANCE = {}
_PROVENANCE_FIFO = []
_PROVENANCE_FIFO_MAX = 64


def _provenance_register(obj, info):
    import weakref

    oid = id(obj)
    try:
        ref = weakref.ref(
            obj, lambda _r, oid=oid: _PROVENANCE.pop(oid, None))
        _PROVENANCE[oid] = (ref, info)
        return
    except TypeError:
        pass
    _PROVENANCE[oid] = (lambda obj=obj: obj, info)
    _PROVENANCE_FIFO.append(oid)
    if len(_PROVENANCE_FIFO) > _PROVENANCE_FIFO_MAX:
        _PROVENANCE.pop(_PROVENANCE_FIFO.pop(0), None)


def _provenance_lookup(obj):
    entry = _PROVENANCE.get(id(obj))
    if entry is not None and entry[0]() is obj:
        return entry[1]
    return None


def _immutable_artifact(obj):
    """True only for objects that CANNOT have been mutated in place
    since load — the soundness condition for provenance-based persist
    (identity alone cannot detect in-place mutation of mutable types)."""
    if obj is None or isinstance(obj, (bytes, str, int, float, bool,
                                       complex)):
        return True
    np = sys.modules.get("numpy")
    if np is not None and isinstance(obj, np.ndarray):
        return not obj.flags.writeable
    return False


class TaskDataStore(object):
    METADATA_ATTEMPT = "attempt"
    METADATA_DATA = "data"
    METADATA_DONE = "DONE.lock"

    def __init__(self, flow_datastore, run_id, step_name, task_id,
                 attempt=None, mode="r", data_metadata=None):
        self._fds = flow_datastore
        self._ca_store = flow_datastore.ca_store
        self._storage = flow_datastore.storage
        self.run_id = str(run_id)
        self.step_name = step_name
        self.task_id = str(task_id)
        self._mode = mode
        self._attempt = attempt
        # name -> {"sha","size","type","encoding"}
        self._objects = {}
        self._info = {}
        self._cache = {}

        if mode == "r":
            if data_metadata is not None:
                self._objects = data_metadata
            else:
                if attempt is None:
                    self._attempt = self.latest_done_attempt()
                if self._attempt is not None:
                    idx = self._load_json(self._marker_path(
                        self.METADATA_DATA, self._attempt))
                    if idx is not None:
                        self._objects = idx.get("artifacts", {})
        elif mode == "w":
            if attempt is None:
                self._attempt = 0
        elif mode != "d":
            raise DataException("Unknown datastore mode %r" % mode)

    # ------------------------------------------------------------------ paths
    @property
    def pathspec(self):
        return "/".join((self._fds.flow_name, self.run_id, self.step_name,
                         self.task_id))

    @property
    def attempt(self):
        return self._attempt

    def _task_root(self):
        return self._storage.path_join(self._fds.flow_name, self.run_id,
                                       self.step_name, self.task_id)

    def _marker_path(self, kind, attempt):
        return self._storage.path_join(
            self._task_root(), "%d.%s" % (attempt, kind)
        ) if kind != self.METADATA_DONE else self._storage.path_join(
            self._task_root(), "%d.%s" % (attempt, self.METADATA_DONE))

    def _meta_path(self, name, attempt):
        return self._storage.path_join(self._task_root(),
                                       "%d.meta.%s.json" % (attempt, name))

    def _load_json(self, path):
        for _p, blob, _m in self._storage.load_bytes([path]):
            if blob is None:
                return None
            return json.loads(blob.decode("utf-8"))
        return None

    def _save_json(self, path, obj, overwrite=True):
        data = json.dumps(obj).encode("utf-8")
        self._storage.save_bytes(iter([(path, (data, None))]),
                                 overwrite=overwrite)

    # ------------------------------------------------------------- lifecycle
    def init_task(self):
        assert self._mode == "w"
        self._save_json(
            self._marker_path(self.METADATA_ATTEMPT, self._attempt),
            {"time": time.time(), "attempt": self._attempt})

    def latest_done_attempt(self):
        """Highest attempt with a DONE marker, else None.

        Reference does a 3-marker batched probe (flow_datastore.py:190-228);
        locally a single list_content is cheaper.
        """
        from ..config import MAX_ATTEMPTS

        paths = [self._marker_path(self.METADATA_DONE, a)
                 for a in range(MAX_ATTEMPTS)]
        done = self._storage.is_file(paths)
        for a in range(MAX_ATTEMPTS - 1, -1, -1):
            if done[a]:
                return a
        return None

    def has_done_attempt(self):
        return self.latest_done_attempt() is not None

    def done(self):
        """Commit: write the artifact index, then the DONE marker (last)."""
        assert self._mode == "w"
        self._save_json(self._marker_path(self.METADATA_DATA, self._attempt),
                        {"artifacts": self._objects})
        self._save_json(self._marker_path(self.METADATA_DONE, self._attempt),
                        {"time": time.time()})

    # ------------------------------------------------------------- artifacts
    def passdown(self, other, names=None):
        """Zero-copy propagation: copy index entries (sha refs) from another
        task datastore — no blob IO (reference passdown_partial)."""
        assert self._mode == "w"
        for name, info in other._objects.items():
            if names is not None and name not in names:
                continue
            if name not in self._objects:
                self._objects[name] = info

    def save_artifacts(self, name_obj_pairs):
        assert self._mode == "w"
        names, blobs, encodings, types = [], [], [], []
        for name, obj in name_obj_pairs:
            data, enc = serializers.serialize(obj)
            names.append(name)
            blobs.append(data)
            encodings.append(enc)
            types.append(serializers.type_name(obj))
        results = self._ca_store.save_blobs(blobs)
        for name, blob, enc, tname, (_uri, key) in zip(
                names, blobs, encodings, types, results):
            self._objects[name] = {
                "sha": key,
                "size": len(blob),
                "type": tname,
                "encoding": enc,
            }

    def persist(self, flow):
        """Persist all artifact attributes of a flow instance.

        Loaded-and-immutable inputs are persisted by PROVENANCE: if the
        attribute is still the exact object the lazy loader returned
        (flowspec.__getattr__ records (id, index_info)) and the object
        cannot have been mutated in place (bytes/str/scalars, read-only
        numpy views from the numpy-v1 codec), the original index entry
        is reused — no re-serialize, no re-hash. For a 1 GiB read-only
        input that is ~2 s of per-task CPU saved (config 2,
        profiles/bench_results_r02.md)."""
        pairs = []
        for name, obj in flow._artifacts_to_persist():
            info = _provenance_lookup(obj) if \
                _immutable_artifact(obj) else None
            if info is not None:
                self._objects[name] = dict(info)
            else:
                pairs.append((name, obj))
        self.save_artifacts(pairs)

    def load_artifacts(self, names):
        """Yield (name, obj)."""
        want = {}
        for name in names:
            if name in self._cache:
                yield name, self._cache[name]
                continue
            info = self._objects.get(name)
            if info is None:
                raise DataArtifactMissingError(
                    "Artifact '%s' not found in task %s"
                    % (name, self.pathspec))
            want.setdefault(info["sha"], []).append(name)
        if not want:
            return
        for sha, data in self._ca_store.load_blobs(list(want)):
            for name in want[sha]:
                obj = serializers.deserialize(
                    data, self._objects[name]["encoding"])
                self._cache[name] = obj
                if _immutable_artifact(obj):
                    _provenance_register(obj, self._objects[name])
                yield name, obj

    def __contains__(self, name):
        return name in self._objects

    def __getitem__(self, name):
        for _n, obj in self.load_artifacts([name]):
            return obj
        raise KeyError(name)

    def get(self, name, default=None):
        if name in self._objects:
            return self[name]
        return default

    def artifact_sha(self, name):
        info = self._objects.get(name)
        return info["sha"] if info else None

    def artifact_info(self, name):
        return self._objects.get(name)

    def artifact_names(self):
        return list(self._objects)

    def items(self):
        for name in self._objects:
            yield name, self[name]

    @property
    def index(self):
        return dict(self._objects)

    # --------------------------------------------------------- task metadata
    def save_metadata(self, name, obj):
        self._save_json(self._meta_path(name, self._attempt), obj)

    def load_metadata(self, name, attempt=None):
        a = self._attempt if attempt is None else attempt
        if a is None:
            return None
        return self._load_json(self._meta_path(name, a))

    # ----------------------------------------------------------------- logs
    def _log_path(self, stream, attempt):
        return self._storage.path_join(self._task_root(),
                                       "%d.%s.log" % (attempt, stream))

    def save_logs(self, stream, data):
        if isinstance(data, str):
            data = data.encode("utf-8")
        self._storage.save_bytes(
            iter([(self._log_path(stream, self._attempt), (data, None))]),
            overwrite=True)

    def load_logs(self, stream, attempt=None):
        a = self._attempt if attempt is None else attempt
        if a is None:
            return ""
        for _p, blob, _m in self._storage.load_bytes(
                [self._log_path(stream, a)]):
            return blob.decode("utf-8", "replace") if blob else ""
        return ""

    # ---------------------------------------------------------------- clone
    def clone(self, origin):
        """Resume support: copy the origin task's artifact index and commit.
        Metadata-only — CAS dedup means no blob moves (clone_util.py:10)."""
        assert self._mode == "w"
        self.init_task()
        self._objects = dict(origin._objects)
        # the scheduler drives successors off these two metadata entries
        for name in ("transition", "foreach_stack", "control_mapper_tasks"):
            value = origin.load_metadata(name)
            if value is not None:
                self.save_metadata(name, value)
        self.save_metadata("attempt_ok", {"ok": True, "cloned": True,
                                          "origin": origin.pathspec})
        self.done()
