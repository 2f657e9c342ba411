"""Local JSON metadata provider.

Parity target: /root/reference/metaflow/plugins/metadata_providers/local.py —
runs/tasks/attempts/tags as JSON files under ``<ds_root>/<flow>/_meta``.
Run ids are microsecond timestamps (deliberately lock-free, like the
reference local.py:64-69); task ids come from the scheduler's in-process
sequence.
"""

import getpass
import json
import os
import time


class LocalMetadataProvider(object):
    TYPE = "local"

    def __init__(self, flow_name, storage):
        self.flow_name = flow_name
        self._storage = storage

    # ------------------------------------------------------------------ paths
    def _meta_root(self):
        return self._storage.path_join(self.flow_name, "_meta")

    def _run_path(self, run_id):
        return self._storage.path_join(self._meta_root(), str(run_id),
                                       "run.json")

    def _task_path(self, run_id, step_name, task_id):
        return self._storage.path_join(
            self._meta_root(), str(run_id),
            "task.%s.%s.json" % (step_name, task_id))

    def _heartbeat_path(self, run_id):
        return self._storage.path_join(self._meta_root(), str(run_id),
                                       "heartbeat.json")

    def _save(self, path, obj):
        self._storage.save_bytes(
            iter([(path, (json.dumps(obj).encode("utf-8"), None))]),
            overwrite=True)

    def _load(self, path):
        for _p, blob, _m in self._storage.load_bytes([path]):
            if blob is None:
                return None
            try:
                return json.loads(blob.decode("utf-8"))
            except ValueError:
                return None
        return None

    # ------------------------------------------------------------------- ids
    def new_run_id(self, tags=None):
        run_id = str(int(time.time() * 1e6))
        self.register_run(run_id, tags)
        return run_id

    # -------------------------------------------------------------- register
    def register_run(self, run_id, tags=None, origin_run_id=None):
        self._save(self._run_path(run_id), {
            "flow": self.flow_name,
            "run_id": str(run_id),
            "user": _username(),
            "tags": sorted(tags or []),
            "origin_run_id": origin_run_id,
            "ts_epoch": time.time(),
            "status": "running",
        })

    def update_run_info(self, run_id, extra):
        info = self._load(self._run_path(run_id)) or {}
        info.update(extra)
        self._save(self._run_path(run_id), info)

    def register_run_done(self, run_id, success):
        info = self._load(self._run_path(run_id)) or {}
        info["status"] = "successful" if success else "failed"
        info["finished_at"] = time.time()
        self._save(self._run_path(run_id), info)

    def register_task(self, run_id, step_name, task_id, attempt=0,
                      metadata=None):
        path = self._task_path(run_id, step_name, task_id)
        info = self._load(path) or {
            "flow": self.flow_name,
            "run_id": str(run_id),
            "step_name": step_name,
            "task_id": str(task_id),
            "user": _username(),
            "ts_epoch": time.time(),
            "attempts": {},
        }
        att = info["attempts"].setdefault(str(attempt), {})
        att["ts_epoch"] = time.time()
        if metadata:
            att.setdefault("metadata", {}).update(metadata)
        self._save(path, info)

    def register_metadata(self, run_id, step_name, task_id, attempt,
                          metadata):
        self.register_task(run_id, step_name, task_id, attempt, metadata)

    def heartbeat(self, run_id, step_name=None, task_id=None):
        """Run-level liveness; with step_name/task_id, task-level too
        (reference: heartbeats for task+run, heartbeat.py:21)."""
        self._save(self._heartbeat_path(run_id), {"ts": time.time()})
        if step_name and task_id:
            self._save(self._task_heartbeat_path(run_id, step_name,
                                                 task_id),
                       {"ts": time.time()})

    def _task_heartbeat_path(self, run_id, step_name, task_id):
        return self._storage.path_join(
            self._meta_root(), str(run_id),
            "task_heartbeat_%s_%s.json" % (step_name, task_id))

    def task_heartbeat_ts(self, run_id, step_name, task_id):
        hb = self._load(self._task_heartbeat_path(run_id, step_name,
                                                  task_id)) or {}
        return hb.get("ts")

    # ----------------------------------------------------------------- query
    @staticmethod
    def _run_sort_key(name):
        # run ids are creation-time microseconds (resume runs prefix the
        # origin id with "resume"), so DIRECTORY NAMES order runs
        # chronologically without opening any run.json
        digits = name[6:] if name.startswith("resume") else name
        try:
            return int(digits)
        except ValueError:
            return 0

    def iter_runs(self, limit=None):
        """Newest-first run records, loading at most ``limit`` run.json
        files: ordering comes from the directory names alone (µs
        timestamps), so a flow with a million runs pays for the page it
        reads, not the history (reference metadata.py:352 query
        routing / paging)."""
        names = [self._storage.basename(path)
                 for path, is_file in
                 self._storage.list_content([self._meta_root()])
                 if not is_file]
        names.sort(key=self._run_sort_key, reverse=True)
        count = 0
        for name in names:
            if limit is not None and count >= limit:
                return
            info = self._load(self._storage.path_join(
                self._meta_root(), name, "run.json"))
            if info:
                count += 1
                yield info

    def list_runs(self, limit=None):
        return list(self.iter_runs(limit=limit))

    def get_run(self, run_id):
        return self._load(self._run_path(run_id))

    def get_task(self, run_id, step_name, task_id):
        return self._load(self._task_path(run_id, step_name, task_id))

    def list_tasks(self, run_id, step_name=None):
        out = []
        base = self._storage.path_join(self._meta_root(), str(run_id))
        for path, is_file in self._storage.list_content([base]):
            name = self._storage.basename(path)
            if is_file and name.startswith("task."):
                info = self._load(path)
                if info and (step_name is None
                             or info["step_name"] == step_name):
                    out.append(info)
        return out

    def add_run_tags(self, run_id, tags):
        info = self._load(self._run_path(run_id)) or {}
        info["tags"] = sorted(set(info.get("tags", [])) | set(tags))
        self._save(self._run_path(run_id), info)

    def remove_run_tags(self, run_id, tags):
        info = self._load(self._run_path(run_id)) or {}
        info["tags"] = sorted(set(info.get("tags", [])) - set(tags))
        self._save(self._run_path(run_id), info)

    def replace_run_tags(self, run_id, removals, additions):
        """Atomic remove+add (reference client/core.py replace_tag): one
        read-modify-write so a concurrent reader never sees the
        intermediate state."""
        info = self._load(self._run_path(run_id)) or {}
        info["tags"] = sorted(
            (set(info.get("tags", [])) - set(removals)) | set(additions))
        self._save(self._run_path(run_id), info)


def _username():
    try:
        return getpass.getuser()
    except Exception:
        return os.environ.get("USER", "unknown")


METADATA_PROVIDERS = {"local": LocalMetadataProvider}
