"""Client API: read-only object tree Metaflow -> Flow -> Run -> Step -> Task
-> DataArtifact with namespace filtering.

Parity target: /root/reference/metaflow/client/core.py:227-2765. Backed
directly by the local datastore + metadata files.
"""

import os

from ..datastore import FlowDataStore
from ..datastore.storage import LocalStorage
from ..exceptions import NamespaceMismatchError, NotFoundError
from ..metadata.local import LocalMetadataProvider, _username

_current_namespace = "user:%s" % _username()


def namespace(ns):
    """Set the active namespace; None disables filtering. Returns it."""
    global _current_namespace
    _current_namespace = ns
    return ns


def get_namespace():
    return _current_namespace


def default_namespace():
    """Reset to the user's own namespace; returns it (reference
    client/core.py default_namespace)."""
    global _current_namespace
    _current_namespace = "user:%s" % _username()
    return _current_namespace


def get_metadata():
    """Describe the active metadata provider as "TYPE@location"
    (reference get_metadata)."""
    return "local@%s" % _datastore_root()


def default_metadata():
    """Reset metadata selection to the default local provider;
    returns its description (reference default_metadata)."""
    return get_metadata()


def default_namespace():
    return namespace("user:%s" % _username())


def _datastore_root():
    root = os.environ.get("MFX_DATASTORE_SYSROOT_LOCAL")
    if root:
        return root
    return LocalStorage.get_datastore_root_from_config(create_on_absent=False)


def _storage():
    return LocalStorage(_datastore_root())


def _check_namespace(tags):
    if _current_namespace is None:
        return True
    return _current_namespace in (tags or [])


class MetaflowObject(object):
    def __repr__(self):
        return "%s('%s')" % (type(self).__name__, self.pathspec)


class Metaflow(MetaflowObject):
    """Entry point: all flows in the datastore."""

    @property
    def flows(self):
        return list(self)

    def __iter__(self):
        storage = _storage()
        root = storage.root
        if root is None or not os.path.isdir(root):
            return
        for name in sorted(os.listdir(root)):
            if name.startswith(".") or name.startswith("_"):
                continue
            if os.path.isdir(os.path.join(root, name)):
                yield Flow(name)

    def __repr__(self):
        return "Metaflow()"


class Flow(MetaflowObject):
    def __init__(self, name):
        self.id = name
        self.pathspec = name
        self._storage = _storage()
        self._fds = FlowDataStore(name, self._storage)
        self._meta = LocalMetadataProvider(name, self._storage)

    @property
    def latest_run(self):
        for run in self:
            return run
        return None

    @property
    def latest_successful_run(self):
        for run in self:
            if run.successful:
                return run
        return None

    def runs(self, *tags, limit=None):
        """Newest-first runs with all the given tags; ``limit`` bounds
        how many matching runs are yielded (the metadata provider pages
        lazily, so a deep history costs what you read)."""
        n = 0
        for run in self:
            if all(t in run.tags for t in tags):
                yield run
                n += 1
                if limit is not None and n >= limit:
                    return

    def __iter__(self):
        iter_fn = getattr(self._meta, "iter_runs", self._meta.list_runs)
        for info in iter_fn():
            tags = _full_tags(info)
            if _check_namespace(tags):
                yield Run("%s/%s" % (self.id, info["run_id"]), _info=info)

    def __getitem__(self, run_id):
        return Run("%s/%s" % (self.id, run_id))


def _full_tags(info):
    tags = list(info.get("tags", []))
    user = info.get("user")
    if user:
        tags.append("user:%s" % user)
    return tags


class Run(MetaflowObject):
    def __init__(self, pathspec, _info=None):
        parts = pathspec.split("/")
        if len(parts) != 2:
            raise NotFoundError("Run pathspec must be 'Flow/run_id'.")
        self.flow_name, self.id = parts
        self.pathspec = pathspec
        self._storage = _storage()
        self._fds = FlowDataStore(self.flow_name, self._storage)
        self._meta = LocalMetadataProvider(self.flow_name, self._storage)
        self._info = _info or self._meta.get_run(self.id)
        if self._info is None:
            raise NotFoundError("Run %s not found." % pathspec)
        if not _check_namespace(_full_tags(self._info)):
            raise NamespaceMismatchError(_current_namespace)

    @property
    def tags(self):
        return _full_tags(self._info)

    @property
    def successful(self):
        info = self._meta.get_run(self.id) or {}
        if info.get("status") == "successful":
            return True
        # fall back to the end task's DONE marker
        try:
            return self["end"].task.successful
        except Exception:
            return False

    @property
    def finished(self):
        info = self._meta.get_run(self.id) or {}
        return info.get("status") in ("successful", "failed")

    @property
    def is_alive(self):
        """True if the run is unfinished and heartbeat is fresh (<60 s)."""
        import time

        if self.finished:
            return False
        hb = self._meta._load(self._meta._heartbeat_path(self.id)) or {}
        return time.time() - hb.get("ts", 0) < 60

    @property
    def data(self):
        end = self["end"].task
        return end.data if end else None

    def steps(self):
        return list(self)

    def __iter__(self):
        for step_name in self._fds.list_steps(self.id):
            if step_name.startswith("_"):
                continue
            yield Step("%s/%s" % (self.pathspec, step_name))

    def __getitem__(self, step_name):
        steps = self._fds.list_steps(self.id)
        if step_name not in steps:
            raise NotFoundError("Step %s/%s not found."
                                % (self.pathspec, step_name))
        return Step("%s/%s" % (self.pathspec, step_name))

    def add_tags(self, tags):
        self._meta.add_run_tags(self.id, tags)
        self._info = self._meta.get_run(self.id)

    def add_tag(self, tag):
        self.add_tags([tag])

    def remove_tags(self, tags):
        self._meta.remove_run_tags(self.id, tags)
        self._info = self._meta.get_run(self.id)

    def remove_tag(self, tag):
        self.remove_tags([tag])

    def replace_tags(self, removals, additions):
        """Atomically remove and add tags (reference Run.replace_tags)."""
        replace = getattr(self._meta, "replace_run_tags", None)
        if replace is not None:
            replace(self.id, removals, additions)
            self._info = self._meta.get_run(self.id)
        else:
            self.remove_tags(removals)
            self.add_tags(additions)

    def replace_tag(self, removal, addition):
        self.replace_tags([removal], [addition])

    @property
    def end_task(self):
        try:
            return self["end"].task
        except NotFoundError:
            return None

    @property
    def dag(self):
        """The run's static graph as persisted at execution time
        ({step: {type, in_funcs, out_funcs, ...}} — reference
        runtime_dag behavior: the DAG travels with the run)."""
        try:
            params = self._fds.get_task_datastore(self.id,
                                                  "_parameters", "0")
            return params.get("_graph_info")
        except Exception:
            return None

    @property
    def code_package_key(self):
        return (self._meta.get_run(self.id) or {}).get("code_package_key")

    def extract_code(self, dest):
        """Extract this run's code snapshot into dest (reproducibility)."""
        from ..package import CodePackage

        key = self.code_package_key
        if key is None:
            raise NotFoundError("Run %s has no code package." % self.pathspec)
        return CodePackage.extract(self._fds, key, dest)


class Step(MetaflowObject):
    def __init__(self, pathspec):
        parts = pathspec.split("/")
        if len(parts) != 3:
            raise NotFoundError("Step pathspec must be 'Flow/run/step'.")
        self.flow_name, self.run_id, self.id = parts
        self.pathspec = pathspec
        self._storage = _storage()
        self._fds = FlowDataStore(self.flow_name, self._storage)

    @property
    def task(self):
        """The first (often only) task."""
        for t in self:
            return t
        return None

    def tasks(self):
        return list(self)

    def __iter__(self):
        task_ids = self._fds.list_tasks(self.run_id, self.id)

        def sort_key(t):
            main = t.split("_node_")[0]
            node = t.split("_node_")[1] if "_node_" in t else "0"
            try:
                return (int(main), int(node))
            except ValueError:
                return (1 << 30, 0)

        for task_id in sorted(task_ids, key=sort_key):
            yield Task("%s/%s" % (self.pathspec, task_id))

    def __getitem__(self, task_id):
        return Task("%s/%s" % (self.pathspec, task_id))

    @property
    def parent(self):
        return Run("%s/%s" % (self.flow_name, self.run_id))

    @property
    def control_task(self):
        for t in self:
            if "_node_" not in t.id:
                return t
        return None


class Task(MetaflowObject):
    def __init__(self, pathspec):
        parts = pathspec.split("/")
        if len(parts) != 4:
            raise NotFoundError("Task pathspec must be 'Flow/run/step/task'.")
        self.flow_name, self.run_id, self.step_name, self.id = parts
        self.pathspec = pathspec
        self._storage = _storage()
        self._fds = FlowDataStore(self.flow_name, self._storage)
        self._ds = self._fds.get_task_datastore(self.run_id, self.step_name,
                                                self.id)
        if os.environ.get("MFX_CLIENT_CACHE", "0") == "1":
            from .filecache import FileBlobCache

            self._fds.ca_store.set_blob_cache(FileBlobCache())

    @property
    def successful(self):
        if self._ds.attempt is None:
            return False
        ok = self._ds.load_metadata("attempt_ok") or {}
        return bool(ok.get("ok"))

    @property
    def finished(self):
        return self._ds.attempt is not None

    @property
    def is_alive(self):
        """True if this task is mid-execution: not finished and its
        task-level heartbeat is fresh (<60 s)."""
        import time

        if self.finished:
            return False
        from ..metadata.local import LocalMetadataProvider

        meta = LocalMetadataProvider(self.flow_name, self._storage)
        ts = meta.task_heartbeat_ts(self.run_id, self.step_name, self.id)
        return ts is not None and time.time() - ts < 60

    @property
    def exception(self):
        return self._ds.get("_exception")

    @property
    def data(self):
        return MetaflowData(self._ds)

    @property
    def artifacts(self):
        return MetaflowData(self._ds)

    @property
    def index(self):
        frames = self._ds.load_metadata("foreach_stack") or []
        return frames[-1][3] if frames else None

    @property
    def foreach_stack(self):
        return self._ds.load_metadata("foreach_stack") or []

    @staticmethod
    def _plain(raw):
        from ..mflog import parse

        out = []
        for line in raw.splitlines():
            p = parse(line)
            out.append(p.msg if p else line)
        return "\n".join(out) + ("\n" if out else "")

    @property
    def stdout(self):
        return self._plain(self._ds.load_logs("stdout"))

    @property
    def stderr(self):
        return self._plain(self._ds.load_logs("stderr"))

    def loglines(self, stream="stdout"):
        """Structured MFLogline records for this task."""
        from ..mflog import merge_logs

        raw = self._ds.load_logs(stream).encode()
        return list(merge_logs([raw]))

    @property
    def card_html(self):
        """The rendered @card HTML for this task (with the scheduler's
        rocprof kernel-breakdown spliced in when @card(profile=True)),
        or None."""
        from ..plugins.card_decorator import get_card

        return get_card(self._ds)

    @property
    def parent(self):
        return Step("%s/%s/%s" % (self.flow_name, self.run_id,
                                  self.step_name))

    @property
    def parent_tasks(self):
        """Tasks whose outputs fed this task (foreach-aware: matched by
        graph in_funcs + foreach-stack prefix)."""
        graph = self._graph_info()
        if graph is None:
            return []
        node = graph.get(self.step_name, {})
        my_stack = self.foreach_stack
        out = []
        for in_step in node.get("in_funcs", []):
            for t in Step("%s/%s/%s" % (self.flow_name, self.run_id,
                                        in_step)):
                ts = t.foreach_stack
                if ts == my_stack or ts == my_stack[:-1] \
                        or my_stack == ts[:-1] or ts[:len(my_stack)] \
                        == my_stack:
                    out.append(t)
        return out

    @property
    def child_tasks(self):
        graph = self._graph_info()
        if graph is None:
            return []
        node = graph.get(self.step_name, {})
        my_stack = self.foreach_stack
        out = []
        for out_step in node.get("out_funcs", []):
            for t in Step("%s/%s/%s" % (self.flow_name, self.run_id,
                                        out_step)):
                ts = t.foreach_stack
                if ts == my_stack or my_stack == ts[:-1] \
                        or ts == my_stack[:-1] \
                        or my_stack[:len(ts)] == ts:
                    out.append(t)
        return out

    def _graph_info(self):
        try:
            params = self._fds.get_task_datastore(self.run_id,
                                                  "_parameters", "0")
            return params.get("_graph_info")
        except Exception:
            return None

    def __iter__(self):
        for name in self._ds.artifact_names():
            if not name.startswith("_"):
                yield DataArtifact(self, name)

    def __getitem__(self, name):
        if name not in self._ds:
            raise NotFoundError("Artifact %s/%s not found."
                                % (self.pathspec, name))
        return DataArtifact(self, name)


class MetaflowData(object):
    """Attribute access to a task's artifacts (task.data.model)."""

    def __init__(self, task_ds):
        object.__setattr__(self, "_ds", task_ds)

    def __getattr__(self, name):
        ds = object.__getattribute__(self, "_ds")
        if name in ds:
            return ds[name]
        raise AttributeError(name)

    def __contains__(self, name):
        return name in object.__getattribute__(self, "_ds")

    def _artifacts(self):
        return object.__getattribute__(self, "_ds").artifact_names()

    def __repr__(self):
        names = [n for n in self._artifacts() if not n.startswith("_")]
        return "<MetaflowData: %s>" % ", ".join(sorted(names))


class DataArtifact(MetaflowObject):
    def __init__(self, task, name):
        self._task = task
        self.id = name
        self.pathspec = "%s/%s" % (task.pathspec, name)

    @property
    def data(self):
        return self._task._ds[self.id]

    @property
    def sha(self):
        return self._task._ds.artifact_sha(self.id)

    @property
    def size(self):
        info = self._task._ds.artifact_info(self.id) or {}
        return info.get("size")
