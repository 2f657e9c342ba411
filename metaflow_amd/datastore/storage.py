"""Byte-level storage backends.

Parity target: /root/reference/metaflow/datastore/datastore_storage.py:26
(DataStoreStorage ABC) and plugins/datastores/local_storage.py:12
(atomic tmp+rename writes). Batch-oriented contract so cloud backends can
pipeline; the local impl is the primary backend on a single MI355X node.
"""

import json
import os
import tempfile


class DataStoreStorage(object):
    TYPE = None

    def __init__(self, root=None):
        self.root = root

    # --- path helpers ---------------------------------------------------
    @classmethod
    def path_join(cls, *components):
        return "/".join(c.strip("/") for c in components if c)

    @classmethod
    def basename(cls, path):
        return path.rsplit("/", 1)[-1]

    # --- contract ---------------------------------------------------------
    def is_file(self, paths):
        """[bool] per path."""
        raise NotImplementedError

    def size_file(self, path):
        raise NotImplementedError

    def list_content(self, paths):
        """Yield (path, is_file) for each direct child of each path."""
        raise NotImplementedError

    def save_bytes(self, path_and_bytes_iter, overwrite=False):
        """path_and_bytes_iter yields (path, (bytes, metadata_dict|None))."""
        raise NotImplementedError

    def load_bytes(self, paths):
        """Yield (path, bytes_or_None, metadata_dict_or_None)."""
        raise NotImplementedError

    def full_uri(self, path):
        return self.path_join(self.root, path)


class LocalStorage(DataStoreStorage):
    TYPE = "local"

    @classmethod
    def get_datastore_root_from_config(cls, create_on_absent=True):
        from ..config import DATASTORE_LOCAL_DIR, from_conf

        root = from_conf("DATASTORE_SYSROOT_LOCAL")
        if root is None:
            # walk up from cwd looking for an existing .mfx dir, like the
            # reference does for .metaflow; else create in cwd
            cur = os.getcwd()
            while True:
                candidate = os.path.join(cur, DATASTORE_LOCAL_DIR)
                if os.path.isdir(candidate):
                    return candidate
                parent = os.path.dirname(cur)
                if parent == cur:
                    break
                cur = parent
            root = os.path.join(os.getcwd(), DATASTORE_LOCAL_DIR)
            if create_on_absent:
                os.makedirs(root, exist_ok=True)
        return root

    def _abs(self, path):
        return os.path.join(self.root, path.lstrip("/"))

    def is_file(self, paths):
        return [os.path.isfile(self._abs(p)) for p in paths]

    def size_file(self, path):
        try:
            return os.path.getsize(self._abs(path))
        except OSError:
            return None

    def info_file(self, path):
        ap = self._abs(path)
        if not os.path.isfile(ap):
            return False, None
        meta = None
        meta_path = ap + "_meta"
        if os.path.isfile(meta_path):
            try:
                with open(meta_path) as f:
                    meta = json.load(f)
            except Exception:
                meta = None
        return True, meta

    def list_content(self, paths):
        for path in paths:
            ap = self._abs(path)
            if not os.path.isdir(ap):
                continue
            for entry in sorted(os.listdir(ap)):
                if entry.endswith("_meta"):
                    continue
                full = os.path.join(ap, entry)
                yield self.path_join(path, entry), os.path.isfile(full)

    def save_bytes(self, path_and_bytes_iter, overwrite=False):
        for path, payload in path_and_bytes_iter:
            if isinstance(payload, tuple):
                data, metadata = payload
            else:
                data, metadata = payload, None
            ap = self._abs(path)
            if not overwrite and os.path.exists(ap):
                continue
            os.makedirs(os.path.dirname(ap), exist_ok=True)
            # atomic write: tmp in the same dir + rename
            fd, tmp = tempfile.mkstemp(dir=os.path.dirname(ap),
                                       prefix=".tmp_")
            try:
                with os.fdopen(fd, "wb") as f:
                    if hasattr(data, "read"):
                        while True:
                            chunk = data.read(1 << 20)
                            if not chunk:
                                break
                            f.write(chunk)
                    elif isinstance(data, (list, tuple)):
                        # scatter write: avoids concatenating header +
                        # payload (zero-copy path for large blobs)
                        for part in data:
                            f.write(part)
                    else:
                        f.write(data)
                os.replace(tmp, ap)
            except Exception:
                try:
                    os.unlink(tmp)
                except OSError:
                    pass
                raise
            if metadata is not None:
                with open(ap + "_meta", "w") as f:
                    json.dump(metadata, f)

    def load_bytes(self, paths):
        for path in paths:
            ap = self._abs(path)
            if os.path.isfile(ap):
                with open(ap, "rb") as f:
                    data = f.read()
                meta = None
                if os.path.isfile(ap + "_meta"):
                    try:
                        with open(ap + "_meta") as f:
                            meta = json.load(f)
                    except Exception:
                        meta = None
                yield path, data, meta
            else:
                yield path, None, None

    def create_exclusive(self, path, data):
        """Atomically create a file; returns True iff WE created it (the
        leader-election primitive: O_CREAT|O_EXCL)."""
        ap = self._abs(path)
        os.makedirs(os.path.dirname(ap), exist_ok=True)
        try:
            fd = os.open(ap, os.O_CREAT | os.O_EXCL | os.O_WRONLY, 0o644)
        except FileExistsError:
            return False
        with os.fdopen(fd, "wb") as f:
            f.write(data)
        return True

    def load_file_path(self, path):
        """Local fast path: return the filesystem path for a key (lets the
        native CAS engine mmap/pread instead of round-tripping bytes)."""
        ap = self._abs(path)
        return ap if os.path.isfile(ap) else None


STORAGE_IMPLS = {"local": LocalStorage}
