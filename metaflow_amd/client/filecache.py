"""LRU on-disk blob cache for the Client API.

Parity target: /root/reference/metaflow/client/filecache.py (FileCache /
FileBlobCache). Caches CAS blobs under ~/.mfx_cache keyed by sha, with a
size-bounded LRU eviction; plugged into ContentAddressedStore via
set_blob_cache so repeated client reads skip the datastore.
"""

import os

DEFAULT_MAX_BYTES = 2 << 30  # 2 GiB


class FileBlobCache(object):
    def __init__(self, root=None, max_bytes=DEFAULT_MAX_BYTES):
        self.root = root or os.path.expanduser("~/.mfx_cache/blobs")
        os.makedirs(self.root, exist_ok=True)
        self.max_bytes = max_bytes

    def _path(self, key):
        return os.path.join(self.root, key[:2], key)

    def load_key(self, key):
        path = self._path(key)
        try:
            with open(path, "rb") as f:
                data = f.read()
            os.utime(path)  # LRU touch
            return data
        except OSError:
            return None

    def store_key(self, key, blob):
        path = self._path(key)
        if os.path.exists(path):
            return
        os.makedirs(os.path.dirname(path), exist_ok=True)
        tmp = path + ".tmp%d" % os.getpid()
        try:
            with open(tmp, "wb") as f:
                f.write(blob)
            os.replace(tmp, path)
        except OSError:
            try:
                os.unlink(tmp)
            except OSError:
                pass
            return
        self._evict()

    def _evict(self):
        entries = []
        total = 0
        for dirpath, _d, names in os.walk(self.root):
            for n in names:
                p = os.path.join(dirpath, n)
                try:
                    st = os.stat(p)
                except OSError:
                    continue
                entries.append((st.st_mtime, st.st_size, p))
                total += st.st_size
        if total <= self.max_bytes:
            return
        entries.sort()  # oldest first
        for _mtime, size, p in entries:
            if total <= self.max_bytes:
                break
            try:
                os.unlink(p)
                total -= size
            except OSError:
                pass

    def clear(self):
        import shutil

        shutil.rmtree(self.root, ignore_errors=True)
        os.makedirs(self.root, exist_ok=True)
