"""Bulk object IO: the local analog of the reference's S3 datatool.

Parity target: /root/reference/metaflow/plugins/datatools/s3/ (S3.get_many/
put_many/put_files with a worker pool). On one MI355X node the backend is
the local filesystem (or any DataStoreStorage); the value is the same
batched, parallel API — thread-pooled reads/writes with retry — so user
steps can move many objects at device-filling throughput.

    with ObjectStore(root="/data/cache") as store:
        store.put_many([("a", b"..."), ("b", b"...")])
        blobs = store.get_many(["a", "b"])
        store.put_files([("model.bin", "/tmp/model.bin")])
"""

import os
import random
import shutil
import time
from collections import namedtuple
from concurrent.futures import ThreadPoolExecutor

from .exceptions import MFXException

StoredObject = namedtuple("StoredObject", ["key", "path", "size"])


def _maybe_inject_failure():
    """Test hook (reference parity: s3op --inject-failure): when
    MFX_INJECT_IO_FAILURES=P (0..100), IO ops raise a transient OSError
    with probability P% — exercised by the retry machinery tests."""
    p = float(os.environ.get("MFX_INJECT_IO_FAILURES", "0"))
    if p > 0 and random.random() * 100 < p:
        raise OSError("injected transient IO failure")


class ObjectStore(object):
    def __init__(self, root, max_workers=None, retries=3):
        self.root = os.path.abspath(root)
        os.makedirs(self.root, exist_ok=True)
        self._pool = ThreadPoolExecutor(
            max_workers=max_workers or min(32, (os.cpu_count() or 8) * 2))
        self._retries = retries

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self._pool.shutdown(wait=True)
        return False

    def _path(self, key):
        p = os.path.normpath(os.path.join(self.root, key.lstrip("/")))
        if not p.startswith(self.root):
            raise MFXException("Key escapes the store root: %r" % key)
        return p

    def _with_retries(self, fn, *args):
        err = None
        for attempt in range(self._retries):
            try:
                _maybe_inject_failure()
                return fn(*args)
            except OSError as e:
                err = e
                time.sleep(0.05 * (2 ** attempt))
        raise MFXException("IO failed after %d retries: %s"
                           % (self._retries, err))

    # ------------------------------------------------------------------ puts
    def put(self, key, blob):
        def do(k, b):
            path = self._path(k)
            os.makedirs(os.path.dirname(path), exist_ok=True)
            tmp = path + ".tmp%d" % os.getpid()
            with open(tmp, "wb") as f:
                f.write(b)
            os.replace(tmp, path)
            return StoredObject(k, path, len(b))

        return self._with_retries(do, key, blob)

    def put_many(self, key_blob_pairs):
        futures = [self._pool.submit(self.put, k, b)
                   for k, b in key_blob_pairs]
        return [f.result() for f in futures]

    def put_files(self, key_path_pairs):
        def do(k, src):
            path = self._path(k)
            os.makedirs(os.path.dirname(path), exist_ok=True)
            tmp = path + ".tmp%d" % os.getpid()
            shutil.copyfile(src, tmp)
            os.replace(tmp, path)
            return StoredObject(k, path, os.path.getsize(path))

        futures = [self._pool.submit(self._with_retries, do, k, p)
                   for k, p in key_path_pairs]
        return [f.result() for f in futures]

    # ------------------------------------------------------------------ gets
    def get(self, key):
        def do(k):
            path = self._path(k)
            # large objects go through the native parallel-pread engine
            # (cas_engine.cpp: 16 MiB chunks across a thread pool,
            # single allocation) — Python's f.read() is single-threaded
            try:
                if os.path.getsize(path) >= (32 << 20):
                    from .ops import cas_native

                    return cas_native.engine().load_blob_parallel(path, 0)
            except ImportError:
                pass
            with open(path, "rb") as f:
                return f.read()

        return self._with_retries(do, key)

    def get_many(self, keys):
        futures = [self._pool.submit(self.get, k) for k in keys]
        return [f.result() for f in futures]

    def list_paths(self, prefix=""):
        base = self._path(prefix) if prefix else self.root
        out = []
        for dirpath, _dirs, names in os.walk(base):
            for n in names:
                if n.endswith(".tmp%d" % os.getpid()):
                    continue
                full = os.path.join(dirpath, n)
                out.append(StoredObject(
                    os.path.relpath(full, self.root), full,
                    os.path.getsize(full)))
        return sorted(out)

    def info(self, key):
        path = self._path(key)
        if not os.path.exists(path):
            return None
        return StoredObject(key, path, os.path.getsize(path))
