"""Content-addressed blob store.

Parity target: /root/reference/metaflow/datastore/content_addressed_store.py
(save_blobs :41, load_blobs :122, _pack_v1 gzip :211, dedup-by-existence
:107). MI355X-first differences:

* hash is SHA-256 of the RAW content (dedup is codec-independent);
* blobs are self-describing: an 8-byte header ``MFXB <ver> <codec> <pad>``
  precedes the payload, so mixed codecs coexist in one store;
* large blobs (>= CAS_COMPRESS_MAX_SIZE, default 4 MiB) skip gzip — tensor
  shards are incompressible and compressing them would bottleneck the
  artifact-GB/s path;
* when the native C++ engine (_mfx_cas: multi-threaded SHA-256 + direct
  file writes) is importable and the backing storage is the local
  filesystem, save/load of large blobs route through it.
"""

import gzip
import hashlib
import os
import zlib

from ..config import CAS_COMPRESS_MAX_SIZE, CAS_GZIP_LEVEL, CAS_NATIVE

MAGIC = b"MFXB"
CODEC_RAW = 0
CODEC_GZIP = 1
HEADER_LEN = 8

# blobs >= this hash as a Merkle root over 8 MiB leaves across a thread pool
# (hashlib releases the GIL; memoryview slices avoid copies: ~6 GB/s on 8
# cores vs ~1 GB/s single-threaded)
PARALLEL_KEY_MIN = 16 << 20
_LEAF = 8 << 20
_hash_pool = None


def parallel_key(blob):
    global _hash_pool
    if len(blob) < PARALLEL_KEY_MIN:
        return hashlib.sha256(blob).hexdigest()
    if _hash_pool is None:
        from concurrent.futures import ThreadPoolExecutor

        _hash_pool = ThreadPoolExecutor(max_workers=os.cpu_count() or 8)
    mv = memoryview(blob)
    n = (len(blob) + _LEAF - 1) // _LEAF

    def leaf(i):
        return hashlib.sha256(mv[i * _LEAF:(i + 1) * _LEAF]).digest()

    digests = b"".join(_hash_pool.map(leaf, range(n)))
    return hashlib.sha256(b"MFXP1" + digests).hexdigest()


class _IoEngine(object):
    """load-only engine backed by the STANDALONE _mfx_io extension (no
    torch link, ~1 ms import): torch-less task subprocesses get the
    parallel-pread read path without paying the 1.5 s torch import the
    full _mfx_cas engine would drag in (ops/cas_native.py)."""

    __slots__ = ("_mod",)

    def __init__(self, mod):
        self._mod = mod

    def load_blob_parallel(self, path, header_skip):
        return self._mod.load_file(path, header_skip)


def _native_engine():
    if not CAS_NATIVE:
        return None
    try:
        from ..ops import cas_native

        return cas_native.engine()
    except Exception:
        pass
    try:
        from ..ops import _mfx_io

        return _IoEngine(_mfx_io)
    except Exception:
        return None


def pack(data, codec):
    header = MAGIC + bytes([1, codec, 0, 0])
    if codec == CODEC_GZIP:
        return header + zlib.compress(data, CAS_GZIP_LEVEL)
    return header + data


def unpack(blob):
    if blob[:4] != MAGIC:
        # legacy/foreign blob: try gzip, else raw (reference behavior)
        try:
            return gzip.decompress(blob)
        except OSError:
            return blob
    codec = blob[5]
    payload = blob[HEADER_LEN:]
    if codec == CODEC_GZIP:
        return zlib.decompress(payload)
    return payload


class ContentAddressedStore(object):
    def __init__(self, prefix, storage):
        self._prefix = prefix
        self._storage = storage
        self._blob_cache = None

    def set_blob_cache(self, cache):
        self._blob_cache = cache

    def _key_path(self, key):
        return self._storage.path_join(self._prefix, key[:2], key)

    @staticmethod
    def compute_key(data):
        return hashlib.sha256(data).hexdigest()

    def save_blobs(self, blob_iter, raw=False, len_hint=0):
        """Save an iterable of bytes objects; returns list of (uri, key).

        Dedup: existing keys are never rewritten (content-addressed keys are
        write-once, which is what makes resume/clone metadata-only).
        """
        results = []
        to_save = []
        for blob in blob_iter:
            # keys are opaque; loads always use the key stored in the
            # artifact index, so the Merkle fast path is safe
            key = parallel_key(blob)
            path = self._key_path(key)
            results.append((self._storage.full_uri(path), key))
            to_save.append((key, path, blob))

        # existence check first (dedup)
        exists = self._storage.is_file([p for _, p, _ in to_save])

        def _packed():
            raw_header = MAGIC + bytes([1, CODEC_RAW, 0, 0])
            for (key, path, blob), present in zip(to_save, exists):
                if present:
                    continue
                big = raw or len(blob) >= CAS_COMPRESS_MAX_SIZE
                if big:
                    # zero-copy: header + payload written separately
                    yield path, ([raw_header, blob], None)
                else:
                    yield path, (pack(bytes(blob), CODEC_GZIP), None)

        self._storage.save_bytes(_packed(), overwrite=False)
        return results

    def save_stream(self, chunk_iter, nbytes):
        """Streaming raw-blob save: consume ``chunk_iter`` (bytes-like
        chunks, 8 MiB-aligned except the last) while hashing Merkle
        leaves in a thread pool and appending to a tmp file — the
        checkpoint path's D2H copies, hashing and file writes all
        overlap instead of running serially (config 4's GB/s metric).
        Returns (uri, key); identical content yields the same key as
        save_blobs (leaf convention shared with parallel_key), and a
        dedup hit just drops the tmp file.

        Local-filesystem storage only; callers fall back to save_blobs
        elsewhere."""
        import hashlib
        import os as _os

        global _hash_pool
        if _hash_pool is None:
            from concurrent.futures import ThreadPoolExecutor

            _hash_pool = ThreadPoolExecutor(
                max_workers=_os.cpu_count() or 8)

        import queue
        import threading

        root_dir = self._storage._abs(self._prefix)
        _os.makedirs(root_dir, exist_ok=True)
        tmp = _os.path.join(root_dir, ".stream.%d.%d.tmp"
                            % (_os.getpid(), id(chunk_iter)))
        futures = []
        small = nbytes < PARALLEL_KEY_MIN
        plain = hashlib.sha256() if small else None
        raw_header = MAGIC + bytes([1, CODEC_RAW, 0, 0])
        # dedicated writer thread: the leaf bytes objects are hashed by
        # the pool AND written by the writer, so the main thread's only
        # serial work is the leaf copy out of the caller's (pinned)
        # buffer — write, hash, copy and the caller's D2H all overlap
        wq = queue.Queue(maxsize=128)
        werr = []

        def _writer():
            try:
                with open(tmp, "wb") as f:
                    f.write(raw_header)
                    while True:
                        item = wq.get()
                        if item is None:
                            return
                        f.write(item)
            except Exception as ex:  # surfaced after join
                werr.append(ex)
                while wq.get() is not None:
                    pass

        wt = threading.Thread(target=_writer, daemon=True)
        wt.start()
        try:
            for chunk in chunk_iter:
                mv = memoryview(chunk)
                for off in range(0, len(mv), _LEAF):
                    # bytes() copy: the caller reuses its pinned halves
                    # once we return from this iteration
                    leaf = bytes(mv[off:off + _LEAF])
                    if small:
                        plain.update(leaf)
                    else:
                        futures.append(_hash_pool.submit(
                            lambda b: hashlib.sha256(b).digest(), leaf))
                    wq.put(leaf)
            wq.put(None)
            wt.join()
            if werr:
                raise werr[0]
            if small:
                key = plain.hexdigest()
            else:
                digests = b"".join(fu.result() for fu in futures)
                key = hashlib.sha256(b"MFXP1" + digests).hexdigest()
            path = self._key_path(key)
            ap = self._storage._abs(path)
            if _os.path.isfile(ap):
                _os.unlink(tmp)  # dedup: content already stored
            else:
                _os.makedirs(_os.path.dirname(ap), exist_ok=True)
                _os.replace(tmp, ap)
            return self._storage.full_uri(path), key
        except Exception:
            try:
                wq.put(None)
            except Exception:
                pass
            try:
                _os.unlink(tmp)
            except OSError:
                pass
            raise

    def blob_file(self, key):
        """Local raw-blob fast path: (filesystem_path, payload_offset) if
        this key is stored as an uncompressed MFXB file on local disk,
        else None. Lets large-tensor consumers (parallel/checkpoint.py)
        ``readinto`` pinned memory instead of round-tripping bytes."""
        path = self._storage.load_file_path(self._key_path(key))
        if path is None:
            return None
        try:
            with open(path, "rb") as f:
                hdr = f.read(HEADER_LEN)
        except OSError:
            return None
        if len(hdr) < HEADER_LEN or hdr[:4] != MAGIC or hdr[5] != CODEC_RAW:
            return None
        return path, HEADER_LEN

    def load_blobs(self, keys, force_raw=False):
        """Yield (key, bytes) for each key.

        Large raw blobs on local disk go through the native engine's
        parallel-pread path (one allocation, chunked pread across a
        thread pool) — the Python path pays a single-threaded read()
        plus a header-slice copy, which capped load at ~2.4 GB/s vs
        8 GB/s save in round 1 (profiles/bench_results_r01.md)."""
        missing = []
        for key in keys:
            hit = None
            if self._blob_cache is not None:
                hit = self._blob_cache.load_key(key)
            if hit is not None:
                yield key, hit
            else:
                missing.append(key)
        if not missing:
            return
        engine = _native_engine()
        slow = []
        if engine is not None:
            for key in missing:
                loc = self.blob_file(key)
                if loc is None:
                    slow.append(key)
                    continue
                path, off = loc
                data = engine.load_blob_parallel(path, off)
                if self._blob_cache is not None:
                    self._blob_cache.store_key(key, data)
                yield key, data
        else:
            slow = missing
        if not slow:
            return
        paths = {self._key_path(k): k for k in slow}
        for path, blob, _meta in self._storage.load_bytes(list(paths)):
            key = paths[path]
            if blob is None:
                from ..exceptions import DataArtifactMissingError

                raise DataArtifactMissingError(
                    "Blob %s not found in content-addressed store." % key)
            data = unpack(blob)
            if self._blob_cache is not None:
                self._blob_cache.store_key(key, data)
            yield key, data
