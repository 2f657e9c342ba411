"""Sharded tensor checkpointing through the content-addressed store.

BASELINE config 4: @checkpoint save/load of 70B random-init shards across
288 GB HBM x8. Design:

* one CAS blob per tensor holding the RAW storage bytes (no per-blob
  framing: dtype/shape live in the index artifact), so the save path is
  zero-copy from (pinned) host memory to the write syscall;
* content-hashed keys: an unchanged shard between two checkpoints dedups
  to a pure existence check — no bytes move;
* GPU tensors stage D2H through a reusable pinned buffer with
  non-blocking copies on a dedicated side stream, chunked so HBM-sized
  tensors never need a second host-sized intermediate;
* the index artifact (name -> {sha, dtype, shape}) is saved like any
  artifact, so `resume` and the Client API see checkpoints natively.
"""

_PIN_BUF_BYTES = 512 << 20  # pinned staging buffer for D2H
_pin_buf = None
_side_stream = None


def _get_pin_buf(torch):
    global _pin_buf
    if _pin_buf is None:
        try:
            _pin_buf = torch.empty(_PIN_BUF_BYTES, dtype=torch.uint8,
                                   pin_memory=True)
        except RuntimeError:
            _pin_buf = torch.empty(_PIN_BUF_BYTES, dtype=torch.uint8)
    return _pin_buf


def _tensor_to_buffer(t):
    """(memoryview_of_raw_bytes, dtype_str, shape). GPU tensors stream
    through the pinned staging buffer on a side stream."""
    import numpy as np
    import torch

    global _side_stream
    t = t.detach().contiguous()
    shape = tuple(t.shape)
    dtype = str(t.dtype).replace("torch.", "")
    if not t.is_cuda:
        arr = t.reshape(-1).view(torch.uint8).numpy()
        return memoryview(arr), dtype, shape

    nbytes = t.element_size() * t.numel()
    flat = t.reshape(-1).view(torch.uint8)
    host = np.empty(nbytes, dtype=np.uint8)
    pin = _get_pin_buf(torch)
    pin_np = pin.numpy()
    if _side_stream is None:
        _side_stream = torch.cuda.Stream()
    # order the side-stream copies after every kernel the producing
    # stream still has in flight (e.g. the fused AdamW of the step being
    # checkpointed) — without this the D2H can capture stale bytes. The
    # .contiguous() above also ran on the current stream, so one
    # wait_stream covers both.
    _side_stream.wait_stream(torch.cuda.current_stream())
    offset = 0
    with torch.cuda.stream(_side_stream):
        while offset < nbytes:
            n = min(_PIN_BUF_BYTES, nbytes - offset)
            pin[:n].copy_(flat[offset:offset + n], non_blocking=True)
            _side_stream.synchronize()
            host[offset:offset + n] = pin_np[:n]
            offset += n
    return memoryview(host), dtype, shape


def _gpu_chunks(t, torch):
    """Yield the tensor's bytes as pinned-host chunks: D2H of half i+1
    overlaps the caller's consumption of half i (hash+write in
    cas.save_stream). Chunks are 8 MiB-aligned (Merkle leaf contract)."""
    global _side_stream
    nbytes = t.element_size() * t.numel()
    flat = t.reshape(-1).view(torch.uint8)
    pin = _get_pin_buf(torch)
    half = _PIN_BUF_BYTES // 2  # 256 MiB: a multiple of the 8 MiB leaf
    views = [pin[:half], pin[half:]]
    np_views = [v.numpy() for v in views]
    if _side_stream is None:
        _side_stream = torch.cuda.Stream()
    _side_stream.wait_stream(torch.cuda.current_stream())
    events = [torch.cuda.Event(), torch.cuda.Event()]
    with torch.cuda.stream(_side_stream):
        n0 = min(half, nbytes)
        views[0][:n0].copy_(flat[:n0], non_blocking=True)
        events[0].record(_side_stream)
        offset, i = 0, 0
        while offset < nbytes:
            n = min(half, nbytes - offset)
            nxt = offset + n
            if nxt < nbytes:
                n2 = min(half, nbytes - nxt)
                views[i ^ 1][:n2].copy_(flat[nxt:nxt + n2],
                                        non_blocking=True)
                events[i ^ 1].record(_side_stream)
            events[i].synchronize()
            # NOTE: save_stream copies each 8 MiB leaf out of this view
            # before hashing, so reusing the half next iteration is safe
            yield memoryview(np_views[i][:n])
            offset = nxt
            i ^= 1


def _gpu_chunks_torch(t, torch):
    """Like _gpu_chunks but yields torch PINNED slices (for the native
    StreamSaver, which reads them GIL-free in C++)."""
    global _side_stream
    nbytes = t.element_size() * t.numel()
    flat = t.reshape(-1).view(torch.uint8)
    pin = _get_pin_buf(torch)
    half = _PIN_BUF_BYTES // 2
    views = [pin[:half], pin[half:]]
    if _side_stream is None:
        _side_stream = torch.cuda.Stream()
    _side_stream.wait_stream(torch.cuda.current_stream())
    events = [torch.cuda.Event(), torch.cuda.Event()]
    with torch.cuda.stream(_side_stream):
        n0 = min(half, nbytes)
        views[0][:n0].copy_(flat[:n0], non_blocking=True)
        events[0].record(_side_stream)
        offset, i = 0, 0
        while offset < nbytes:
            n = min(half, nbytes - offset)
            nxt = offset + n
            if nxt < nbytes:
                n2 = min(half, nbytes - nxt)
                views[i ^ 1][:n2].copy_(flat[nxt:nxt + n2],
                                        non_blocking=True)
                events[i ^ 1].record(_side_stream)
            events[i].synchronize()
            # feed() completes (hash + write) before the generator
            # resumes, so reusing this half next round is safe
            yield views[i][:n]
            offset = nxt
            i ^= 1


def _save_gpu_tensor_native(t, cas, torch):
    """GPU shard -> CAS with zero Python-side data work: D2H into
    pinned halves on the side stream while the C++ StreamSaver hashes
    Merkle leaves across threads and writes the previous half
    (cas_engine.cpp). Returns (sha, nbytes)."""
    import os as _os

    from ..datastore.cas import CODEC_RAW, MAGIC
    from ..ops import _mfx_cas

    t = t.detach().contiguous()
    nbytes = t.element_size() * t.numel()
    root_dir = cas._storage._abs(cas._prefix)
    _os.makedirs(root_dir, exist_ok=True)
    tmp = _os.path.join(root_dir, ".ckpt.%d.%x.tmp"
                        % (_os.getpid(), id(t)))
    header = MAGIC + bytes([1, CODEC_RAW, 0, 0])
    try:
        saver = _mfx_cas.StreamSaver(tmp, header, 0)
        for chunk in _gpu_chunks_torch(t, torch):
            saver.feed(chunk)
        key = saver.finish()
        ap = cas._storage._abs(cas._key_path(key))
        if _os.path.isfile(ap):
            _os.unlink(tmp)  # dedup
        else:
            _os.makedirs(_os.path.dirname(ap), exist_ok=True)
            _os.replace(tmp, ap)
        return key, nbytes
    except Exception:
        try:
            _os.unlink(tmp)
        except OSError:
            pass
        raise


def save_state_dict(task_datastore, state_dict, name="checkpoint"):
    """Persist a state dict; returns {tensor_name: {sha, dtype, shape}}.

    The save is pipelined at SHARD granularity: while shard i's
    hash+write runs on an IO thread (hashlib and os.write release the
    GIL), shard i+1's D2H staging proceeds on the main thread — the two
    stages measure ~10 GB/s each on an MI355X box
    (benchmarks/bench_ckpt_stages.py), so overlapping them is the whole
    config-4 win; finer-grained streaming (8 MiB leaves through a
    queue) measured SLOWER (~4.5 GB/s) from pure Python overhead."""
    import torch
    from concurrent.futures import ThreadPoolExecutor

    cas = task_datastore._ca_store
    index = {}
    other = {}
    pending = None  # (key, future, dtype, shape, nbytes)

    def _resolve(p):
        k, fut, dtype, shape, nbytes = p
        (_uri, sha), = fut.result()
        index[k] = {"sha": sha, "dtype": dtype, "shape": list(shape),
                    "nbytes": nbytes}

    can_native = hasattr(cas._storage, "_abs")
    if can_native:
        try:
            from ..ops import _mfx_cas  # noqa: F401
        except Exception:
            can_native = False

    with ThreadPoolExecutor(max_workers=1) as io_pool:
        for key, value in state_dict.items():
            if isinstance(value, torch.Tensor):
                if value.is_cuda and can_native:
                    # native path: C++ hashes+writes straight from the
                    # pinned halves (no host-side numpy copies at all);
                    # the pinned buffer is shared, so drain any pending
                    # io_pool shard first
                    if pending is not None:
                        _resolve(pending)
                        pending = None
                    sha, nbytes = _save_gpu_tensor_native(value, cas,
                                                          torch)
                    t = value
                    index[key] = {
                        "sha": sha,
                        "dtype": str(t.dtype).replace("torch.", ""),
                        "shape": list(t.shape), "nbytes": nbytes}
                    continue
                buf, dtype, shape = _tensor_to_buffer(value)
                if pending is not None:
                    _resolve(pending)
                pending = (key,
                           io_pool.submit(cas.save_blobs, [buf], True),
                           dtype, shape, len(buf))
            else:
                other[key] = value
        if pending is not None:
            _resolve(pending)
    task_datastore.save_artifacts([
        ("_checkpoint_%s_index" % name, index),
        ("_checkpoint_%s_meta" % name, other),
    ])
    task_datastore.save_metadata("checkpoint_%s" % name,
                                 {"tensors": len(index)})
    return index


def _load_file_to_device(path, off, nbytes, dtype, shape, device, torch):
    """Pipelined file -> pinned -> HBM load: readinto one pinned half
    while the other half's async H2D is in flight on the side stream."""
    global _side_stream
    dest = torch.empty(shape, dtype=dtype, device=device)
    flat = dest.reshape(-1).view(torch.uint8)
    pin = _get_pin_buf(torch)
    half = _PIN_BUF_BYTES // 2
    views = [pin[:half], pin[half:]]
    np_views = [memoryview(v.numpy()) for v in views]
    if _side_stream is None:
        _side_stream = torch.cuda.Stream()
    events = [torch.cuda.Event(), torch.cuda.Event()]
    recorded = [False, False]
    with open(path, "rb") as f:
        f.seek(off)
        offset, i = 0, 0
        with torch.cuda.stream(_side_stream):
            while offset < nbytes:
                n = min(half, nbytes - offset)
                if recorded[i]:
                    events[i].synchronize()  # pin half i free to reuse?
                got = f.readinto(np_views[i][:n])
                if got != n:
                    raise IOError("short read from %s" % path)
                flat[offset:offset + n].copy_(views[i][:n],
                                              non_blocking=True)
                events[i].record(_side_stream)
                recorded[i] = True
                offset += n
                i ^= 1
        _side_stream.synchronize()
    return dest


def _load_file_to_cpu(path, off, nbytes, dtype, shape, torch):
    """Single-copy CPU load: readinto a fresh writable buffer (the old
    bytes -> bytearray path copied twice)."""
    import numpy as np

    arr = np.empty(nbytes, dtype=np.uint8)
    with open(path, "rb") as f:
        f.seek(off)
        if f.readinto(memoryview(arr)) != nbytes:
            raise IOError("short read from %s" % path)
    return torch.from_numpy(arr).view(dtype).reshape(shape)


def load_state_dict(task_datastore, name="checkpoint", map_location="cpu"):
    import torch

    index = task_datastore["_checkpoint_%s_index" % name]
    meta = task_datastore.get("_checkpoint_%s_meta" % name, {})
    cas = task_datastore._ca_store
    out = dict(meta)
    sha_to_names = {}
    for key, info in index.items():
        sha_to_names.setdefault(info["sha"], []).append(key)
    import warnings

    to_cuda = str(map_location) != "cpu"
    slow_shas = []
    for sha, names in sha_to_names.items():
        loc = cas.blob_file(sha)
        if loc is None:
            slow_shas.append(sha)
            continue
        path, off = loc
        for key in names:
            info = index[key]
            dtype = getattr(torch, info["dtype"])
            nbytes = info["nbytes"]
            if to_cuda:
                out[key] = _load_file_to_device(
                    path, off, nbytes, dtype, info["shape"], map_location,
                    torch)
            else:
                out[key] = _load_file_to_cpu(path, off, nbytes, dtype,
                                             info["shape"], torch)
    for sha, blob in cas.load_blobs(slow_shas):
        for key in sha_to_names[sha]:
            info = index[key]
            dtype = getattr(torch, info["dtype"])
            if not to_cuda:
                # writable copy (restored tensors get trained on)
                t = torch.frombuffer(bytearray(blob), dtype=torch.uint8)
                t = t.view(dtype).reshape(info["shape"])
            else:
                # zero host copies: view the blob read-only, H2D is the
                # only copy
                with warnings.catch_warnings():
                    warnings.simplefilter("ignore")
                    t = torch.frombuffer(blob, dtype=torch.uint8)
                t = t.view(dtype).reshape(info["shape"]).to(map_location)
            out[key] = t
    return out
