#!/usr/bin/env python3
"""Stage-by-stage timing of the checkpoint save pipeline to find the
actual bottleneck: D2H, leaf copy, pool hashing, file write, and the
composed save_stream."""

import hashlib
import os
import sys
import tempfile
import time
from concurrent.futures import ThreadPoolExecutor

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

GB = 1 << 30


def main():
    import numpy as np
    import torch

    n = 2 * GB
    print("cpus:", os.cpu_count(), flush=True)

    # 1. single-thread sha256
    buf = np.random.bytes(256 << 20)
    t = time.time()
    hashlib.sha256(buf).digest()
    st = (256 << 20) / (time.time() - t) / GB
    print("sha256 single-thread: %.2f GB/s" % st, flush=True)

    # 2. pool hash of 8 MiB leaves
    pool = ThreadPoolExecutor(max_workers=os.cpu_count())
    big = np.random.bytes(n)
    mv = memoryview(big)
    t = time.time()
    futs = [pool.submit(lambda b: hashlib.sha256(b).digest(),
                        bytes(mv[o:o + (8 << 20)]))
            for o in range(0, n, 8 << 20)]
    [f.result() for f in futs]
    print("pool leaf hash (+copy): %.2f GB/s" % (n / (time.time() - t)
                                                 / GB), flush=True)

    # 3. leaf copies alone
    t = time.time()
    for o in range(0, n, 8 << 20):
        bytes(mv[o:o + (8 << 20)])
    print("leaf copies: %.2f GB/s" % (n / (time.time() - t) / GB),
          flush=True)

    # 4. file write
    tmp = tempfile.mktemp(prefix="ckpt_stage_")
    t = time.time()
    with open(tmp, "wb") as f:
        for o in range(0, n, 8 << 20):
            f.write(mv[o:o + (8 << 20)])
    print("file write: %.2f GB/s" % (n / (time.time() - t) / GB),
          flush=True)
    os.unlink(tmp)

    if torch.cuda.is_available():
        dev = torch.device("cuda")
        t_gpu = torch.randn(n // 2, dtype=torch.bfloat16, device=dev)
        pin = torch.empty(512 << 20, dtype=torch.uint8, pin_memory=True)
        halves = [pin[:256 << 20], pin[256 << 20:]]
        flat = t_gpu.reshape(-1).view(torch.uint8)
        side = torch.cuda.Stream()
        torch.cuda.synchronize()
        t = time.time()
        with torch.cuda.stream(side):
            off = 0
            i = 0
            while off < n:
                c = min(256 << 20, n - off)
                halves[i][:c].copy_(flat[off:off + c], non_blocking=True)
                off += c
                i ^= 1
        torch.cuda.synchronize()
        print("pure D2H (alternating pinned halves): %.2f GB/s"
              % (n / (time.time() - t) / GB), flush=True)

        # composed save_stream from GPU
        from metaflow_amd.datastore.cas import ContentAddressedStore
        from metaflow_amd.datastore.storage import LocalStorage
        from metaflow_amd.parallel.checkpoint import _gpu_chunks

        d = tempfile.mkdtemp(prefix="ckpt_stage_cas_")
        store = ContentAddressedStore("data", LocalStorage(d))
        torch.cuda.synchronize()
        t = time.time()
        store.save_stream(_gpu_chunks(t_gpu, torch), n)
        print("save_stream(GPU tensor): %.2f GB/s"
              % (n / (time.time() - t) / GB), flush=True)
        import shutil

        shutil.rmtree(d)


if __name__ == "__main__" and "--probe-pinned" not in sys.argv:
    main()


def probe_pinned():
    """Is reading TORCH-PINNED host memory slow (uncached mapping)?"""
    import numpy as np
    import torch

    n = 1 << 30
    pin = torch.empty(n, dtype=torch.uint8, pin_memory=True)
    pin_np = pin.numpy()
    pin_np[::4096] = 1  # fault in
    page = np.empty(n, dtype=np.uint8)

    t = time.time()
    for o in range(0, n, 8 << 20):
        bytes(memoryview(pin_np)[o:o + (8 << 20)])
    print("leaf copies FROM PINNED: %.2f GB/s"
          % (n / (time.time() - t) / GB), flush=True)

    t = time.time()
    np.copyto(page, pin_np)
    print("np.copyto pinned->pageable: %.2f GB/s"
          % (n / (time.time() - t) / GB), flush=True)

    cpu_t = torch.from_numpy(page)
    t = time.time()
    cpu_t.copy_(pin)
    print("torch copy pinned->pageable: %.2f GB/s"
          % (n / (time.time() - t) / GB), flush=True)

    t = time.time()
    h = hashlib.sha256()
    for o in range(0, 256 << 20, 8 << 20):
        h.update(memoryview(pin_np)[o:o + (8 << 20)])
    print("sha256 straight from pinned: %.2f GB/s"
          % ((256 << 20) / (time.time() - t) / GB), flush=True)


if __name__ == "__main__" and "--probe-pinned" in sys.argv:
    probe_pinned()


def probe_compose():
    """Decompose save_stream overhead: host-source streaming vs the old
    buffer-then-save path, with and without the GPU in the loop."""
    import shutil

    import numpy as np
    import torch

    from metaflow_amd.datastore.cas import ContentAddressedStore
    from metaflow_amd.datastore.storage import LocalStorage

    n = 2 * GB
    big = np.random.bytes(n)
    chunks = [memoryview(big)[o:o + (256 << 20)]
              for o in range(0, n, 256 << 20)]

    d = tempfile.mkdtemp(prefix="ckpt_cmp_")
    store = ContentAddressedStore("data", LocalStorage(d))
    t = time.time()
    store.save_stream(iter(list(chunks)), n)
    print("save_stream(host chunks): %.2f GB/s"
          % (n / (time.time() - t) / GB), flush=True)
    shutil.rmtree(d)

    d = tempfile.mkdtemp(prefix="ckpt_cmp2_")
    store = ContentAddressedStore("data", LocalStorage(d))
    t = time.time()
    store.save_blobs([big], raw=True)
    print("save_blobs(host, old path): %.2f GB/s"
          % (n / (time.time() - t) / GB), flush=True)
    shutil.rmtree(d)

    if torch.cuda.is_available():
        from metaflow_amd.parallel import checkpoint as CK

        t_gpu = torch.randn(n // 2, dtype=torch.bfloat16, device="cuda")
        torch.cuda.synchronize()
        t = time.time()
        buf, _d, _s = CK._tensor_to_buffer(t_gpu)
        print("_tensor_to_buffer(D2H serial): %.2f GB/s"
              % (n / (time.time() - t) / GB), flush=True)
        d = tempfile.mkdtemp(prefix="ckpt_cmp3_")
        store = ContentAddressedStore("data", LocalStorage(d))
        t = time.time()
        store.save_blobs([buf], raw=True)
        print("save_blobs(of that buffer): %.2f GB/s"
              % (n / (time.time() - t) / GB), flush=True)
        shutil.rmtree(d)
        # chunk generator alone (no hash/write): drain _gpu_chunks
        t = time.time()
        total = 0
        for mv in CK._gpu_chunks(t_gpu, torch):
            total += len(mv)
        print("_gpu_chunks drain only: %.2f GB/s"
              % (total / (time.time() - t) / GB), flush=True)


if __name__ == "__main__" and "--probe-compose" in sys.argv:
    probe_compose()
