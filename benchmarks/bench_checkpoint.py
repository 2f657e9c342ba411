#!/usr/bin/env python3
"""BASELINE config 4: @checkpoint save/load GB/s of big random-init shards.

    python benchmarks/bench_checkpoint.py [--gb 8] [--device cuda]

On a GPU this measures the pinned-buffer D2H staging + Merkle hash + CAS
write path per rank (the driver's 8-GPU run gives the x8 aggregate; the
per-rank number is the per-GPU shard bandwidth). The 70B x8 job is this
shard path with ~17.5 GB params + 70 GB Adam state per rank.
"""

import argparse
import json
import os
import sys
import tempfile
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gb", type=float, default=4.0)
    p.add_argument("--device", default=None)
    p.add_argument("--shard-mb", type=int, default=512)
    args = p.parse_args()

    import torch

    from metaflow_amd.datastore import FlowDataStore
    from metaflow_amd.datastore.storage import LocalStorage
    from metaflow_amd.parallel.checkpoint import (
        load_state_dict,
        save_state_dict,
    )

    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    tmp = tempfile.mkdtemp(prefix="mfx_ckpt_bench_")
    fds = FlowDataStore("CkptBench", LocalStorage(tmp))
    ds = fds.get_task_datastore("1", "train", "1", attempt=0, mode="w")
    ds.init_task()

    n_shards = max(1, int(args.gb * 1024 / args.shard_mb))
    elems = (args.shard_mb << 20) // 2  # bf16
    state = {
        "shard_%d" % i: torch.randn(elems, dtype=torch.bfloat16,
                                    device=device)
        for i in range(n_shards)
    }
    total_gb = n_shards * args.shard_mb / 1024
    if device.startswith("cuda"):
        torch.cuda.synchronize()

    t = time.time()
    save_state_dict(ds, state, name="bench")
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    save_s = time.time() - t

    # second save of identical content: dedup (content-addressed no-op)
    t = time.time()
    save_state_dict(ds, state, name="bench2")
    dedup_s = time.time() - t

    t = time.time()
    back = load_state_dict(ds, name="bench", map_location="cpu")
    load_s = time.time() - t
    assert len([k for k in back if k.startswith("shard_")]) == n_shards
    assert torch.equal(back["shard_0"],
                       state["shard_0"].cpu()), "cpu load corrupt"

    load_dev_s = None
    if device.startswith("cuda"):
        # the real restore path: CAS file -> pinned -> HBM, overlapped
        del back
        t = time.time()
        back = load_state_dict(ds, name="bench", map_location=device)
        torch.cuda.synchronize()
        load_dev_s = time.time() - t
        assert back["shard_0"].is_cuda
        assert torch.equal(back["shard_0"], state["shard_0"]), \
            "device load corrupt"

    ds.done()
    import subprocess

    subprocess.run(["rm", "-rf", tmp], check=False)
    print(json.dumps({
        "metric": "checkpoint GB/s",
        "value": total_gb / save_s,
        "unit": "GB/s",
        "higher_is_better": True,
        "save_gbps": total_gb / save_s,
        "dedup_save_gbps": total_gb / dedup_s,
        "load_gbps": total_gb / load_s,
        "load_to_gpu_gbps":
            (total_gb / load_dev_s) if load_dev_s else None,
        "config": {"gb": total_gb, "shards": n_shards, "device": device},
    }), flush=True)


if __name__ == "__main__":
    main()
