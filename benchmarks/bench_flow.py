#!/usr/bin/env python3
"""BASELINE config 1: 3-step LinearFlow wall time on the local runtime +
filesystem datastore, CPU only (pure engine plumbing: scheduler,
subprocess tasks, CAS artifacts, metadata, logs).

    python benchmarks/bench_flow.py [--runs 5]
"""

import argparse
import json
import os
import subprocess
import sys
import tempfile
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

FLOW = """
from metaflow_amd import FlowSpec, step

class Linear3(FlowSpec):
    @step
    def start(self):
        self.x = 1
        self.next(self.middle)

    @step
    def middle(self):
        self.x += 1
        self.next(self.end)

    @step
    def end(self):
        assert self.x == 2

if __name__ == "__main__":
    Linear3()
"""


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--runs", type=int, default=5)
    args = p.parse_args()

    tmp = tempfile.mkdtemp(prefix="mfx_flow_bench_")
    flow_file = os.path.join(tmp, "linear3.py")
    with open(flow_file, "w") as f:
        f.write(FLOW)
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"

    times = []
    for i in range(args.runs + 1):  # first run is warmup (imports)
        t = time.time()
        proc = subprocess.run(
            [sys.executable, flow_file, "--quiet", "--datastore-root",
             os.path.join(tmp, "ds"), "run"],
            capture_output=True, text=True, env=env, timeout=300)
        dt = time.time() - t
        assert proc.returncode == 0, proc.stderr[-2000:]
        if i > 0:
            times.append(dt)

    subprocess.run(["rm", "-rf", tmp], check=False)
    best = min(times)
    print(json.dumps({
        "metric": "LinearFlow wall seconds",
        "value": best,
        "unit": "s",
        "higher_is_better": False,
        "mean_s": sum(times) / len(times),
        "best_s": best,
        "runs": args.runs,
        "tasks_per_run": 3,
        "config": {"flow": "3-step linear", "datastore": "local fs"},
    }), flush=True)


if __name__ == "__main__":
    main()
