"""Fire-and-forget sidecar subprocesses fed NDJSON over stdin.

Parity target: /root/reference/metaflow/sidecar/ (SidecarSubProcess :57,
worker dispatch, lossy/no-ack by design, auto-restart). Sidecars here run
``python -m metaflow_amd.sidecar <worker_name>`` and read one JSON message
per line; messages are best-effort (never block the task).

Built-in workers:
  heartbeat  — periodic run-liveness updates into the metadata provider
  gpu_monitor — samples rocm-smi/torch.cuda memory+utilization into a
                JSONL file next to the task logs (the rocprof-adjacent
                telemetry sidecar from SURVEY §2.5)
"""

import json
import os
import subprocess
import sys
import time


# Self-contained heartbeat loop for the LOCAL metadata provider: spawned
# as `python -S -c ...` so the child never imports metaflow_amd (the full
# package import costs ~0.3-0.5 s of CPU, which DOUBLED the cost of every
# small task — the config-2 fan-out is task-CPU bound, see
# profiles/bench_results_r02.md). Paths must mirror
# metadata/local.py:_heartbeat_path/_task_heartbeat_path.
_LOCAL_HB_SCRIPT = r"""
import json, os, select, sys, time
ctx = json.loads(os.environ["MFX_SIDECAR_CONTEXT"])
base = os.path.join(ctx["datastore_root"], ctx["flow_name"], "_meta",
                    str(ctx["run_id"]))
paths = [os.path.join(base, "heartbeat.json")]
if ctx.get("step_name") and ctx.get("task_id"):
    paths.append(os.path.join(base, "task_heartbeat_%s_%s.json"
                              % (ctx["step_name"], ctx["task_id"])))
os.makedirs(base, exist_ok=True)
while True:
    payload = json.dumps({"ts": time.time()}).encode()
    for p in paths:
        try:
            tmp = p + ".tmp%d" % os.getpid()
            with open(tmp, "wb") as f:
                f.write(payload)
            os.replace(tmp, p)
        except OSError:
            pass
    r, _w, _x = select.select([sys.stdin], [], [], 5.0)
    if r and not sys.stdin.buffer.read(1):
        break  # stdin EOF: parent gone / terminate()
"""


class SidecarSubProcess(object):
    def __init__(self, worker_name, context=None):
        self.worker_name = worker_name
        self.context = context or {}
        self._proc = None
        self._restarts = 0
        self._start()

    def _start(self):
        if self.worker_name == "heartbeat" and \
                self.context.get("provider") != "service":
            argv = [sys.executable, "-S", "-u", "-c", _LOCAL_HB_SCRIPT]
        else:
            argv = [sys.executable, "-m", "metaflow_amd.sidecar",
                    self.worker_name]
        try:
            self._proc = subprocess.Popen(
                argv,
                stdin=subprocess.PIPE,
                stdout=subprocess.DEVNULL,
                stderr=subprocess.DEVNULL,
                env=dict(os.environ,
                         MFX_SIDECAR_CONTEXT=json.dumps(self.context)),
            )
        except Exception:
            self._proc = None

    def send(self, msg_type, payload=None):
        """Lossy send; restarts a dead sidecar at most 3 times."""
        if self._proc is None:
            return
        if self._proc.poll() is not None:
            if self._restarts >= 3:
                return
            self._restarts += 1
            self._start()
            if self._proc is None:
                return
        try:
            line = json.dumps({"type": msg_type,
                               "payload": payload or {}}) + "\n"
            self._proc.stdin.write(line.encode())
            self._proc.stdin.flush()
        except Exception:
            pass

    def terminate(self):
        if self._proc is not None:
            self.send("shutdown")
            try:
                self._proc.stdin.close()
            except Exception:
                pass
            try:
                self._proc.wait(timeout=2)
            except Exception:
                self._proc.kill()


# -------------------------------------------------------------- workers
def _heartbeat_worker(context):
    """Write run heartbeats every ~10 s until stdin closes."""
    from .datastore.storage import LocalStorage

    flow = context["flow_name"]
    run_id = context["run_id"]
    root = context["datastore_root"]
    if context.get("provider") == "service":
        from .metadata.service import ServiceMetadataProvider

        meta = ServiceMetadataProvider(flow)
    else:
        from .metadata.local import LocalMetadataProvider

        meta = LocalMetadataProvider(flow, LocalStorage(root))
    step_name = context.get("step_name")
    task_id = context.get("task_id")

    def beat():
        try:
            meta.heartbeat(run_id, step_name, task_id)
        except TypeError:  # provider without task-level support
            meta.heartbeat(run_id)

    beat()
    import select

    while True:
        ready, _w, _x = select.select([sys.stdin], [], [], 10.0)
        if ready:
            line = sys.stdin.readline()
            if not line:
                break
            try:
                if json.loads(line).get("type") == "shutdown":
                    break
            except ValueError:
                pass
        beat()


def _gpu_monitor_worker(context):
    """Sample GPU memory/utilization into a JSONL file every ~5 s."""
    out_path = context.get("out_path", "/tmp/mfx_gpu_monitor.jsonl")
    import select

    def sample():
        rec = {"ts": time.time()}
        try:
            import torch

            if torch.cuda.is_available():
                free, total = torch.cuda.mem_get_info()
                rec["mem_used_gb"] = (total - free) / 1e9
                rec["mem_total_gb"] = total / 1e9
        except Exception:
            pass
        try:
            smi = subprocess.run(
                ["rocm-smi", "--showuse", "--json"],
                capture_output=True, text=True, timeout=5)
            if smi.returncode == 0:
                rec["rocm_smi"] = json.loads(smi.stdout)
        except Exception:
            pass
        with open(out_path, "a") as f:
            f.write(json.dumps(rec) + "\n")

    sample()  # immediate first sample: short tasks get telemetry too
    while True:
        ready, _w, _x = select.select([sys.stdin], [], [], 5.0)
        if ready:
            line = sys.stdin.readline()
            if not line:
                break
            try:
                if json.loads(line).get("type") == "shutdown":
                    break
            except ValueError:
                pass
        sample()


def _monitor_worker(context):
    """Append monitor measure/count/gauge messages to a JSONL file."""
    out_path = context.get("out_path", "/tmp/mfx_monitor.jsonl")
    while True:
        line = sys.stdin.readline()
        if not line:
            break
        try:
            msg = json.loads(line)
        except ValueError:
            continue
        if msg.get("type") == "shutdown":
            break
        with open(out_path, "a") as f:
            f.write(json.dumps(msg) + "\n")


WORKERS = {
    "heartbeat": _heartbeat_worker,
    "gpu_monitor": _gpu_monitor_worker,
    "monitor": _monitor_worker,
}


def main():
    worker = sys.argv[1]
    context = json.loads(os.environ.get("MFX_SIDECAR_CONTEXT", "{}"))
    fn = WORKERS.get(worker)
    if fn is None:
        sys.exit(2)
    try:
        fn(context)
    except KeyboardInterrupt:
        pass


if __name__ == "__main__":
    main()
