"""Concurrency and robustness: simultaneous runs in one datastore, CAS
concurrent writers, status CLI, Runner.resume, gpu_monitor sidecar."""

import os
import subprocess
import sys
import time

from .test_runtime import FLOWS, REPO, latest_run_id, run_flow


def test_concurrent_runs_share_datastore(tmp_datastore):
    """Two flows running at once in the same datastore root must not
    interfere (µs run ids, write-once CAS, per-run metadata)."""
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"

    def launch(flow_file):
        return subprocess.Popen(
            [sys.executable, os.path.join(FLOWS, flow_file), "--quiet",
             "--datastore-root", tmp_datastore, "run"],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
            text=True)

    procs = [launch("linear_flow.py"), launch("foreach_flow.py"),
             launch("branch_flow.py")]
    for p in procs:
        out, err = p.communicate(timeout=300)
        assert p.returncode == 0, err[-1500:]
    # all three flows recorded successful runs
    for flow in ("LinearFlow", "ForeachFlow", "BranchFlow"):
        assert os.path.isdir(os.path.join(tmp_datastore, flow, "_meta"))


def test_cas_concurrent_same_blob(tmp_path):
    """Many threads saving the SAME content concurrently: one physical
    blob, no corruption (write-once + atomic rename)."""
    from concurrent.futures import ThreadPoolExecutor

    from metaflow_amd.datastore.cas import ContentAddressedStore
    from metaflow_amd.datastore.storage import LocalStorage

    cas = ContentAddressedStore("data", LocalStorage(str(tmp_path)))
    blob = os.urandom(1 << 20)

    def save(_):
        return cas.save_blobs([blob])[0][1]

    with ThreadPoolExecutor(8) as pool:
        keys = list(pool.map(save, range(16)))
    assert len(set(keys)) == 1
    [(k, data)] = list(cas.load_blobs([keys[0]]))
    assert data == blob


def test_status_cli(tmp_datastore):
    run_flow("linear_flow.py", tmp_datastore, "run")
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_DATASTORE_SYSROOT_LOCAL"] = tmp_datastore
    proc = subprocess.run(
        [sys.executable, "-m", "metaflow_amd", "status"],
        env=env, capture_output=True, text=True, timeout=60)
    assert proc.returncode == 0
    assert "LinearFlow" in proc.stdout
    runs = subprocess.run(
        [sys.executable, "-m", "metaflow_amd", "runs", "LinearFlow"],
        env=env, capture_output=True, text=True, timeout=60)
    assert "ok" in runs.stdout


def test_runner_resume(tmp_datastore, tmp_path, monkeypatch):
    monkeypatch.setenv("RESUME_COUNTER_DIR", str(tmp_path))
    monkeypatch.setenv("RESUME_FAIL", "1")
    monkeypatch.setenv("MFX_NUM_GPUS", "0")
    monkeypatch.setenv(
        "PYTHONPATH", REPO + os.pathsep + os.environ.get("PYTHONPATH", ""))
    from metaflow_amd.exceptions import MFXException
    from metaflow_amd.runner import Runner

    r = Runner(os.path.join(FLOWS, "resume_flow.py"),
               datastore_root=tmp_datastore)
    try:
        r.run()
        raise AssertionError("expected failure")
    except MFXException:
        pass
    orig = latest_run_id(tmp_datastore, "ResumeFlow")
    monkeypatch.setenv("RESUME_FAIL", "0")
    ex = r.resume(orig)
    assert ex.returncode == 0


def test_gpu_monitor_sidecar(tmp_path):
    from metaflow_amd.sidecar import SidecarSubProcess

    out = tmp_path / "mon.jsonl"
    sc = SidecarSubProcess("gpu_monitor", {"out_path": str(out)})
    try:
        deadline = time.time() + 20
        while time.time() < deadline and not out.exists():
            time.sleep(0.3)
        assert out.exists(), "gpu monitor wrote no samples"
    finally:
        sc.terminate()


def test_mfx_card_cli(tmp_datastore, tmp_path):
    """`python -m metaflow_amd card <pathspec>` writes the task's HTML."""
    run_flow("card_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "CardFlow")
    task_id = os.listdir(
        os.path.join(tmp_datastore, "CardFlow", run_id, "start"))[0]
    out = tmp_path / "card.html"
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_DATASTORE_SYSROOT_LOCAL"] = tmp_datastore
    proc = subprocess.run(
        [sys.executable, "-m", "metaflow_amd", "card",
         "CardFlow/%s/start/%s" % (run_id, task_id), "--out", str(out)],
        env=env, capture_output=True, text=True, timeout=120)
    assert proc.returncode == 0, proc.stderr[-1000:]
    assert "custom html" in out.read_text()
