"""HTTP metadata service + provider round-trip, and a full flow run with
--metadata service."""

import os
import socket
import subprocess
import sys
import time

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture
def service(tmp_path):
    port = _free_port()
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    proc = subprocess.Popen(
        [sys.executable, "-m", "metaflow_amd.metadata.service",
         "--root", str(tmp_path / "meta_root"), "--port", str(port)],
        env=env, stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    url = "http://127.0.0.1:%d" % port
    # wait for readiness
    import urllib.request

    for _ in range(100):
        try:
            urllib.request.urlopen(url + "/version", timeout=1)
            break
        except Exception:
            time.sleep(0.1)
    else:
        proc.kill()
        raise RuntimeError("service did not start")
    yield url
    proc.terminate()
    proc.wait(timeout=5)


def test_provider_roundtrip(service):
    from metaflow_amd.metadata.service import ServiceMetadataProvider

    p = ServiceMetadataProvider("TestFlow", url=service)
    run_id = p.new_run_id(tags=["exp:1"])
    assert run_id
    p.register_task(run_id, "start", "1", 0, {"hello": 1})
    p.heartbeat(run_id)
    p.add_run_tags(run_id, ["extra"])
    info = p.get_run(run_id)
    assert "exp:1" in info["tags"] and "extra" in info["tags"]
    tasks = p.list_tasks(run_id)
    assert tasks and tasks[0]["step_name"] == "start"
    p.register_run_done(run_id, True)
    assert p.get_run(run_id)["status"] == "successful"
    assert p.get_task(run_id, "start", "1")["task_id"] == "1"


def test_flow_run_with_service_metadata(service, tmp_datastore,
                                        monkeypatch):
    from .test_runtime import run_flow

    monkeypatch.setenv("MFX_SERVICE_URL", service)
    run_flow("linear_flow.py", tmp_datastore, "--metadata", "service",
             "run")
    from metaflow_amd.metadata.service import ServiceMetadataProvider

    p = ServiceMetadataProvider("LinearFlow", url=service)
    runs = p.list_runs()
    assert runs, "service has no runs registered"
    run_id = runs[0]["run_id"]
    info = p.get_run(run_id)
    assert info and info["status"] == "successful"
    assert len(p.list_tasks(run_id)) >= 3
    # heartbeat sidecar went through the service provider too
    assert p._request("GET", "/version")["api_version"] == 1


def test_local_metadata_paging(tmp_path):
    """iter_runs(limit=N) loads only N run records, newest first, with
    ordering derived from directory names (no full-history scan)."""
    from metaflow_amd.datastore.storage import LocalStorage
    from metaflow_amd.metadata.local import LocalMetadataProvider

    meta = LocalMetadataProvider("PagedFlow", LocalStorage(str(tmp_path)))
    ids = []
    for _ in range(25):
        rid = meta.new_run_id()
        meta.register_run(rid)
        ids.append(rid)
    page = list(meta.iter_runs(limit=5))
    assert len(page) == 5
    assert [r["run_id"] for r in page] == sorted(ids, reverse=True)[:5]
    # resume-prefixed ids sort with their origin's timestamp
    meta.register_run("resume%s" % ids[0])
    newest = next(iter(meta.iter_runs(limit=1)))
    assert newest["run_id"] in ("resume%s" % ids[0], ids[-1])
    assert len(meta.list_runs()) == 26
