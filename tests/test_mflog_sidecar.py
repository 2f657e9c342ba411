"""mflog structured-log protocol + sidecar subprocess tests."""

import time

from metaflow_amd import mflog


def test_decorate_parse_roundtrip():
    line = mflog.decorate("step/1", "hello world", counter=7)
    p = mflog.parse(line)
    assert p.source == "step/1"
    assert p.counter == 7
    assert p.msg == "hello world"
    assert mflog.parse(b"not structured") is None


def test_decorate_stream():
    data = b"line1\nline2\n"
    out = mflog.decorate_stream("s/2", data)
    lines = out.splitlines()
    assert len(lines) == 2
    assert mflog.parse(lines[0]).msg == "line1"
    assert mflog.parse(lines[1]).counter == 1


def test_merge_logs_chronological():
    a = b"\n".join([
        mflog.decorate("a", "a0", 0, ts="2026-01-01T00:00:00.000000"),
        mflog.decorate("a", "a1", 1, ts="2026-01-01T00:00:02.000000"),
    ])
    b = b"\n".join([
        mflog.decorate("b", "b0", 0, ts="2026-01-01T00:00:01.000000"),
        mflog.decorate("b", "b1", 1, ts="2026-01-01T00:00:03.000000"),
    ])
    merged = [l.msg for l in mflog.merge_logs([a, b])]
    assert merged == ["a0", "b0", "a1", "b1"]


def test_heartbeat_sidecar(tmp_path):
    from metaflow_amd.datastore.storage import LocalStorage
    from metaflow_amd.metadata.local import LocalMetadataProvider
    from metaflow_amd.sidecar import SidecarSubProcess

    root = str(tmp_path)
    sc = SidecarSubProcess("heartbeat", {
        "flow_name": "F", "run_id": "123", "datastore_root": root})
    try:
        meta = LocalMetadataProvider("F", LocalStorage(root))
        deadline = time.time() + 15
        hb = None
        while time.time() < deadline:
            hb = meta._load(meta._heartbeat_path("123"))
            if hb:
                break
            time.sleep(0.3)
        assert hb and hb.get("ts"), "no heartbeat written"
    finally:
        sc.terminate()


def test_task_heartbeat_written(tmp_datastore):
    """Every task writes task-level heartbeats (reference: task+run
    liveness); a finished task reports is_alive False via the client."""
    import os

    from .test_runtime import latest_run_id, run_flow

    run_flow("linear_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "LinearFlow")
    meta_dir = os.path.join(tmp_datastore, "LinearFlow", "_meta", run_id)
    hb = [f for f in os.listdir(meta_dir)
          if f.startswith("task_heartbeat_")]
    # one per task attempt: start, middle..., end (>= 2 at minimum)
    assert len(hb) >= 2, hb

    os.environ["MFX_DATASTORE_SYSROOT_LOCAL"] = tmp_datastore
    import importlib

    import metaflow_amd.client as client

    importlib.reload(client)
    client.namespace(None)
    task = client.Run("LinearFlow/%s" % run_id)["start"].task
    assert task.finished and not task.is_alive
