"""Monitor/event-logger singletons and metaflow_amd_extensions plugin
discovery."""

import json
import os
import subprocess
import sys
import textwrap

from .test_runtime import FLOWS, REPO, run_flow


def test_monitor_null_default():
    from metaflow_amd.monitor import NullMonitor, get_system_monitor

    mon = get_system_monitor()
    assert isinstance(mon, NullMonitor)
    with mon.measure("x"), mon.count("y"):
        pass  # no-ops must not raise


def test_debug_monitor_emits(capsys_dummy=None):
    from metaflow_amd.monitor import DebugEventLogger, DebugMonitor

    mon = DebugMonitor()
    with mon.measure("block"):
        pass
    mon.gauge("g", 3)
    DebugEventLogger().log({"event": "x"})


def test_sidecar_monitor_writes(tmp_path):
    """measure/count through the monitor sidecar land in the JSONL out."""
    import time

    from metaflow_amd.monitor import SidecarMonitor

    out = tmp_path / "mon.jsonl"
    os.environ["MFX_MONITOR_OUT"] = str(out)
    try:
        mon = SidecarMonitor()
        mon.init_environment()
        with mon.measure("unit.block"):
            pass
        with mon.count("unit.runs"):
            pass
        deadline = time.time() + 15
        while time.time() < deadline:
            if out.exists() and len(out.read_text().splitlines()) >= 2:
                break
            time.sleep(0.2)
        mon.terminate()
        lines = [json.loads(l) for l in out.read_text().splitlines()]
        kinds = {l["type"] for l in lines}
        assert "measure" in kinds and "count" in kinds, lines
        names = {l["payload"]["name"] for l in lines}
        assert {"unit.block", "unit.runs"} <= names
    finally:
        os.environ.pop("MFX_MONITOR_OUT", None)


def test_task_event_logger_fires(tmp_datastore):
    """MFX_EVENT_LOGGER=debug: task subprocesses log task_start events."""
    proc = run_flow("linear_flow.py", tmp_datastore, "run",
                    env_extra={"MFX_EVENT_LOGGER": "debug"})
    assert "task_start" in proc.stderr
    assert '"step": "start"' in proc.stderr


def test_extension_package_discovery(tmp_path, tmp_datastore):
    """A metaflow_amd_extensions package on PYTHONPATH contributes a step
    decorator usable via --with."""
    ext = tmp_path / "metaflow_amd_extensions" / "myext"
    ext.mkdir(parents=True)
    # namespace package: no __init__.py at the top level
    (ext / "__init__.py").write_text(textwrap.dedent("""
        from metaflow_amd.decorators import StepDecorator

        class StampDecorator(StepDecorator):
            name = "stamp"

            def task_pre_step(self, step_name, task_datastore, metadata,
                              run_id, task_id, flow, graph, retry_count,
                              max_user_code_retries, ubf_context, inputs):
                flow.stamped = "by-extension"

        STEP_DECORATORS = [StampDecorator]
    """))
    env = dict(os.environ)
    env["PYTHONPATH"] = os.pathsep.join(
        [str(tmp_path), REPO, env.get("PYTHONPATH", "")])
    env["MFX_NUM_GPUS"] = "0"
    proc = subprocess.run(
        [sys.executable, os.path.join(FLOWS, "linear_flow.py"), "--quiet",
         "--datastore-root", tmp_datastore, "--with", "stamp", "run"],
        env=env, capture_output=True, text=True, timeout=300)
    assert proc.returncode == 0, proc.stderr[-2000:]

    from .test_runtime import latest_run_id, read_artifact

    run_id = latest_run_id(tmp_datastore, "LinearFlow")
    assert read_artifact(tmp_datastore, "LinearFlow", run_id, "start",
                         "stamped") == "by-extension"


def test_system_context_phase(tmp_datastore):
    """Telemetry records carry the process's execution phase (reference
    system_context.py): task subprocesses log phase=task."""
    proc = run_flow("linear_flow.py", tmp_datastore, "run",
                    env_extra={"MFX_EVENT_LOGGER": "debug"})
    assert '"phase": "task"' in proc.stderr

    from metaflow_amd import system_context as sc

    assert sc.phase_from_subcommand("run") == sc.SCHEDULER
    assert sc.phase_from_subcommand("step") == sc.TASK
    assert sc.phase_from_subcommand("dump") == sc.CLIENT
