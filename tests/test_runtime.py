"""End-to-end runtime tests: run real flows as subprocesses (CPU)."""

import json
import os
import subprocess
import sys

import pytest

FLOWS = os.path.join(os.path.dirname(os.path.abspath(__file__)), "flows")
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_flow(flow_file, datastore_root, *args, check=True, timeout=180,
             env_extra=None):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    if env_extra:
        env.update(env_extra)
    cmd = [
        sys.executable, os.path.join(FLOWS, flow_file),
        "--datastore-root", datastore_root,
    ] + list(args)
    proc = subprocess.run(cmd, capture_output=True, text=True, env=env,
                          timeout=timeout)
    if check and proc.returncode != 0:
        raise AssertionError(
            "flow failed rc=%d\nstdout:\n%s\nstderr:\n%s"
            % (proc.returncode, proc.stdout, proc.stderr))
    return proc


def latest_run_id(datastore_root, flow_name):
    meta = os.path.join(datastore_root, flow_name, "_meta")
    runs = sorted(os.listdir(meta))
    return runs[-1]


def read_artifact(datastore_root, flow_name, run_id, step, name):
    """Read an artifact via the client API."""
    os.environ["MFX_DATASTORE_SYSROOT_LOCAL"] = datastore_root
    import importlib

    import metaflow_amd.client as client

    importlib.reload(client)
    client.namespace(None)
    task = client.Task("%s/%s/%s/%s" % (
        flow_name, run_id, step,
        next(iter(client.Step("%s/%s/%s" % (flow_name, run_id, step)))).id))
    return getattr(task.data, name)


def test_linear_flow(tmp_datastore):
    run_flow("linear_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "LinearFlow")
    assert read_artifact(tmp_datastore, "LinearFlow", run_id, "end",
                         "final") == 31


def test_linear_flow_with_param(tmp_datastore):
    run_flow("linear_flow.py", tmp_datastore, "run", "--alpha", "5")
    run_id = latest_run_id(tmp_datastore, "LinearFlow")
    assert read_artifact(tmp_datastore, "LinearFlow", run_id, "end",
                         "final") == 51


def test_branch_flow(tmp_datastore):
    run_flow("branch_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "BranchFlow")
    assert read_artifact(tmp_datastore, "BranchFlow", run_id, "join",
                         "total") == 3


def test_foreach_flow(tmp_datastore):
    run_flow("foreach_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "ForeachFlow")
    assert read_artifact(tmp_datastore, "ForeachFlow", run_id, "join",
                         "total") == 30


def test_parallel_flow(tmp_datastore):
    run_flow("parallel_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "ParallelFlow")
    assert read_artifact(tmp_datastore, "ParallelFlow", run_id, "join",
                         "ok") is True


def test_show_command(tmp_datastore):
    proc = run_flow("linear_flow.py", tmp_datastore, "show")
    assert "start" in proc.stdout
    assert "middle" in proc.stdout


def test_dump_command(tmp_datastore):
    run_flow("linear_flow.py", tmp_datastore, "run")
    run_id = latest_run_id(tmp_datastore, "LinearFlow")
    # find end task id via the datastore layout
    end_dir = os.path.join(tmp_datastore, "LinearFlow", run_id, "end")
    task_id = os.listdir(end_dir)[0]
    proc = run_flow("linear_flow.py", tmp_datastore, "dump",
                    "%s/end/%s" % (run_id, task_id))
    assert "final" in proc.stdout


def test_run_id_file(tmp_datastore, tmp_path):
    rid = tmp_path / "runid.txt"
    run_flow("linear_flow.py", tmp_datastore, "run",
             "--run-id-file", str(rid))
    assert rid.read_text() == latest_run_id(tmp_datastore, "LinearFlow")


def test_logs_and_dot_cli(tmp_datastore, tmp_path):
    """The `logs` and `output-dot` flow CLI commands."""
    flow = tmp_path / "echo_flow.py"
    flow.write_text(
        "from metaflow_amd import FlowSpec, step\n"
        "class EchoFlow(FlowSpec):\n"
        "    @step\n"
        "    def start(self):\n"
        "        print('log-marker-xyz')\n"
        "        self.next(self.end)\n"
        "    @step\n"
        "    def end(self):\n"
        "        pass\n"
        "if __name__ == '__main__':\n"
        "    EchoFlow()\n")
    import subprocess
    import sys

    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    run = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "run"],
        capture_output=True, text=True, env=env, timeout=180)
    assert run.returncode == 0
    run_id = latest_run_id(tmp_datastore, "EchoFlow")
    task_id = os.listdir(
        os.path.join(tmp_datastore, "EchoFlow", run_id, "start"))[0]
    logs = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "logs", "%s/start/%s" % (run_id, task_id)],
        capture_output=True, text=True, env=env, timeout=120)
    assert logs.returncode == 0
    assert "log-marker-xyz" in logs.stdout
    dot = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "output-dot"],
        capture_output=True, text=True, env=env, timeout=120)
    assert dot.returncode == 0
    assert "digraph" in dot.stdout and "start" in dot.stdout


def test_foreach_stack_values(tmp_datastore, tmp_path):
    """foreach_stack() returns the resolved input VALUE per nesting
    level (reference flowspec.py:654-684), not the variable name
    (advisor finding r1 #4)."""
    import subprocess
    import sys

    flow = tmp_path / "fstack_flow.py"
    flow.write_text(
        "from metaflow_amd import FlowSpec, step\n\n"
        "class FStackFlow(FlowSpec):\n"
        "    @step\n"
        "    def start(self):\n"
        "        self.letters = ['a', 'b']\n"
        "        self.next(self.outer, foreach='letters')\n\n"
        "    @step\n"
        "    def outer(self):\n"
        "        self.digits = [10, 20, 30]\n"
        "        self.next(self.inner, foreach='digits')\n\n"
        "    @step\n"
        "    def inner(self):\n"
        "        stack = self.foreach_stack()\n"
        "        assert len(stack) == 2, stack\n"
        "        (i0, n0, v0), (i1, n1, v1) = stack\n"
        "        assert n0 == 2 and v0 == ['a', 'b'][i0], stack\n"
        "        assert n1 == 3 and v1 == [10, 20, 30][i1], stack\n"
        "        assert v1 == self.input, stack\n"
        "        self.next(self.join_inner)\n\n"
        "    @step\n"
        "    def join_inner(self, inputs):\n"
        "        self.next(self.join_outer)\n\n"
        "    @step\n"
        "    def join_outer(self, inputs):\n"
        "        self.next(self.end)\n\n"
        "    @step\n"
        "    def end(self):\n"
        "        pass\n\n"
        "if __name__ == '__main__':\n"
        "    FStackFlow()\n")
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    proc = subprocess.run(
        [sys.executable, str(flow), "--datastore-root", tmp_datastore,
         "run"], env=env, capture_output=True, text=True, timeout=300)
    assert proc.returncode == 0, proc.stderr[-3000:]


def test_sched_timing_breakdown(tmp_path, tmp_datastore):
    """MFX_SCHED_TIMING=1 prints the scheduler's per-phase serial-time
    breakdown at run end (launch / exit / exit.logs)."""
    import os
    import subprocess
    import sys
    import textwrap

    flow = tmp_path / "timing_flow.py"
    flow.write_text(textwrap.dedent("""
        from metaflow_amd import FlowSpec, step

        class TFlow(FlowSpec):
            @step
            def start(self):
                self.items = [0, 1, 2]
                self.next(self.work, foreach="items")

            @step
            def work(self):
                self.v = self.input
                self.next(self.join)

            @step
            def join(self, inputs):
                self.next(self.end)

            @step
            def end(self):
                pass

        if __name__ == "__main__":
            TFlow()
    """))
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    env["MFX_SCHED_TIMING"] = "1"
    proc = subprocess.run(
        [sys.executable, str(flow), "--quiet", "--datastore-root",
         tmp_datastore, "run"],
        env=env, capture_output=True, text=True, timeout=300)
    assert proc.returncode == 0, proc.stderr[-2000:]
    out = proc.stdout
    for phase in ("launch", "exit", "exit.logs"):
        assert "[mfx-sched-timing] %s" % phase in out, out[-1500:]
